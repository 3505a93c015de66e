"""Property-based invariants (hypothesis): BlockManager accounting under
random op sequences, flow-queue ordering, parser robustness on arbitrary
bytes. ADVICE-tier hardening: these are the data structures whose
accounting bugs would corrupt KV or strand requests silently."""
import os
import string

from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from llm_d_inference_scheduler_amd.engine.kvcache import (BlockManager,
                                                          block_hashes)

# DEEP_EXAMPLES=1000 turns the suite into a long fuzz campaign
_N = int(os.environ.get("DEEP_EXAMPLES", "60"))
SMALL = settings(max_examples=_N, deadline=None,
                 suppress_health_check=[HealthCheck.too_slow])


@st.composite
def op_sequences(draw):
    n_ops = draw(st.integers(10, 60))
    ops = []
    for i in range(n_ops):
        kind = draw(st.sampled_from(["prompt", "grow", "free", "take"]))
        ops.append((kind, draw(st.integers(0, 7)),
                    draw(st.integers(1, 70))))
    return ops


class TestBlockManagerInvariants:
    @SMALL
    @given(op_sequences(), st.integers(8, 64))
    def test_accounting_never_breaks(self, ops, num_blocks):
        mgr = BlockManager(num_blocks)
        live = {}          # seq -> token count requested
        taken = []
        prompts = {s: list(range(10 + s, 10 + s + 64)) for s in range(8)}
        for kind, s, n in ops:
            sid = f"s{s}"
            if kind == "prompt" and sid not in mgr.tables:
                h = block_hashes(prompts[s][:n])
                matched = mgr.allocate_prompt(sid, h, max(1, n))
                assert matched <= max(0, n - 1)
                if not mgr.allocate(sid, n):
                    mgr.free(sid)
                    continue
                live[sid] = n
                full = n // 16
                for b in range(full):
                    mgr.register_block(sid, b, int(h[b]))
            elif kind == "grow" and sid in mgr.tables:
                cur = live.get(sid, 0)
                if mgr.allocate(sid, cur + n):
                    live[sid] = cur + n
            elif kind == "free" and sid in mgr.tables:
                mgr.free(sid)
                live.pop(sid, None)
            elif kind == "take":
                got = mgr.take_blocks(min(n, 4))
                if got is not None:
                    taken.extend(got)
                    if len(taken) > 8:
                        mgr.release_blocks(taken)
                        taken = []
            # --- invariants after every op ---
            allocated = set()
            for t in mgr.tables.values():
                for blk in t:
                    allocated.add(blk)
            # a block is never in two tables unless refcounted > 1
            refsum = sum(mgr._refcnt)
            table_refs = sum(len(t) for t in mgr.tables.values())
            assert refsum == table_refs + len(taken)
            assert 0 <= mgr.free_blocks <= num_blocks
            for blk in allocated:
                assert mgr._refcnt[blk] >= 1
                assert blk not in mgr._free_lru
        for sid in list(mgr.tables):
            mgr.free(sid)
        mgr.release_blocks(taken)
        assert mgr.free_blocks == num_blocks

    @SMALL
    @given(st.lists(st.integers(0, 2 ** 31 - 1), min_size=0, max_size=100))
    def test_block_hashes_deterministic_prefix_property(self, tokens):
        h1 = block_hashes(tokens)
        h2 = block_hashes(tokens)
        assert list(h1) == list(h2)
        if len(tokens) >= 32:
            # chained property: a longer prompt with identical prefix
            # shares exactly the leading block hashes
            h3 = block_hashes(tokens + [1, 2, 3, 4])
            assert list(h3[:len(h1)]) == list(h1)


class TestQueueProperties:
    @SMALL
    @given(st.lists(st.tuples(st.integers(0, 1 << 30), st.integers(1, 999)),
                    min_size=1, max_size=80))
    def test_maxminheap_orders_by_key(self, items):
        from llm_d_inference_scheduler_amd import _router_core as rc
        h = rc.MaxMinHeap()
        for i, (key, _) in enumerate(items):
            h.push(i, float(key))
        keys = []
        for _ in range(len(items)):
            popped = h.pop()        # (id, key, bytes) of the min-key item
            assert popped is not None
            keys.append(float(items[int(popped[0])][0]))
        assert keys == sorted(keys)
        assert h.pop() is None

    @SMALL
    @given(st.lists(st.integers(0, 1 << 20), min_size=1, max_size=60))
    def test_listqueue_fifo(self, ids):
        from llm_d_inference_scheduler_amd import _router_core as rc
        q = rc.ListQueue()
        for pos, i in enumerate(ids):
            q.push(pos, 0.0, i)
        out = [int(q.pop()[0]) for _ in range(len(ids))]
        assert out == list(range(len(ids)))     # strict FIFO by insertion
        assert q.pop() is None


class TestParserFuzz:
    @SMALL
    @given(st.binary(min_size=0, max_size=300))
    def test_openai_parser_never_crashes(self, body):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            OpenAIParser
        res = OpenAIParser().parse_request(body, {}, "/v1/completions")
        assert res.error is not None or res.request is not None or res.skip

    @SMALL
    @given(st.binary(min_size=0, max_size=300))
    def test_vllm_grpc_parser_never_crashes(self, body):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VllmGrpcParser
        res = VllmGrpcParser().parse_request(
            body, {}, "/vllm.VllmEngine/Generate")
        assert res.error is not None or res.request is not None or res.skip

    @SMALL
    @given(st.text(alphabet=string.printable, max_size=200))
    def test_vertexai_parser_never_crashes(self, text):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VertexAIParser
        res = VertexAIParser().parse_request(text.encode(), {}, "/v1")
        assert res.error is not None or res.request is not None or res.skip


class TestKVBlockIndexInvariants:
    @settings(max_examples=_N, deadline=None)
    @given(st.lists(st.tuples(
        st.sampled_from(["store", "evict", "spec", "remove"]),
        st.integers(0, 2),                     # endpoint
        st.lists(st.integers(0, 30), max_size=6)), max_size=60))
    def test_size_and_match_consistency(self, events):
        """Index size equals the number of live (endpoint, hash) confirmed
        entries, and match_longest never reports beyond its inputs."""
        from llm_d_inference_scheduler_amd.datalayer.kvblock import \
            KVBlockIndex
        ix = KVBlockIndex(speculative_ttl_s=60.0)
        live = set()
        for kind, ep_i, hashes in events:
            ep = f"gpu{ep_i}"
            if kind == "store":
                ix.apply_events(ep, hashes, [])
                live |= {(ep, h) for h in hashes}
            elif kind == "evict":
                ix.apply_events(ep, [], hashes)
                live -= {(ep, h) for h in hashes}
            elif kind == "spec":
                ix.add_speculative(ep, hashes)
            else:
                ix.remove_endpoint(ep)
                live = {(e, h) for e, h in live if e != ep}
        assert ix.size == len({h for _, h in live})
        m = ix.match_longest([1, 2, 3], ["gpu0", "gpu1", "gpu2"])
        for ep, n in m.items():
            assert 0 <= n <= 3
            for h in [1, 2, 3][:n]:
                # confirmed or speculative coverage must actually exist
                assert (ep, h) in live or (h, ep) in ix._spec


class TestEngineLifecycleProperty:
    @settings(max_examples=max(10, _N // 6), deadline=None,
              suppress_health_check=[HealthCheck.too_slow])
    @given(st.lists(st.tuples(st.integers(1, 60),      # prompt len
                              st.integers(1, 6),       # max_tokens
                              st.booleans(),           # abort it mid-flight?
                              st.integers(-1, 10)),    # priority
                    min_size=1, max_size=5),
           st.integers(6, 24))                          # pool blocks
    def test_requests_always_terminate(self, reqs, blocks):
        """Every submitted request either completes or is aborted — no
        stranded sequences, and the pool leaks no blocks (vLLM-style
        invariant over random shapes incl. pool churn, exhaustion
        preemption AND priority-driven early preemption)."""
        import torch
        from llm_d_inference_scheduler_amd.engine import (EngineRequest,
                                                          EngineWorker)
        from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
        w = EngineWorker(TINY_LLAMA, "cpu", kv_blocks=blocks,
                         dtype=torch.float32)
        done = set()
        for i, (plen, mx, abort, prio) in enumerate(reqs):
            w.add_request(EngineRequest(f"r{i}", list(range(plen)),
                                        max_tokens=mx, priority=prio))
        aborted = {f"r{i}" for i, (_, _, a, _) in enumerate(reqs) if a}
        stepped = 0
        while w.has_work and stepped < 400:
            if stepped == 2 and aborted:
                for rid in aborted:
                    w.abort(rid)
            for out in w.step():
                if out.finished:
                    done.add(out.request_id)
            stepped += 1
        assert stepped < 400, "engine failed to drain"
        assert done | aborted >= {f"r{i}" for i in range(len(reqs))}
        assert not w.mgr.tables, "leaked sequence tables"


class TestFlowControlInvariants:
    @SMALL
    @given(st.lists(st.tuples(
            st.sampled_from(["submit", "tick", "tick_blocked"]),
            st.integers(-1, 10),                   # priority
            st.integers(1, 50),                    # byte size
            st.integers(0, 2)),                    # flow
            min_size=1, max_size=120))
    def test_every_item_reaches_one_terminal_state(self, ops):
        """Random submit/tick/blocked-tick sequences: items end
        dispatched/rejected/displaced exactly once, dispatched callbacks
        match DISPATCHED outcomes, and the queue drains once dispatch
        unblocks (processor.go exactly-once finalization invariants)."""
        from llm_d_inference_scheduler_amd.flowcontrol import (
            BandConfig, FlowController, FlowControlRequest, FlowRegistry,
            QueueOutcome)
        from llm_d_inference_scheduler_amd.scheduling.types import \
            LLMRequest
        reg = FlowRegistry(bands=[BandConfig(10), BandConfig(0),
                                  BandConfig(-1, max_items=6)],
                           global_max_bytes=600)
        dispatched = []
        state = {"ok": True}
        fc = FlowController(reg, lambda item: state["ok"] and
                            (dispatched.append(item) or True))
        items = []
        for i, (kind, prio, size, flow) in enumerate(ops):
            if kind == "submit":
                it = FlowControlRequest(
                    request=LLMRequest(request_id=f"r{i}", model="m",
                                       prompt="x"), flow_key=f"f{flow}",
                    priority=prio, byte_size=size, ttl_s=60.0)
                items.append(it)
                fc.submit(it)
            else:
                state["ok"] = kind == "tick"
                fc.tick()
        state["ok"] = True
        for _ in range(20):
            fc.tick()
        for it in items:
            if it.outcome is not None:
                assert it.outcome in (QueueOutcome.DISPATCHED,
                                      QueueOutcome.REJECTED_CAPACITY,
                                      QueueOutcome.EVICTED_DISPLACED)
        for d in dispatched:
            assert d.outcome == QueueOutcome.DISPATCHED
        assert fc.queued_len == 0
        n_disp = sum(1 for i in items
                     if i.outcome == QueueOutcome.DISPATCHED)
        assert n_disp == len(dispatched)
        fc.stop()
