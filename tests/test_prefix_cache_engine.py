"""Engine-level automatic prefix caching + pipelined decode semantics.

Reference analog: the router's approx-prefix producer *estimates* vLLM's
prefix cache (reference approximateprefix/plugin.go:214-230); in this build
the engine IS the model server, so the cache is real — these tests pin the
block-reuse, resurrection and correctness (identical greedy outputs with
and without cache hits) contracts.
"""
import numpy as np
import pytest
import torch

from llm_d_inference_scheduler_amd.engine import EngineRequest, EngineWorker
from llm_d_inference_scheduler_amd.engine.kvcache import (BlockManager,
                                                          block_hashes)
from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA


def run_to_completion(worker, max_steps=200):
    outs = []
    for _ in range(max_steps):
        outs.extend(worker.step())
        if not worker.has_work:
            break
    return outs


def make_worker(**kw):
    kw.setdefault("kv_blocks", 256)
    kw.setdefault("dtype", torch.float32)
    return EngineWorker(TINY_LLAMA, "cpu", **kw)


PROMPT = list(range(7, 7 + 48))          # 3 full blocks exactly


class TestBlockManagerPrefixCache:
    def test_miss_then_hit(self):
        mgr = BlockManager(32)
        h = block_hashes(PROMPT)
        assert len(h) == 3
        assert mgr.allocate_prompt("a", h, len(PROMPT)) == 0
        mgr.allocate("a", 48)
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        # block-aligned prompt: last block must be recomputed (cap at
        # prompt_len-1), so the match is 2 blocks = 32 tokens
        assert mgr.allocate_prompt("b", h, len(PROMPT)) == 32
        assert mgr.tables["b"][:2] == mgr.tables["a"][:2]

    def test_longer_prompt_full_match(self):
        mgr = BlockManager(32)
        long_prompt = PROMPT + [1, 2, 3]
        h = block_hashes(long_prompt)
        mgr.allocate_prompt("a", h, len(long_prompt))
        mgr.allocate("a", len(long_prompt))
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        assert mgr.allocate_prompt("b", h, len(long_prompt)) == 48

    def test_resurrect_from_free_lru(self):
        mgr = BlockManager(32)
        h = block_hashes(PROMPT)
        mgr.allocate_prompt("a", h, len(PROMPT))
        mgr.allocate("a", 48)
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        blocks_a = list(mgr.tables["a"])
        mgr.free("a")                       # content survives in free LRU
        assert mgr.free_blocks == 32
        assert mgr.allocate_prompt("b", h, len(PROMPT)) == 32
        assert mgr.tables["b"][:2] == blocks_a[:2]

    def test_eviction_invalidates(self):
        mgr = BlockManager(4)
        h = block_hashes(PROMPT)
        mgr.allocate_prompt("a", h, len(PROMPT))
        mgr.allocate("a", 48)
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        mgr.free("a")
        # churn through all blocks with an unrelated sequence
        mgr.allocate("x", 64)
        mgr.free("x")
        assert mgr.allocate_prompt("c", h, len(PROMPT)) == 0

    def test_refcount_protects_shared(self):
        mgr = BlockManager(8)
        h = block_hashes(PROMPT)
        mgr.allocate_prompt("a", h, len(PROMPT))
        mgr.allocate("a", 48)
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        mgr.allocate_prompt("b", h, len(PROMPT))
        mgr.allocate("b", 48)
        mgr.free("a")
        # b still holds the shared blocks: they must not be handed out
        assert not mgr.can_allocate(8 * 16)
        taken = mgr.take_blocks(mgr.free_blocks)
        assert set(taken).isdisjoint(set(mgr.tables["b"]))

    def test_hit_rate_stats(self):
        mgr = BlockManager(32)
        h = block_hashes(PROMPT)
        mgr.allocate_prompt("a", h, len(PROMPT))
        mgr.allocate("a", 48)
        for b in range(3):
            mgr.register_block("a", b, int(h[b]))
        mgr.allocate_prompt("b", h, len(PROMPT))
        assert mgr.cached_tokens_total == 32
        assert mgr.queried_tokens_total == 96
        assert 0 < mgr.hit_rate < 1


class TestEnginePrefixCache:
    def test_identical_outputs_on_cache_hit(self):
        w = make_worker()
        prompt = list(np.random.default_rng(3).integers(5, 900, size=50))
        w.add_request(EngineRequest("r1", list(prompt), max_tokens=8))
        outs1 = run_to_completion(w)
        fin1 = [o for o in outs1 if o.finished][0]
        assert fin1.cached_tokens == 0

        w.add_request(EngineRequest("r2", list(prompt), max_tokens=8))
        outs2 = run_to_completion(w)
        fin2 = [o for o in outs2 if o.finished][0]
        assert fin2.cached_tokens == 48      # 3 full blocks reused
        assert fin2.all_tokens == fin1.all_tokens

    def test_shared_prefix_partial_hit(self):
        w = make_worker()
        rng = np.random.default_rng(5)
        shared = list(rng.integers(5, 900, size=32))
        p1 = shared + list(rng.integers(5, 900, size=20))
        p2 = shared + list(rng.integers(5, 900, size=20))
        w.add_request(EngineRequest("r1", p1, max_tokens=4))
        run_to_completion(w)
        w.add_request(EngineRequest("r2", p2, max_tokens=4))
        outs = run_to_completion(w)
        fin = [o for o in outs if o.finished][0]
        assert fin.cached_tokens == 32       # the shared 2 blocks

    def test_cache_disabled(self):
        w = make_worker(prefix_caching=False)
        prompt = list(range(10, 60))
        for rid in ("a", "b"):
            w.add_request(EngineRequest(rid, list(prompt), max_tokens=4))
            outs = run_to_completion(w)
            assert [o for o in outs if o.finished][0].cached_tokens == 0


class TestPipelinedDecode:
    def test_exact_token_counts(self):
        w = make_worker()
        for i in range(4):
            w.add_request(EngineRequest(
                f"r{i}", list(range(3 + i, 40 + i)), max_tokens=6))
        outs = run_to_completion(w)
        fins = {o.request_id: o for o in outs if o.finished}
        assert len(fins) == 4
        for o in fins.values():
            assert o.completion_tokens == 6
            assert len(o.all_tokens) == 6
        assert w.total_generated == 24
        assert not w.running and not w.waiting and w._pending is None

    def test_max_tokens_one_finalizes(self):
        w = make_worker()
        w.add_request(EngineRequest("one", list(range(5, 25)),
                                    max_tokens=1))
        outs = run_to_completion(w)
        fin = [o for o in outs if o.finished]
        assert fin and fin[0].completion_tokens == 1

    def test_abort_in_flight(self):
        w = make_worker()
        w.add_request(EngineRequest("a", list(range(5, 45)), max_tokens=50))
        w.add_request(EngineRequest("b", list(range(6, 46)), max_tokens=6))
        w.step()
        w.step()              # b's tokens now in flight
        w.abort("a")
        outs = run_to_completion(w)
        fins = [o for o in outs if o.finished]
        assert len(fins) == 1 and fins[0].request_id == "b"
        assert w.mgr.usage == pytest.approx(0.0)

    def test_staggered_finish_membership_churn(self):
        w = make_worker()
        for i in range(5):
            w.add_request(EngineRequest(
                f"r{i}", list(range(3 + i, 43 + i)), max_tokens=2 + 2 * i))
        outs = run_to_completion(w)
        fins = {o.request_id: o.completion_tokens
                for o in outs if o.finished}
        assert fins == {f"r{i}": 2 + 2 * i for i in range(5)}


class TestPreemption:
    def test_kv_exhaustion_preempts_and_completes(self):
        """Pool too small for all requests at once: the engine preempts
        (recompute with generation folded into the prompt) instead of
        deadlocking, and every request still finishes with exact counts."""
        w = make_worker(kv_blocks=12)           # 192 tokens of KV
        for i in range(4):
            w.add_request(EngineRequest(
                f"r{i}", list(range(3 + i, 43 + i)), max_tokens=30))
        outs = run_to_completion(w, max_steps=600)
        fins = {o.request_id: o for o in outs if o.finished}
        assert len(fins) == 4, fins.keys()
        for o in fins.values():
            assert o.completion_tokens == 30
        assert w.mgr.usage == pytest.approx(0.0)

    def test_preempted_greedy_tokens_unchanged(self):
        """A preempted+recomputed request produces the same greedy tokens
        as an undisturbed run."""
        prompt = list(range(11, 51))
        w = make_worker(kv_blocks=256)
        w.add_request(EngineRequest("ref", list(prompt), max_tokens=12))
        ref = [o for o in run_to_completion(w) if o.finished][0].all_tokens

        w2 = make_worker(kv_blocks=256)
        w2.add_request(EngineRequest("p", list(prompt), max_tokens=12))
        for _ in range(6):
            w2.step()
        # force-preempt mid-generation
        w2.step()
        outs = list(w2._collect_pending())
        w2._preempt(w2.running[-1])
        for _ in range(200):
            outs.extend(w2.step())
            if not w2.has_work:
                break
        got = [o for o in outs if o.finished][0].all_tokens
        assert got == ref

    def test_preemption_prefers_low_priority(self):
        """Victim selection honors InferenceObjective priority: the
        sheddable request is recomputed, the critical one is untouched."""
        w = make_worker(kv_blocks=8)
        w.add_request(EngineRequest("crit", list(range(5, 37)),
                                    max_tokens=40, priority=10))
        w.add_request(EngineRequest("shed", list(range(200, 232)),
                                    max_tokens=40, priority=-1))
        preempted = []
        orig = w._preempt

        def spy(req):
            preempted.append(req.request_id)
            return orig(req)
        w._preempt = spy
        outs = run_to_completion(w, max_steps=800)
        fins = {o.request_id for o in outs if o.finished}
        assert fins == {"crit", "shed"}
        assert preempted and all(r == "shed" for r in preempted)


class TestStopTokens:
    def test_stop_token_finishes_early(self):
        w = make_worker()
        # run once to learn the greedy continuation, then stop on its
        # 3rd generated token
        w.add_request(EngineRequest("probe", list(range(5, 45)),
                                    max_tokens=8))
        ref = [o for o in run_to_completion(w) if o.finished][0].all_tokens
        w2 = make_worker()
        w2.add_request(EngineRequest("s", list(range(5, 45)), max_tokens=8,
                                     stop_token_ids=[ref[2]]))
        outs = run_to_completion(w2)
        fin = [o for o in outs if o.finished][0]
        assert fin.finish_reason == "stop"
        assert fin.completion_tokens == 3
        assert fin.all_tokens == ref[:3]
        assert not w2.has_work and w2.mgr.usage == pytest.approx(0.0)


class TestPriorityPreemptsBeforeExhaustion:
    def test_critical_waiter_preempts_running_low_priority(self):
        """A higher-priority arrival whose prefill cannot allocate
        preempts the lowest-priority running request immediately instead
        of waiting for full KV exhaustion (round-2 refinement)."""
        w = make_worker(kv_blocks=16)   # 256 token slots
        # low-priority request grabs most of the pool, with a generation
        # long enough that it cannot finish within the probe window
        w.add_request(EngineRequest("low", list(range(150)), max_tokens=90,
                                    priority=-1))
        for _ in range(20):
            w.step()
            if w.running:
                break
        assert w.running and w.running[0].request_id == "low"
        # collect pending so the victim is preemptible
        for _ in range(3):
            w.step()
        # critical arrival needs more than the free space
        w.add_request(EngineRequest("crit", list(range(200, 300)),
                                    max_tokens=4, priority=10))
        preempted = False
        outs = []
        for _ in range(60):
            outs.extend(w.step())
            if any(r.request_id == "crit" for r in w.running):
                preempted = True
                break
        assert preempted, "critical request never started decoding"
        # the low-priority request was requeued (recompute), not lost
        ids_active = {r.request_id for r in w.running} | \
                     {r.request_id for r in w.waiting}
        assert "low" in ids_active
        # run everything to completion: both finish
        for _ in range(300):
            outs.extend(w.step())
            if not w.has_work:
                break
        fin = {o.request_id for o in outs if o.finished and not o.error}
        assert {"low", "crit"} <= fin

    def test_no_preemption_for_equal_priority(self):
        w = make_worker(kv_blocks=16)
        w.add_request(EngineRequest("a", list(range(150)), max_tokens=90,
                                    priority=0))
        for _ in range(20):
            w.step()
            if w.running:
                break
        for _ in range(3):
            w.step()
        w.add_request(EngineRequest("b", list(range(200, 300)),
                                    max_tokens=4, priority=0))
        for _ in range(10):
            w.step()
        # same priority never displaces: "a" keeps running
        assert any(r.request_id == "a" for r in w.running)
