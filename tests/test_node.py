"""NodeRunner: single-rank routing+engine loop, and multi-process gloo P/D
disaggregation with real dist send/recv of KV blocks (world_size=2)."""
import json
import os
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest

EPP_YAML_TIGHT_DISAGG = """
plugins:
  - type: decode-filter
  - type: prefill-filter
  - type: queue-scorer
  - type: kv-cache-utilization-scorer
  - type: prefix-cache-scorer
  - type: max-score-picker
  - type: prefix-based-pd-decider
    parameters: {nonCachedTokens: 16}
  - type: disagg-profile-handler
    parameters:
      pdDecider: prefix-based-pd-decider
schedulingProfiles:
  - name: decode
    plugins:
      - {pluginRef: decode-filter}
      - {pluginRef: prefix-cache-scorer, weight: 3}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
  - name: prefill
    plugins:
      - {pluginRef: prefill-filter}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
"""


def make_req(i, n_prompt=40, max_tokens=4):
    return LLMRequest(request_id=f"r{i}", model="tiny-llama",
                      prompt_tokens=list(range(100, 100 + n_prompt)),
                      prompt="x " * n_prompt, max_tokens=max_tokens)


def run_node_until(node, want, max_steps=300):
    done = []
    for _ in range(max_steps):
        node.step()
        done.extend(node.drain_completions())
        if len(done) >= want:
            break
    return done


class TestSingleRank:
    def test_mono_end_to_end(self):
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=256)
        node = NodeRunner(cfg)
        for i in range(5):
            node.submit(make_req(i))
        done = run_node_until(node, 5)
        assert len(done) == 5
        assert all(not c.error for c in done)
        assert all(c.usage.completion_tokens == 4 for c in done)
        assert all(c.usage.ttft_ms is not None for c in done)
        node.shutdown()

    def test_flow_control_mode(self):
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=256,
                         flow_control=True)
        node = NodeRunner(cfg)
        for i in range(6):
            node.submit(make_req(i, max_tokens=2))
        done = run_node_until(node, 6)
        assert len(done) == 6
        assert node.flow is not None
        assert node.flow.registry.stats.dispatched == 6
        node.shutdown()


def _pd_worker(rank, world_size, init_file, out_file):
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    try:
        cfg = NodeConfig(model=TINY_LLAMA, rank=rank, world_size=world_size,
                         topology="pd:1p1d", device="cpu",
                         dtype=torch.float32, kv_blocks=256,
                         epp_yaml=EPP_YAML_TIGHT_DISAGG, seed=3)
        node = NodeRunner(cfg)
        results = []
        if rank == 0:
            for i in range(4):
                node.submit(make_req(i, n_prompt=48, max_tokens=4))
        for _ in range(200):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
                done_flag = torch.tensor(
                    [1 if len(results) >= 4 else 0])
            else:
                done_flag = torch.tensor([0])
            dist.broadcast(done_flag, src=0)
            if done_flag.item():
                break
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump([{ "id": c.request_id, "tokens": c.tokens,
                             "error": c.error,
                             "completion": c.usage.completion_tokens,
                             "cached": c.usage.cached_tokens}
                           for c in results], f)
        node.shutdown()
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
class TestMultiProcessPD:
    def test_pd_disagg_over_gloo(self, tmp_path):
        """2-process P/D: rank0 prefill+router, rank1 decode; KV blocks move
        via dist.send/recv; completions match a monolithic reference run."""
        init_file = str(tmp_path / "pg_init")
        out_file = str(tmp_path / "out.json")
        mp.start_processes(_pd_worker, args=(2, init_file, out_file),
                           nprocs=2, join=True, start_method="spawn")
        with open(out_file) as f:
            results = json.load(f)
        assert len(results) == 4
        assert all(not r["error"] for r in results)
        assert all(r["completion"] == 4 for r in results)

        # monolithic reference on one engine, same seed: tokens must match
        from llm_d_inference_scheduler_amd.engine import (EngineRequest,
                                                          EngineWorker)
        mono = EngineWorker(TINY_LLAMA, "cpu", dtype=torch.float32,
                            kv_blocks=256, seed=3)
        mono.add_request(EngineRequest(
            "m", prompt_tokens=list(range(100, 148)), max_tokens=4))
        toks = []
        for _ in range(50):
            for o in mono.step():
                toks.extend(o.new_tokens)
            if not mono.has_work:
                break
        for r in results:
            assert r["tokens"] == toks


class TestChunkedDecode:
    def test_chunked_matches_unchunked(self):
        """decode_chunk_tokens splits generation into re-routed chunks
        (sidecar decode.go analog); cumulative result identical to one
        dispatch because the prefix cache replays the continuation."""
        results = {}
        for chunk in (None, 3):
            cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                             device="cpu", dtype=torch.float32,
                             kv_blocks=256, decode_chunk_tokens=chunk,
                             seed=5)
            node = NodeRunner(cfg)
            node.submit(make_req(0, n_prompt=40, max_tokens=8))
            got = []
            for _ in range(120):
                node.step()
                got.extend(node.drain_completions())
                if got:
                    break
            node.shutdown()
            assert got and not got[0].error
            assert got[0].usage.completion_tokens == 8
            results[chunk] = got[0].tokens
        assert results[None] == results[3]


def _pd4_worker(rank, world_size, init_file, out_file):
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    try:
        cfg = NodeConfig(model=TINY_LLAMA, rank=rank, world_size=world_size,
                         topology="pd:1p3d", device="cpu",
                         dtype=torch.float32, kv_blocks=256,
                         epp_yaml=EPP_YAML_TIGHT_DISAGG, seed=3)
        node = NodeRunner(cfg)
        results = []
        if rank == 0:
            for i in range(9):
                node.submit(make_req(i, n_prompt=48, max_tokens=4))
        for _ in range(300):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
                done = torch.tensor([1 if len(results) >= 9 else 0])
            else:
                done = torch.tensor([0])
            dist.broadcast(done, src=0)
            if done.item():
                break
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump([{ "id": c.request_id, "error": c.error,
                             "completion": c.usage.completion_tokens}
                           for c in results], f)
        node.shutdown()
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
class TestMultiProcessPD4:
    def test_pd_1p3d_world4(self, tmp_path):
        """4-process 1p3d: requests spread over 3 decode ranks, KV from the
        shared prefill rank — the shape the driver's 8-GPU run scales up."""
        init_file = str(tmp_path / "pg4_init")
        out_file = str(tmp_path / "out4.json")
        mp.start_processes(_pd4_worker, args=(4, init_file, out_file),
                           nprocs=4, join=True, start_method="spawn")
        with open(out_file) as f:
            results = json.load(f)
        assert len(results) == 9
        assert all(not r["error"] for r in results), results
        assert all(r["completion"] == 4 for r in results)


class TestChunkedStop:
    def test_stop_token_ends_chunked_decode(self):
        from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=256,
                         seed=5)
        node = NodeRunner(cfg)
        node.submit(make_req(0, n_prompt=40, max_tokens=10))
        ref = None
        for _ in range(120):
            node.step()
            done = node.drain_completions()
            if done:
                ref = done[0].tokens
                break
        node.shutdown()
        cfg2 = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                          device="cpu", dtype=torch.float32, kv_blocks=256,
                          decode_chunk_tokens=3, seed=5)
        node = NodeRunner(cfg2)
        req = make_req(0, n_prompt=40, max_tokens=10)
        req.stop_token_ids = [ref[4]]       # stops inside the 2nd chunk
        node.submit(req)
        got = None
        for _ in range(200):
            node.step()
            done = node.drain_completions()
            if done:
                got = done[0]
                break
        node.shutdown()
        assert got is not None and got.finish_reason == "stop"
        assert got.tokens == ref[:5]


def _pd_bp_worker(rank, world_size, init_file, out_file):
    """P/D under tight transfer back-pressure: max_step_transfer_bytes=1
    forces one kv_ready job per step (deterministic deferral on both
    ranks); everything must still complete."""
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    try:
        cfg = NodeConfig(model=TINY_LLAMA, rank=rank, world_size=world_size,
                         topology="pd:1p1d", device="cpu",
                         dtype=torch.float32, kv_blocks=256,
                         epp_yaml=EPP_YAML_TIGHT_DISAGG, seed=3,
                         max_step_transfer_bytes=1)
        node = NodeRunner(cfg)
        results = []
        if rank == 0:
            for i in range(6):
                node.submit(make_req(i, n_prompt=48, max_tokens=3))
        for _ in range(300):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
                done = torch.tensor([1 if len(results) >= 6 else 0])
            else:
                done = torch.tensor([0])
            dist.broadcast(done, src=0)
            if done.item():
                break
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump([{ "id": c.request_id, "error": c.error,
                             "completion": c.usage.completion_tokens}
                           for c in results], f)
        node.shutdown()
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
class TestTransferBackpressure:
    def test_pd_one_job_per_step(self, tmp_path):
        init_file = str(tmp_path / "pgbp_init")
        out_file = str(tmp_path / "outbp.json")
        mp.start_processes(_pd_bp_worker, args=(2, init_file, out_file),
                           nprocs=2, join=True, start_method="spawn")
        with open(out_file) as f:
            results = json.load(f)
        assert len(results) == 6
        assert all(not r["error"] for r in results), results
        assert all(r["completion"] == 3 for r in results)


class TestTransferAsyncUnit:
    """Unit coverage of the async-transfer state machine (single process)."""

    def _node(self):
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=64)
        return NodeRunner(cfg)

    def test_kv_released_frees_prefill_blocks(self):
        node = self._node()
        mgr = node.engine.mgr
        assert mgr.allocate("p1", 32)
        held = mgr.free_blocks
        node._process_messages([{"type": "kv_released", "req_id": "p1",
                                 "dst": 0}])
        assert mgr.free_blocks == held + 2
        assert "p1" not in mgr.tables
        node.shutdown()

    def test_abort_mid_transfer_no_double_free(self):
        """Abort arriving while a pull/recv is in flight: the completion
        callback (not the abort) owns the in-flight blocks' release, and
        they are released exactly once."""
        from llm_d_inference_scheduler_amd.engine import EngineRequest
        node = self._node()
        mgr = node.engine.mgr
        free0 = mgr.free_blocks

        # stub transfer: capture the completion instead of running it
        captured = {}

        def fake_recv(src, local, on_complete=None):
            captured["cb"] = on_complete
        node.transfer.recv_blocks = fake_recv
        node.rank = 0

        req = EngineRequest("rx", list(range(40)), max_tokens=4)
        node._pending_adoption["rx"] = {
            "req": req, "src": 1,
            "reserved": mgr.take_blocks(3)}
        job = {"type": "kv_ready", "req_id": "rx", "src": 1, "dst": 0,
               "blocks": [0, 1, 2], "seq_len": 40, "first_token": 7}
        node._start_kv_job(job)
        assert "cb" in captured
        # abort lands while the copy is in flight
        node.cancel("rx")
        assert mgr.free_blocks == free0 - 3  # still held by the transfer
        captured["cb"]()                      # transfer completes
        assert mgr.free_blocks == free0       # released exactly once
        node.shutdown()


def _pd_pipelined_worker(rank, world_size, init_file, out_file):
    """P/D with the PIPELINED mailbox (dedicated gloo subgroup, one-step-
    deep exchange overlap — the production bench configuration)."""
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    mailbox_group = dist.new_group(backend="gloo")
    ctrl_group = dist.new_group(backend="gloo")
    try:
        cfg = NodeConfig(model=TINY_LLAMA, rank=rank, world_size=world_size,
                         topology="pd:1p1d", device="cpu",
                         dtype=torch.float32, kv_blocks=256,
                         epp_yaml=EPP_YAML_TIGHT_DISAGG, seed=3,
                         mailbox_group=mailbox_group)
        node = NodeRunner(cfg)
        assert node.mailbox.pipelined
        results = []
        if rank == 0:
            for i in range(5):
                node.submit(make_req(i, n_prompt=48, max_tokens=4))
        for _ in range(300):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
                done = torch.tensor([1 if len(results) >= 5 else 0])
            else:
                done = torch.tensor([0])
            dist.broadcast(done, src=0, group=ctrl_group)
            if done.item():
                break
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump([{ "id": c.request_id, "error": c.error,
                             "completion": c.usage.completion_tokens}
                           for c in results], f)
        node.shutdown()
        dist.barrier(group=ctrl_group)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
class TestPipelinedMailbox:
    def test_pd_with_pipelined_exchange(self, tmp_path):
        init_file = str(tmp_path / "pgpl_init")
        out_file = str(tmp_path / "outpl.json")
        mp.start_processes(_pd_pipelined_worker,
                           args=(2, init_file, out_file),
                           nprocs=2, join=True, start_method="spawn")
        with open(out_file) as f:
            results = json.load(f)
        assert len(results) == 5
        assert all(not r["error"] for r in results), results
        assert all(r["completion"] == 4 for r in results)


class TestPrefillLoss:
    def test_stale_adoption_expires_and_frees_reservation(self):
        """A decode rank whose prefill peer died (kv_ready never arrives)
        fails the request with prefill_lost after the TTL and releases
        the sglang-style reservation instead of leaking it."""
        from llm_d_inference_scheduler_amd.engine import EngineRequest
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=64)
        node = NodeRunner(cfg)
        mgr = node.engine.mgr
        free0 = mgr.free_blocks
        node._pending_adoption["dead"] = {
            "req": EngineRequest("dead", list(range(20)), max_tokens=2),
            "src": 1, "reserved": mgr.take_blocks(4), "step": 0}
        node._step = 10
        node._sweep_stale_adoptions(max_age_steps=5)
        assert "dead" not in node._pending_adoption
        assert mgr.free_blocks == free0
        done = [m for m in node._outbox if m.get("type") == "done"]
        assert done and done[0]["error"] == "prefill_lost"
        node.shutdown()


def _mono4_worker(rank, world_size, init_file, out_file):
    """mono DP world-4 with the pipelined mailbox — the exact shape of the
    driver's multi-GPU SCALE run (bench.py --mode mono under torchrun)."""
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    mailbox_group = dist.new_group(backend="gloo")
    ctrl_group = dist.new_group(backend="gloo")
    try:
        cfg = NodeConfig(model=TINY_LLAMA, rank=rank, world_size=world_size,
                         topology="mono", device="cpu",
                         dtype=torch.float32, kv_blocks=256, seed=3,
                         mailbox_group=mailbox_group)
        node = NodeRunner(cfg)
        results = []
        if rank == 0:
            for i in range(12):
                node.submit(make_req(i, n_prompt=40, max_tokens=4))
        for _ in range(300):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
                done = torch.tensor([1 if len(results) >= 12 else 0])
            else:
                done = torch.tensor([0])
            dist.broadcast(done, src=0, group=ctrl_group)
            if done.item():
                break
        if rank == 0:
            # DP spread: all four decode ranks served work
            served = torch.zeros(world_size)
        node.shutdown()
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump({"n": len(results),
                           "errors": [c.error for c in results
                                      if c.error]}, f)
        dist.barrier(group=ctrl_group)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
class TestMonoWorld4:
    def test_mono_dp4_pipelined(self, tmp_path):
        init_file = str(tmp_path / "pgm4_init")
        out_file = str(tmp_path / "outm4.json")
        mp.start_processes(_mono4_worker, args=(4, init_file, out_file),
                           nprocs=4, join=True, start_method="spawn")
        with open(out_file) as f:
            res = json.load(f)
        assert res["n"] == 12 and not res["errors"]


class TestObjectivePriorityEndToEnd:
    def test_critical_objective_preempts_batch_work(self):
        """InferenceObjective priority flows router->assign->engine and
        a critical arrival preempts batch-tier work for KV space."""
        from llm_d_inference_scheduler_amd.api.objectives import \
            InferenceObjective
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=16)
        node = NodeRunner(cfg)
        node.datastore.put_objective(InferenceObjective(
            "critical", priority=10))
        node.datastore.put_objective(InferenceObjective(
            "batch", priority=-1))
        hog = make_req(0, n_prompt=150, max_tokens=90)
        hog.objective_name = "batch"
        node.submit(hog)
        for _ in range(30):
            node.step()
            if node.engine.running:
                break
        assert node.engine.running
        crit = make_req(1, n_prompt=100, max_tokens=3)
        crit.objective_name = "critical"
        node.submit(crit)
        done = []
        for _ in range(80):
            node.step()
            done.extend(node.drain_completions())
            if any(c.request_id == "r1" for c in done):
                break
        crit_done = [c for c in done if c.request_id == "r1"]
        assert crit_done and not crit_done[0].error, done
        # the batch request survives (recompute) and finishes eventually
        for _ in range(400):
            node.step()
            done.extend(node.drain_completions())
            if any(c.request_id == "r0" for c in done):
                break
        assert any(c.request_id == "r0" and not c.error for c in done)
        node.shutdown()
