"""Precise prefix-cache scorer: KV-block index, engine event stream, and
router wiring (reference scorer/preciseprefixcache/precise_prefix_cache.go).
"""
import time

import numpy as np
import torch

from llm_d_inference_scheduler_amd.datalayer.kvblock import KVBlockIndex
from llm_d_inference_scheduler_amd.engine import EngineRequest, EngineWorker
from llm_d_inference_scheduler_amd.engine.kvcache import block_hashes
from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
from llm_d_inference_scheduler_amd.plugins.scorers import \
    PrecisePrefixCacheScorer
from llm_d_inference_scheduler_amd.scheduling.types import (LLMRequest,
                                                            SchedulingContext)


def ctx_for(tokens):
    req = LLMRequest(request_id="r", model="m", prompt="",
                     prompt_tokens=list(tokens), max_tokens=4)
    return SchedulingContext(request=req)


class TestKVBlockIndex:
    def test_events_and_match(self):
        ix = KVBlockIndex()
        ix.apply_events("gpu0", [11, 22, 33], [])
        ix.apply_events("gpu1", [11], [])
        m = ix.match_longest([11, 22, 33, 44], ["gpu0", "gpu1", "gpu2"])
        assert m == {"gpu0": 3, "gpu1": 1, "gpu2": 0}
        ix.apply_events("gpu0", [], [22])
        assert ix.match_longest([11, 22, 33], ["gpu0"])["gpu0"] == 1

    def test_speculative_ttl(self):
        ix = KVBlockIndex(speculative_ttl_s=0.05)
        ix.add_speculative("gpu0", [7, 8])
        assert ix.match_longest([7, 8], ["gpu0"])["gpu0"] == 2
        time.sleep(0.06)
        assert ix.match_longest([7, 8], ["gpu0"])["gpu0"] == 0
        ix.sweep()
        assert not ix._spec

    def test_confirm_replaces_speculative(self):
        ix = KVBlockIndex(speculative_ttl_s=0.01)
        ix.add_speculative("gpu0", [5])
        ix.apply_events("gpu0", [5], [])
        time.sleep(0.02)
        assert ix.match_longest([5], ["gpu0"])["gpu0"] == 1

    def test_remove_endpoint(self):
        ix = KVBlockIndex()
        ix.apply_events("gpu0", [1, 2], [])
        ix.remove_endpoint("gpu0")
        assert ix.match_longest([1], ["gpu0"])["gpu0"] == 0
        assert ix.size == 0


class TestEngineKVEvents:
    def test_store_and_evict_events(self):
        w = EngineWorker(TINY_LLAMA, "cpu", kv_blocks=8,
                         dtype=torch.float32)
        prompt = list(range(10, 58))        # 3 full blocks
        w.add_request(EngineRequest("a", prompt, max_tokens=2))
        for _ in range(20):
            w.step()
            if not w.has_work:
                break
        stored, evicted = w.mgr.drain_events()
        h = block_hashes(prompt)
        assert [int(x) for x in h[:2]] == stored[:2]
        assert not evicted
        # churn the tiny pool -> eviction events for the cached content
        w.add_request(EngineRequest("b", list(range(200, 280)),
                                    max_tokens=2))
        for _ in range(30):
            w.step()
            if not w.has_work:
                break
        _, evicted = w.mgr.drain_events()
        assert set(evicted) & {int(x) for x in h[:2]}


class TestPreciseScorer:
    def test_score_and_speculative(self):
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        sc = PrecisePrefixCacheScorer()
        eps = [make_endpoint(f"gpu{i}", i) for i in range(2)]
        prompt = list(range(100, 164))      # 4 blocks
        h = block_hashes(prompt)
        sc.apply_events("gpu1", [int(x) for x in h[:3]], [])
        scores = sc.score(ctx_for(prompt), eps)
        assert scores["gpu1"] == 0.75 and scores["gpu0"] == 0.0
        # routing adds speculative coverage for the target
        c = ctx_for(prompt)
        sc.pre_request(c, None, eps[0])
        scores = sc.score(ctx_for(prompt), eps)
        assert scores["gpu0"] == 1.0

    def test_node_integration_events_reach_router(self):
        """Single-rank node: engine block events flow through the mailbox
        loop into the scorer's index."""
        from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
        yaml_cfg = """
plugins:
  - type: decode-filter
  - type: precise-prefix-cache-scorer
  - type: queue-scorer
  - type: max-score-picker
schedulingProfiles:
  - name: decode
    plugins:
      - {pluginRef: decode-filter}
      - {pluginRef: precise-prefix-cache-scorer, weight: 2}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
"""
        node = NodeRunner(NodeConfig(model=TINY_LLAMA, epp_yaml=yaml_cfg,
                                     device="cpu", dtype=torch.float32,
                                     kv_blocks=128))
        assert node._precise is not None
        rng = np.random.default_rng(0)
        prompt = [int(x) for x in rng.integers(5, 900, size=64)]
        node.submit(LLMRequest(request_id="q1", model=TINY_LLAMA.name,
                               prompt="", prompt_tokens=prompt,
                               max_tokens=2))
        for _ in range(40):
            node.step()
            if node.drain_completions():
                break
        # engine registered the prompt's full blocks; events applied
        h = block_hashes(prompt)
        m = node._precise.index.match_longest(
            [int(x) for x in h], ["gpu0"])
        assert m["gpu0"] >= 3
        node.shutdown()


class TestEndpointLifecycleNotifications:
    def test_removal_fans_out_to_plugin_hooks(self):
        """Endpoint removal drives plugin remove_endpoint hooks — the
        notification-source surface that the reference uses to tear down
        per-pod ZMQ subscribers (precise_prefix_cache.go:622-691)."""
        from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
        yaml_cfg = """
plugins:
  - type: decode-filter
  - type: precise-prefix-cache-scorer
  - type: queue-scorer
  - type: max-score-picker
schedulingProfiles:
  - name: decode
    plugins:
      - {pluginRef: decode-filter}
      - {pluginRef: precise-prefix-cache-scorer, weight: 2}
      - {pluginRef: max-score-picker}
"""
        node = NodeRunner(NodeConfig(model=TINY_LLAMA, epp_yaml=yaml_cfg,
                                     device="cpu", dtype=torch.float32,
                                     kv_blocks=64))
        node._precise.apply_events("gpu0", [111, 222], [])
        assert node._precise.index.size == 2
        node.datastore.remove_endpoint("gpu0")
        assert node._precise.index.size == 0
        node.shutdown()
