"""E/P/D multimodal path: encode-role vision tower, embedding hand-off over
dist send/recv, prefill with prefix embeds, KV hand-off to decode.
3-process gloo world (1 encode + 1 prefill + 1 decode)."""
import json

import pytest
import torch
import torch.multiprocessing as mp

from llm_d_inference_scheduler_amd.models.configs import TINY_LLAVA
from llm_d_inference_scheduler_amd.models.vision import VisionEncoder
from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
from llm_d_inference_scheduler_amd.scheduling.types import (LLMRequest,
                                                            MultiModalItem)

EPD_YAML = """
plugins:
  - type: decode-filter
  - type: prefill-filter
  - type: encode-filter
  - type: queue-scorer
  - type: kv-cache-utilization-scorer
  - type: max-score-picker
  - type: always-disagg-pd-decider
  - type: always-disagg-multimodal-decider
  - type: disagg-profile-handler
    parameters:
      pdDecider: always-disagg-pd-decider
      encodeDecider: always-disagg-multimodal-decider
schedulingProfiles:
  - name: decode
    plugins:
      - {pluginRef: decode-filter}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: kv-cache-utilization-scorer, weight: 1}
      - {pluginRef: max-score-picker}
  - name: prefill
    plugins:
      - {pluginRef: prefill-filter}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
  - name: encode
    plugins:
      - {pluginRef: encode-filter}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
"""

N_TEXT = 40


def _epd_worker(rank, world_size, init_file, out_file):
    import torch.distributed as dist
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world_size)
    try:
        cfg = NodeConfig(model=TINY_LLAVA, rank=rank, world_size=world_size,
                         topology="epd:1e1p1d", device="cpu",
                         dtype=torch.float32, kv_blocks=256,
                         epp_yaml=EPD_YAML, seed=4)
        node = NodeRunner(cfg)
        results = []
        if rank == 0:
            for i in range(3):
                req = LLMRequest(
                    request_id=f"mm{i}", model="tiny-llava",
                    prompt_tokens=list(range(100, 100 + N_TEXT)),
                    prompt="img request", max_tokens=3)
                # two items, one duplicated URL (dedupe -> 2 unique)
                req.mm_items = [MultiModalItem("image_url", f"http://img/{i}"),
                                MultiModalItem("image_url", "http://img/shared"),
                                MultiModalItem("image_url", f"http://img/{i}")]
                node.submit(req)
            # plus one text-only request: must skip the encode stage
            node.submit(LLMRequest(request_id="txt", model="tiny-llava",
                                   prompt_tokens=list(range(60)),
                                   prompt="plain", max_tokens=3))
        for _ in range(250):
            node.step()
            if rank == 0:
                results.extend(node.drain_completions())
            flag = torch.tensor([1 if (rank == 0 and len(results) >= 4)
                                 else 0])
            dist.broadcast(flag, src=0)
            if flag.item():
                break
        if rank == 0:
            with open(out_file, "w") as f:
                json.dump([{ "id": c.request_id, "tokens": c.tokens,
                             "error": c.error,
                             "completion": c.usage.completion_tokens}
                           for c in results], f)
        node.shutdown()
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
class TestEPD:
    def test_vision_encoder_deterministic(self):
        enc1 = VisionEncoder(TINY_LLAVA, "cpu", dtype=torch.float32, seed=4)
        enc2 = VisionEncoder(TINY_LLAVA, "cpu", dtype=torch.float32, seed=4)
        e1 = enc1.encode_url("http://img/a")
        e2 = enc2.encode_url("http://img/a")
        assert e1.shape == (TINY_LLAVA.vision_patches, TINY_LLAVA.hidden_size)
        assert torch.allclose(e1, e2)
        assert not torch.allclose(e1, enc1.encode_url("http://img/b"))
        # cache dedupe
        assert enc1.encode_url("http://img/a") is e1

    def test_epd_over_gloo(self, tmp_path):
        init_file = str(tmp_path / "pg_init")
        out_file = str(tmp_path / "out.json")
        mp.start_processes(_epd_worker, args=(3, init_file, out_file),
                           nprocs=3, join=True, start_method="spawn")
        with open(out_file) as f:
            results = json.load(f)
        assert len(results) == 4
        by_id = {r["id"]: r for r in results}
        for i in range(3):
            r = by_id[f"mm{i}"]
            assert not r["error"]
            assert r["completion"] == 3
        assert not by_id["txt"]["error"]
        assert by_id["txt"]["completion"] == 3

    def test_prefix_embeds_prefill_single_worker(self):
        """Engine-level: prefix embeds change the generation (they are
        really consumed), and the same embeds reproduce it."""
        from llm_d_inference_scheduler_amd.engine import (EngineRequest,
                                                          EngineWorker)
        enc = VisionEncoder(TINY_LLAVA, "cpu", dtype=torch.float32, seed=4)
        emb = enc.encode_url("http://img/x")
        text = list(range(200, 232))

        def run(embeds):
            w = EngineWorker(TINY_LLAVA, "cpu", dtype=torch.float32,
                             kv_blocks=128, seed=4)
            req = EngineRequest("r", prompt_tokens=(
                [0] * (emb.shape[0] if embeds is not None else 0) + text),
                max_tokens=4, prefix_embeds=embeds)
            w.add_request(req)
            toks = []
            for _ in range(30):
                for o in w.step():
                    toks.extend(o.new_tokens)
                if not w.has_work:
                    break
            return toks

        with_emb = run(emb)
        with_emb2 = run(emb.clone())
        without = run(None)
        assert with_emb == with_emb2
        assert len(with_emb) == 4
        assert with_emb != without  # embeds actually flowed into attention
