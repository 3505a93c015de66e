"""Debug: epd mode at world=1 on GPU produced 0 tok/s. Mimic bench's epd
setup with state prints per step."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.models.configs import \
    LLAVA_1_5_7B_TEXT  # noqa: E402
from llm_d_inference_scheduler_amd.node import (NodeConfig,  # noqa: E402
                                                NodeRunner)
from llm_d_inference_scheduler_amd.scheduling.types import (  # noqa: E402
    LLMRequest, MultiModalItem)

EPD_YAML = open(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             "..", "bench.py")).read()
EPD_YAML = EPD_YAML.split('EPD_YAML = """')[1].split('"""')[0]


def main():
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev.startswith("cuda") else torch.float32
    cfg = NodeConfig(model=LLAVA_1_5_7B_TEXT, world_size=1, topology="mono",
                     device=dev, dtype=dtype, kv_blocks=2048,
                     epp_yaml=EPD_YAML, seed=1)
    node = NodeRunner(cfg)
    import random
    rng = random.Random(0)
    for i in range(8):
        r = LLMRequest(request_id=f"r{i}", model=LLAVA_1_5_7B_TEXT.name,
                       prompt="",
                       prompt_tokens=[rng.randrange(256, 31000)
                                      for _ in range(256)],
                       max_tokens=4)
        r.mm_items = [MultiModalItem("image_url", f"http://img/{i % 2}")]
        node.submit(r)
    done = []
    for s in range(60):
        try:
            node.step()
        except Exception as e:
            import traceback
            traceback.print_exc()
            print("STEP RAISED:", e)
            break
        done.extend(node.drain_completions())
        if s < 6 or s % 10 == 0:
            print(f"step {s}: arrivals={len(node._arrivals)} "
                  f"decisions={len(node._decisions)} "
                  f"waiting={len(node.engine.waiting)} "
                  f"running={len(node.engine.running)} "
                  f"awaiting_embeds={len(node._awaiting_embeds)} "
                  f"encode_jobs={len(node._encode_jobs)} "
                  f"done={len(done)}")
        if len(done) >= 8:
            break
    print("completions:", len(done),
          "errors:", [c.error for c in done if c.error])
    node.shutdown()


if __name__ == "__main__":
    main()
