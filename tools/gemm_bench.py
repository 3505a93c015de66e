"""hipBLASLt GEMM throughput at the serving shapes (run on a GPU box)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.models.llama import \
    _load_tuned_gemms  # noqa: E402

SHAPES = [(4096, 6144, "qkv"), (4096, 4096, "wo"),
          (4096, 28672, "gate_up"), (14336, 4096, "down")]


def bench(m, k, n, tag, iters=30):
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        a @ b
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        a @ b
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    tf = 2 * m * k * n / (us / 1e6) / 1e12
    print(f"M={m:5d} K={k:5d} N={n:5d} {tag:8s}: {us:8.1f} us  {tf:7.0f} TF/s")


if __name__ == "__main__":
    _load_tuned_gemms()
    for m in (1024, 2048, 4096, 8192):
        for k, n, tag in SHAPES:
            bench(m, k, n, tag)
        print()
