"""Offline hipBLASLt GEMM autotune for the serving shapes (run on a GPU
box). Writes profiles/tuned_gemm_gfx950.csv, which LlamaRunner loads at
startup (torch.cuda.tunable.read_file) so production runs use the tuned
solutions without paying tuning stalls.

The decode-step GEMMs are skinny (M = decode batch): hipBLASLt's default
pick launches only ~100-400 workgroups on a 256-CU chip and lands ~2.6x
off the weights-bound floor (profiles/r01_bench_kernel_stats_v3.md).
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch
import torch.cuda.tunable as tunable

OUT = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/tuned_gemm.csv"

# Llama-3-8B layer shapes: (K, N) for x[M,K] @ W[K,N]
LAYER_SHAPES = [
    (4096, 6144),     # wqkv
    (4096, 4096),     # wo
    (4096, 28672),    # wgate_up
    (14336, 4096),    # wdown
]
LOGITS = (4096, 128256)
# decode batches (engine pads nothing; cover the common sizes) + prefill
MS_DECODE = [8, 16, 24, 32, 48, 64, 96, 128, 192, 256]
MS_PREFILL = [1024, 2048, 4096, 8192]
MS_LOGITS = [1, 2, 4, 8, 16, 32, 64, 128]


def main():
    assert torch.cuda.is_available()
    tunable.enable(True)
    tunable.tuning_enable(True)
    tunable.set_max_tuning_duration(50)    # ms per candidate solution
    tunable.set_filename(OUT)
    dev = torch.device("cuda:0")
    shapes = [(m, k, n, False) for m in MS_DECODE + MS_PREFILL
              for (k, n) in LAYER_SHAPES]
    # logits GEMM is x @ lm_head.t() (TransB) — tune that exact layout
    shapes += [(m, LOGITS[0], LOGITS[1], True) for m in MS_LOGITS]
    for i, (m, k, n, tb) in enumerate(shapes):
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        if tb:
            w = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
            _ = a @ w.t()
        else:
            b = torch.randn(k, n, device=dev, dtype=torch.bfloat16)
            _ = a @ b
        torch.cuda.synchronize()
        print(f"[{i + 1}/{len(shapes)}] tuned M={m} K={k} N={n} tb={tb}",
              flush=True)
    # results are flushed to OUT (set_filename) at interpreter exit
    print("tuning done; results flush to", OUT, "at exit")


if __name__ == "__main__":
    main()
