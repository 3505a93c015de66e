"""Decode paged-attention kernel microbenchmark (run on a GPU box).
Times the production kernel at serving shapes with/without sequence split.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.ops import hip_ops  # noqa: E402

ext = hip_ops()
KVH, D, BS, QPG = 8, 128, 16, 4
QH = KVH * QPG


def bench(B, ctx, np_=None, part=512, iters=50, v3=False):
    torch.manual_seed(0)
    max_blocks = (ctx + BS - 1) // BS
    NB = max_blocks * B + 1
    q = torch.randn(B, QH, D, device="cuda").bfloat16()
    kc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
    vc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
    bt = torch.arange(1, NB, dtype=torch.int32,
                      device="cuda").view(B, max_blocks)
    sl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    scale = D ** -0.5

    def run():
        if v3:
            return ext.paged_attention_v3(q, kc, vc, bt, sl, np_ or 1,
                                          part if np_ else ctx + 256, scale)
        if np_:
            return ext.paged_attention_split(q, kc, vc, bt, sl, np_, part,
                                             scale)
        return ext.paged_attention(q, kc, vc, bt, sl, scale)

    for _ in range(10):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        run()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    kv_gb = B * ctx * KVH * D * 2 * 2 / 1e9
    tag = " v3" if v3 else "   "
    print(f"B={B:4d} ctx={ctx:5d} np={np_ or 1:2d}{tag}: {us:8.1f} us  "
          f"({kv_gb / (us / 1e6) / 1e3:6.2f} TB/s effective)")


if __name__ == "__main__":
    for B in (64, 128, 256):
        bench(B, 1152)
        bench(B, 1152, v3=True)
        for np_ in (2, 4, 8):
            part = ((1152 + np_ - 1) // np_ + 63) // 64 * 64
            bench(B, 1152, np_=np_, part=part)
            bench(B, 1152, np_=np_, part=part, v3=True)
    bench(8, 8192, np_=16, part=512)
    bench(8, 8192, np_=16, part=512, v3=True)
    bench(128, 1536)
    bench(128, 1536, v3=True)
    for np_ in (2, 4):
        part = ((1536 + np_ - 1) // np_ + 63) // 64 * 64
        bench(128, 1536, np_=np_, part=part, v3=True)
