"""Decode paged-attention kernel microbenchmark (run on a GPU box).
Times the production kernel at serving shapes with/without sequence split,
against the pipelined v4 (producer/consumer waves, double-buffered logits).
Also checks v4 numerics against v1 before timing.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.ops import hip_ops  # noqa: E402

ext = hip_ops()
KVH, D, BS = 8, 128, 16


def make(B, ctx, qpg, fp8=False, seed=0):
    torch.manual_seed(seed)
    max_blocks = (ctx + BS - 1) // BS
    NB = max_blocks * B + 1
    q = torch.randn(B, KVH * qpg, D, device="cuda").bfloat16()
    kc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
    vc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
    if fp8:
        kc = kc.to(torch.float8_e4m3fn)
        vc = vc.to(torch.float8_e4m3fn)
    bt = torch.arange(1, NB, dtype=torch.int32,
                      device="cuda").view(B, max_blocks)
    sl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    return q, kc, vc, bt, sl


def check(kernel, qpg=4, fp8=False):
    fn = getattr(ext, f"paged_attention_{kernel}")
    for B, ctx, np_, part in ((4, 300, 1, 0), (8, 1152, 1, 0),
                              (8, 1152, 4, 320), (3, 70, 1, 0)):
        q, kc, vc, bt, sl = make(B, ctx, qpg, fp8)
        scale = D ** -0.5
        ref = ext.paged_attention(q, kc, vc, bt, sl, scale)
        got = fn(q, kc, vc, bt, sl, np_,
                 part if np_ > 1 else ctx + 256, scale)
        diff = (ref.float() - got.float()).abs().max().item()
        status = "OK" if diff < 3e-2 else "FAIL"
        print(f"{kernel} numerics qpg={qpg} fp8={int(fp8)} B={B} ctx={ctx} "
              f"np={np_}: max|d|={diff:.4f} {status}")


def bench(B, ctx, np_=None, part=512, iters=50, kernel="v1", qpg=4):
    q, kc, vc, bt, sl = make(B, ctx, qpg)
    scale = D ** -0.5

    def run():
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
    print(f"B={B:4d} ctx={ctx:5d} np={np_ or 1:2d} {kernel}: {us:8.1f} us  "
          f"({kv_gb / (us / 1e6) / 1e3:6.2f} TB/s effective)")


if __name__ == "__main__" and "--chunks" not in sys.argv:
    for B in (64, 128, 256):
        bench(B, 1152)
        for np_ in (2, 4):
            part = ((1152 + np_ - 1) // np_ + 255) // 256 * 256
            bench(B, 1152, np_=np_, part=part)
    bench(8, 8192, np_=16, part=512)
    bench(128, 1536)



def bench_chunk(B, ctx, chunk, iters=50, qpg=4):
    q, kc, vc, bt, sl = make(B, ctx, qpg)
    scale = D ** -0.5
    ref = ext.paged_attention(q, kc, vc, bt, sl, scale, 256)
    got = ext.paged_attention(q, kc, vc, bt, sl, scale, chunk)
    diff = (ref.float() - got.float()).abs().max().item()
    for _ in range(10):
        ext.paged_attention(q, kc, vc, bt, sl, scale, chunk)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext.paged_attention(q, kc, vc, bt, sl, scale, chunk)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    kv_gb = B * ctx * KVH * D * 2 * 2 / 1e9
    print(f"B={B:4d} ctx={ctx:5d} chunk={chunk}: {us:8.1f} us "
          f"({kv_gb / (us / 1e6) / 1e3:6.2f} TB/s) max|d|={diff:.4f}")


if __name__ == "__main__" and "--chunks" in sys.argv:
    for B in (64, 128, 256):
        for chunk in (256, 512):
            bench_chunk(B, 1152, chunk)
    bench_chunk(128, 2048, 256)
    bench_chunk(128, 2048, 512)
