"""Peer-pull KV block copy microbenchmark (run on a GPU box, optionally
under rocprofv3 --stats for the kernel evidence).

Times copy_blocks_peer (kv_cache.hip) — the one-sided gather the decode
rank runs over the prefill rank's IPC-mapped pool — at P/D hand-off sizes
(Llama-3-8B: 64 blocks = one 1024-token request = 128 MB across layers).
On a 1-GPU box both pools live on the same device, so the figure is the
HBM-to-HBM ceiling of the kernel; across GPUs the same kernel streams over
the direct xGMI link (~153 GB/s pair bound).
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from llm_d_inference_scheduler_amd.ops import hip_ops  # noqa: E402

ext = hip_ops()
L, KVH, BS, D = 32, 8, 16, 128  # Llama-3-8B KV geometry


def main():
    NB = 4096
    like = torch.empty(0, dtype=torch.bfloat16, device="cuda")
    src = ext.ipc_alloc_tensor([L, 2, NB, KVH, BS, D], like)
    dst = torch.zeros(L, 2, NB, KVH, BS, D, dtype=torch.bfloat16,
                      device="cuda")
    src.normal_()
    block_bytes = L * 2 * KVH * BS * D * 2
    for n in (4, 16, 64, 256, 1024):
        src_ids = torch.randperm(NB, device="cuda")[:n].to(torch.int32)
        dst_ids = torch.randperm(NB, device="cuda")[:n].to(torch.int32)
        for _ in range(5):
            ext.copy_blocks_peer(src.data_ptr(), dst, src_ids, dst_ids, NB)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            ext.copy_blocks_peer(src.data_ptr(), dst, src_ids, dst_ids, NB)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / iters * 1e6
        gb = n * block_bytes / 1e9
        print(f"n_blocks={n:5d} ({gb*1e3:7.1f} MB): {us:9.1f} us  "
              f"{gb / (us / 1e6):7.1f} GB/s (rd+wr {2*gb/(us/1e6):7.1f})")
    # correctness spot check
    s = torch.tensor([7, 0, 41], dtype=torch.int32, device="cuda")
    d = torch.tensor([2, 30, 11], dtype=torch.int32, device="cuda")
    ext.copy_blocks_peer(src.data_ptr(), dst, s, d, NB)
    torch.cuda.synchronize()
    assert torch.equal(dst[:, :, d.long()], src[:, :, s.long()])
    print("numerics OK")


if __name__ == "__main__":
    main()
