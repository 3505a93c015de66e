"""Exercise each hand-written gfx950 kernel in isolation for a rocprofv3
--pmc counter pass (run on a GPU box under rocprofv3; see profiles/).

Order of dispatches (for attributing csv rows):
  1. hash_prompts (batched chained-xxhash prefix hashing)
  2. table_update (device prefix-table insert)
  3. match_longest (LDS-staged longest-prefix match)
  4. flash_prefill (MFMA prefill attention)
  5. paged_attention (GQA decode attention)
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from llm_d_inference_scheduler_amd import _router_core as rc  # noqa: E402
from llm_d_inference_scheduler_amd.ops import hip_ops  # noqa: E402
from llm_d_inference_scheduler_amd.ops.prefix import _i64  # noqa: E402

ext = hip_ops()
ITERS = 20

# --- prefix kernels: 256-request admission batch, 1024-token prompts ---
rng = np.random.default_rng(0)
toks = rng.integers(256, 100000, size=256 * 1024).astype(np.int32)
offsets = np.arange(0, 257 * 1024, 1024, dtype=np.int64)
seed0 = rc.model_seed("llama-3-8b", "")
tok_d = torch.from_numpy(toks).cuda()
off_d = torch.from_numpy(offsets).cuda()
for _ in range(ITERS):
    hashes, counts = ext.hash_prompts(tok_d, off_d, 16, 256, _i64(seed0))
cap = 1 << 20
keys = torch.zeros(cap, dtype=torch.uint64, device="cuda")
masks = torch.zeros(cap, dtype=torch.uint64, device="cuda")
flat = hashes.view(-1)[:4096].contiguous()
for e in range(8):
    ext.table_update(keys, masks, flat, e, False)
for _ in range(ITERS):
    match = ext.match_longest(keys, masks, hashes, counts, 8)
torch.cuda.synchronize()

# --- attention kernels: serving shapes ---
KVH, D, BS, qpg = 8, 128, 16, 4
QH = KVH * qpg
B, ctx = 256, 1152
max_blocks = (ctx + BS - 1) // BS
NB = max_blocks * B + 1
q = torch.randn(B, QH, D, device="cuda").bfloat16()
kc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
vc = torch.randn(NB, KVH, BS, D, device="cuda").bfloat16()
bt = torch.arange(1, NB, dtype=torch.int32,
                  device="cuda").view(B, max_blocks)
sl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
for _ in range(ITERS):
    ext.paged_attention(q, kc, vc, bt, sl, D ** -0.5)
chunk = 4096
qp = torch.randn(chunk, QH, D, device="cuda").bfloat16()
meta = torch.tensor([[0, chunk, 0]], dtype=torch.int32, device="cuda")
tiles = torch.tensor([(0, v) for v in range(0, chunk * qpg, 128)],
                     dtype=torch.int32, device="cuda")
bt1 = torch.arange(1, (chunk + BS - 1) // BS + 1, dtype=torch.int32,
                   device="cuda").view(1, -1)
for _ in range(ITERS):
    ext.flash_prefill(qp, kc, vc, bt1, meta, tiles, D ** -0.5)
torch.cuda.synchronize()
print("pmc exercise done")
