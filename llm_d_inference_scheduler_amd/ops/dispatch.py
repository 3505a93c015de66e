"""Op dispatch: gfx950 HIP kernels on GPU tensors, torch reference on CPU.

Policy (driver contract): on a GPU box the HIP extension is REQUIRED — a
CUDA-device tensor with `_hip_ops` missing raises immediately rather than
silently falling back to eager torch. CPU tensors (tests, tiny gloo runs)
use ops.ref.
"""
from typing import Optional

import torch

from . import ref

try:
    from .. import _hip_ops as _ext
except ImportError:  # extension not built
    _ext = None


def hip_ops():
    """The raw extension module (router prefix kernels etc.). Fail-loud."""
    if _ext is None:
        raise RuntimeError(
            "_hip_ops extension is not built; run "
            "`python setup.py build_ext --inplace` (gfx950 HIP kernels are "
            "mandatory on GPU hosts — no eager fallback)")
    return _ext


def _use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    hip_ops()  # raises if missing on a GPU path
    return True


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float,
            residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    if _use_hip(x):
        return _ext.rmsnorm(x, w, eps, residual)
    return ref.rmsnorm(x, w, eps, residual)


def rope(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
         positions: torch.Tensor) -> None:
    if _use_hip(q):
        _ext.rope(q, k, cos_sin, positions.to(torch.int32))
        return
    ref.rope(q, k, cos_sin, positions)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if _use_hip(gate_up):
        return _ext.silu_mul(gate_up)
    return ref.silu_mul(gate_up)


def reshape_and_cache(k_new: torch.Tensor, v_new: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slots: torch.Tensor) -> None:
    if _use_hip(k_new):
        _ext.reshape_and_cache(k_new, v_new, k_cache, v_cache,
                               slots.to(torch.int64))
        return
    ref.reshape_and_cache(k_new, v_new, k_cache, v_cache, slots)


import os as _os

# online-softmax chunk of the decode kernel: 512 halves barrier count
# (bisect knob; 256 is the measured default)
_ATTN_CHUNK = int(_os.environ.get("LLMD_ATTN_CHUNK", "256"))


def paged_attention(q: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_tables: torch.Tensor,
                    seq_lens: torch.Tensor, scale: float,
                    max_seq_len: Optional[int] = None) -> torch.Tensor:
    if _use_hip(q):
        bt = block_tables.to(torch.int32)
        sl = seq_lens.to(torch.int32)
        # flash-decoding sequence split: the plain kernel launches only
        # B*KVH workgroups — far short of the 256-CU chip's >=2 WG/CU need.
        # Partition long sequences so the grid fills the chip; a combine
        # kernel merges the unnormalized partials.
        n_wgs = q.shape[0] * k_cache.shape[1]
        if max_seq_len is not None and n_wgs < 1024 and max_seq_len > 512:
            target_np = min(64, max(1, 1024 // max(1, n_wgs)))
            # partition size: ceil-divide then round up to the kernel's
            # 256-token online-softmax chunk
            c = _ATTN_CHUNK
            part = max(c, ((max_seq_len + target_np - 1) // target_np
                           + c - 1) // c * c)
            np_ = (max_seq_len + part - 1) // part
            if np_ > 1:
                return _ext.paged_attention_split(q, k_cache, v_cache, bt,
                                                  sl, np_, part, scale,
                                                  _ATTN_CHUNK)
        return _ext.paged_attention(q, k_cache, v_cache, bt, sl, scale,
                                    _ATTN_CHUNK)
    return ref.paged_attention(q, k_cache, v_cache, block_tables, seq_lens,
                               scale)


def flash_prefill(q: torch.Tensor, k_cache: torch.Tensor,
                  v_cache: torch.Tensor, block_tables: torch.Tensor,
                  seq_meta: torch.Tensor, tiles: torch.Tensor,
                  scale: float) -> torch.Tensor:
    """Fused MFMA causal varlen prefill attention (GPU-only; callers fall
    back to the composed reference path when unsupported)."""
    return hip_ops().flash_prefill(q, k_cache, v_cache, block_tables,
                                   seq_meta, tiles, scale)


def move_blocks(pool: torch.Tensor, staging: torch.Tensor,
                block_ids: torch.Tensor, is_scatter: bool) -> None:
    if _use_hip(pool):
        _ext.move_blocks(pool, staging, block_ids.to(torch.int32), is_scatter)
        return
    # CPU reference: pool [L, 2, NB, KVH, BS, D]; staging [n, L, 2, KVH, BS, D]
    ids = block_ids.long()
    if is_scatter:
        pool[:, :, ids] = staging.permute(1, 2, 0, 3, 4, 5)
    else:
        staging.copy_(pool[:, :, ids].permute(2, 0, 1, 3, 4, 5))
