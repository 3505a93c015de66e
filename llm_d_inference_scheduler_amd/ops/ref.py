"""Reference torch implementations of the gfx950 kernels.

Used (a) as the CPU execution path (tiny configs in tests / gloo multi-proc
runs) and (b) as the fp32 ground truth the GPU numerics tests compare the
HIP kernels against (tests/test_gpu_kernels.py). Math is fp32 regardless of
input dtype, mirroring the kernels' internal precision.
"""
import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float,
            residual: torch.Tensor = None):
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
        residual.copy_(xf.to(residual.dtype))
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * w.float()
    return y.to(x.dtype)


def rope_table(max_pos: int, head_dim: int, theta: float,
               device="cpu") -> torch.Tensor:
    """cos/sin table [max_pos, head_dim/2, 2] fp32 (host-precomputed)."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv_freq)
    table = torch.stack([torch.cos(ang), torch.sin(ang)], dim=-1)
    return table.to(torch.float32).contiguous().to(device)


def rope(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
         positions: torch.Tensor) -> None:
    """In-place NeoX rotate-half RoPE. q: [T, QH, D], k: [T, KVH, D]."""
    half = q.shape[-1] // 2
    cs = cos_sin[positions.long()]          # [T, half, 2]
    cos = cs[..., 0].unsqueeze(1)           # [T, 1, half]
    sin = cs[..., 1].unsqueeze(1)
    for t in (q, k):
        tf = t.float()
        x1, x2 = tf[..., :half], tf[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        t.copy_(torch.cat([o1, o2], dim=-1).to(t.dtype))


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    inter = gate_up.shape[-1] // 2
    g = gate_up[..., :inter].float()
    u = gate_up[..., inter:].float()
    return (g * torch.sigmoid(g) * u).to(gate_up.dtype)


def reshape_and_cache(k_new: torch.Tensor, v_new: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slots: torch.Tensor) -> None:
    """k_new/v_new: [T, KVH, D]; caches: [NB, KVH, BS, D]; slots: [T]."""
    bs = k_cache.shape[2]
    for t in range(k_new.shape[0]):
        s = int(slots[t])
        if s < 0:
            continue
        blk, row = s // bs, s % bs
        k_cache[blk, :, row, :] = k_new[t]
        v_cache[blk, :, row, :] = v_new[t]


def paged_attention(q: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_tables: torch.Tensor,
                    seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """Decode GQA attention. q: [B, QH, D] -> out [B, QH, D]."""
    B, QH, D = q.shape
    KVH, BS = k_cache.shape[1], k_cache.shape[2]
    qpg = QH // KVH
    out = torch.empty_like(q)
    for b in range(B):
        S = int(seq_lens[b])
        nb = (S + BS - 1) // BS
        blocks = block_tables[b, :nb].long()
        k = k_cache[blocks]                # [nb, KVH, BS, D]
        v = v_cache[blocks]
        k = k.permute(1, 0, 2, 3).reshape(KVH, nb * BS, D)[:, :S].float()
        v = v.permute(1, 0, 2, 3).reshape(KVH, nb * BS, D)[:, :S].float()
        qb = q[b].float().view(KVH, qpg, D)
        scores = torch.einsum("hgd,hsd->hgs", qb, k) * scale
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hgs,hsd->hgd", p, v)
        out[b] = o.reshape(QH, D).to(q.dtype)
    return out


def gather_prefix(k_cache: torch.Tensor, v_cache: torch.Tensor,
                  block_table: torch.Tensor, seq_len: int):
    """Gather one sequence's K/V [S, KVH, D] from the paged pool."""
    BS = k_cache.shape[2]
    nb = (seq_len + BS - 1) // BS
    blocks = block_table[:nb].long()
    k = k_cache[blocks].permute(0, 2, 1, 3)   # [nb, BS, KVH, D]
    v = v_cache[blocks].permute(0, 2, 1, 3)
    S = nb * BS
    k = k.reshape(S, k.shape[2], k.shape[3])[:seq_len]
    v = v.reshape(S, v.shape[2], v.shape[3])[:seq_len]
    return k, v
