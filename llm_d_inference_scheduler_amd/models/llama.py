"""Llama-family model runner (random-init weights, paged KV, bf16).

MI355X-native execution path: plain-GEMM projections go through
hipBLASLt/rocBLAS via torch.matmul; the fused hot ops (residual+RMSNorm,
RoPE, SiLU*up, GQA paged decode attention, KV scatter) are the gfx950 HIP
kernels in csrc/hip dispatched through ops/. Prefill attention is composed
chunked torch-matmul attention over KV gathered from the paged pool (an
LDS-tiled MFMA flash kernel is the planned replacement; no Triton, no SDPA
multi-backend dispatch is used anywhere).

The reference router never runs a model — its pods do (vLLM). This runner
is the per-GPU worker engine's model, SURVEY.md §2.12 "worker shim".
"""
import math
import os
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch

from .. import ops
from ..ops import ref as ops_ref
from ..utils.logging import get_logger
from .configs import ModelConfig

log = get_logger("models.llama")

_TUNED_GEMM_LOADED = False


def _load_tuned_gemms() -> None:
    """Load offline hipBLASLt autotune results (tools/tune_gemms.py) so the
    skinny decode GEMMs use tuned solutions instead of the default pick
    (~2.6x off the weights-bound floor, profiles/r01_bench_kernel_stats_v3)."""
    global _TUNED_GEMM_LOADED
    if _TUNED_GEMM_LOADED:
        return
    _TUNED_GEMM_LOADED = True
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "..", "..", "profiles", "tuned_gemm_gfx950.csv")
    if os.environ.get("LLMD_NO_TUNED_GEMM", "0") == "1":
        # bisect knob: TunableOp is the prime suspect for the
        # multi-stream wedge (profiles/r02_notes.md)
        return
    if os.path.exists(path):
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(path)
        log.info("loaded tuned GEMM solutions", path=path)


@dataclass
class ForwardBatch:
    """One model invocation over a token-flattened batch.

    For decode: T == number of sequences, each contributing its newest token.
    For prefill: T == sum of chunk lengths; `seq_starts` delimits sequences.
    """
    input_ids: torch.Tensor          # [T] int64
    positions: torch.Tensor         # [T] int32
    slot_mapping: torch.Tensor      # [T] int64 into the paged pool
    is_decode: bool
    # decode-only:
    block_tables: Optional[torch.Tensor] = None   # [B, max_blocks] int32
    seq_lens: Optional[torch.Tensor] = None       # [B] int32 (incl. new token)
    max_seq_len: Optional[int] = None             # host-known max (no sync)
    # prefill-only:
    seq_starts: Optional[List[int]] = None        # len B+1 offsets into T
    ctx_lens: Optional[List[int]] = None          # [B] total ctx after chunk
    prefill_block_tables: Optional[List[torch.Tensor]] = None
    # fused flash-prefill metadata (GPU path; see flash_prefill.hip)
    prefill_bt: Optional[torch.Tensor] = None     # [S, max_blocks] int32
    prefill_meta: Optional[torch.Tensor] = None   # [S, 3] start,chunk,prior
    prefill_tiles: Optional[torch.Tensor] = None  # [n_tiles, 2] seq,vrow0
    logit_rows: Optional[torch.Tensor] = None     # rows needing logits
    # multimodal: rows whose input embedding is provided (encode hand-off)
    embed_rows: Optional[torch.Tensor] = None     # [n] int64 into T
    embed_values: Optional[torch.Tensor] = None   # [n, hidden]


class LlamaRunner:
    def __init__(self, config: ModelConfig, device: torch.device,
                 dtype: torch.dtype = torch.bfloat16, seed: int = 0):
        self.cfg = config
        self.device = device
        self.dtype = dtype
        self.scale = 1.0 / math.sqrt(config.head_dim)
        c = config
        if self.device.type == "cuda":
            _load_tuned_gemms()
            # init directly on the GPU (16 GB of weights for Llama-3-8B —
            # CPU-side init would dominate startup). Same seed => identical
            # weights on every rank (required by the P/D KV hand-off).
            torch.cuda.manual_seed_all(seed)

            def w(*shape, std=0.02):
                t = torch.empty(*shape, dtype=torch.float32,
                                device=self.device)
                t.normal_(0.0, std)
                return t.to(dtype)
        else:
            gen = torch.Generator(device="cpu").manual_seed(seed)

            def w(*shape, std=0.02):
                t = torch.empty(*shape, dtype=torch.float32)
                t.normal_(0.0, std, generator=gen)
                return t.to(dtype).to(device)

        out_std = 0.02 / math.sqrt(2 * c.num_layers)
        self.embed = w(c.vocab_size, c.hidden_size)
        self.layers: List[Dict[str, torch.Tensor]] = []
        for _ in range(c.num_layers):
            self.layers.append({
                "input_norm": torch.ones(c.hidden_size, dtype=dtype,
                                         device=device),
                "wqkv": w(c.hidden_size, c.q_size + 2 * c.kv_size),
                "wo": w(c.q_size, c.hidden_size, std=out_std),
                "post_norm": torch.ones(c.hidden_size, dtype=dtype,
                                        device=device),
                **({"q_norm": torch.ones(c.head_dim, dtype=dtype,
                                         device=device),
                    "k_norm": torch.ones(c.head_dim, dtype=dtype,
                                         device=device)}
                   if c.qk_norm else {}),
                "wgate_up": w(c.hidden_size, 2 * c.intermediate_size),
                "wdown": w(c.intermediate_size, c.hidden_size, std=out_std),
            })
        self.final_norm = torch.ones(c.hidden_size, dtype=dtype, device=device)
        self.lm_head = self.embed if c.tie_embeddings else \
            w(c.vocab_size, c.hidden_size)
        self.cos_sin = ops_ref.rope_table(c.max_position, c.head_dim,
                                          c.rope_theta, device=device)

    def weight_bytes(self) -> int:
        total = self.embed.numel() + self.lm_head.numel() + \
            self.final_norm.numel()
        for layer in self.layers:
            total += sum(t.numel() for t in layer.values())
        return total * self.embed.element_size()

    # ------------------------------------------------------------------
    def forward(self, batch: ForwardBatch, kv_pool: torch.Tensor,
                embeddings_out: bool = False) -> torch.Tensor:
        """Returns logits [n_logit_rows, vocab] (or final hidden states for
        embeddings_out). kv_pool: [L, 2, NB, KVH, BS, D]."""
        c = self.cfg
        hidden = self.embed[batch.input_ids]
        if batch.embed_rows is not None and batch.embed_rows.numel():
            hidden[batch.embed_rows] = batch.embed_values.to(hidden.dtype)
        residual = None
        pos32 = batch.positions.to(torch.int32)
        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = hidden.clone()
                normed = ops.rmsnorm(hidden, layer["input_norm"], c.rms_eps)
            else:
                normed = ops.rmsnorm(hidden, layer["input_norm"], c.rms_eps,
                                     residual=residual)
            qkv = normed @ layer["wqkv"]
            q, k, v = qkv.split([c.q_size, c.kv_size, c.kv_size], dim=-1)
            T = q.shape[0]
            q = q.view(T, c.num_heads, c.head_dim).contiguous()
            k = k.view(T, c.num_kv_heads, c.head_dim).contiguous()
            v = v.view(T, c.num_kv_heads, c.head_dim).contiguous()
            if c.qk_norm:
                # Qwen3 per-head QK-RMSNorm before RoPE: each head's
                # D-vector normalized with a learned gain (the rmsnorm
                # kernel treats [T*H, D] rows like any hidden dim)
                q = ops.rmsnorm(q.view(T * c.num_heads, c.head_dim),
                                layer["q_norm"],
                                c.rms_eps).view(T, c.num_heads, c.head_dim)
                k = ops.rmsnorm(k.view(T * c.num_kv_heads, c.head_dim),
                                layer["k_norm"],
                                c.rms_eps).view(T, c.num_kv_heads,
                                                c.head_dim)
            ops.rope(q, k, self.cos_sin, pos32)
            k_cache = kv_pool[li, 0]
            v_cache = kv_pool[li, 1]
            ops.reshape_and_cache(k, v, k_cache, v_cache, batch.slot_mapping)
            if batch.is_decode:
                attn = ops.paged_attention(q, k_cache, v_cache,
                                           batch.block_tables, batch.seq_lens,
                                           self.scale,
                                           max_seq_len=batch.max_seq_len)
            else:
                attn = self._prefill_attention(batch, q, k_cache, v_cache)
            o = attn.view(T, c.q_size) @ layer["wo"]
            normed2 = ops.rmsnorm(o, layer["post_norm"], c.rms_eps,
                                  residual=residual)
            gate_up = normed2 @ layer["wgate_up"]
            act = ops.silu_mul(gate_up)
            hidden = act @ layer["wdown"]
        final = ops.rmsnorm(hidden, self.final_norm, c.rms_eps,
                            residual=residual)
        # `residual` now holds the pre-norm sum; `final` the normed states
        if embeddings_out:
            return final
        rows = batch.logit_rows
        sel = final if rows is None else final[rows]
        return sel @ self.lm_head.t()

    def _prefill_attention(self, batch: ForwardBatch, q: torch.Tensor,
                           k_cache: torch.Tensor,
                           v_cache: torch.Tensor) -> torch.Tensor:
        """Causal varlen attention per sequence. GPU: fused MFMA flash
        kernel (flash_prefill.hip) — S never materialized. CPU fallback:
        composed chunked torch attention over pool-gathered KV."""
        c = self.cfg
        if (q.is_cuda and batch.prefill_tiles is not None
                and c.head_dim == 128
                and c.num_heads // c.num_kv_heads in (1, 2, 4, 8)):
            return ops.flash_prefill(
                q, k_cache, v_cache, batch.prefill_bt, batch.prefill_meta,
                batch.prefill_tiles, self.scale)
        qpg = c.num_heads // c.num_kv_heads
        out = torch.empty_like(q)
        starts = batch.seq_starts
        for i in range(len(starts) - 1):
            s, e = starts[i], starts[i + 1]
            chunk = e - s
            ctx = batch.ctx_lens[i]
            prior = ctx - chunk
            kk, vv = ops_ref.gather_prefix(k_cache, v_cache,
                                           batch.prefill_block_tables[i], ctx)
            # [KVH, ctx, D]
            kk = kk.permute(1, 0, 2)
            vv = vv.permute(1, 0, 2)
            qi = q[s:e].view(chunk, c.num_kv_heads, qpg,
                             c.head_dim).permute(1, 2, 0, 3)  # [KVH,qpg,chunk,D]
            scores = torch.einsum("hgtd,hsd->hgts", qi.float(), kk.float())
            scores *= self.scale
            # causal mask: query t (global pos prior+t) sees keys <= prior+t
            t_idx = torch.arange(chunk, device=q.device).view(1, 1, chunk, 1)
            s_idx = torch.arange(ctx, device=q.device).view(1, 1, 1, ctx)
            scores.masked_fill_(s_idx > t_idx + prior, float("-inf"))
            p = torch.softmax(scores, dim=-1)
            o = torch.einsum("hgts,hsd->hgtd", p, vv.float())
            out[s:e] = o.permute(2, 0, 1, 3).reshape(
                chunk, c.num_heads, c.head_dim).to(q.dtype)
        return out
