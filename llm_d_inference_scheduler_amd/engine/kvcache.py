"""Paged KV pool + prefix-caching block allocator (per-GPU worker engine).

Pool layout [L, 2, NB, KVH, BS, D] bf16 — one (layer, K/V, block) is a
contiguous MFMA-tile-aligned chunk consumed zero-repack by the decode
attention kernel and the xGMI transfer engine (kv_cache.hip
header). Sizing targets the 288 GB HBM3E budget: Llama-3-8B bf16 KV is
128 KB/token -> ~1.9M tokens of residency beside the 16 GB of weights.

The BlockManager implements engine-level automatic prefix caching: full
blocks are content-addressed by the same chained xxhash64 the router's
approx-prefix producer uses (reference approximateprefix/hashing.go:35-99 —
there it only *estimates* vLLM's cache; here the engine IS the model server,
so the cache is real). Freed blocks keep their content in an LRU and are
resurrected on hash hit; eviction happens only on reuse.
"""
from collections import OrderedDict
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from ..models.configs import ModelConfig

BLOCK_SIZE = 16  # tokens per KV block (reference default, types.go:92)


def block_hashes(prompt_tokens: Sequence[int], block_size: int = BLOCK_SIZE,
                 seed: int = 0x9E3779B97F4A7C15) -> np.ndarray:
    """Chained per-block content hashes of the full prompt blocks."""
    from .. import _router_core as rc
    toks = np.asarray(prompt_tokens, dtype=np.int32)
    n_full = len(toks) // block_size
    if n_full == 0:
        return np.zeros(0, dtype=np.uint64)
    return rc.hash_tokens(toks[:n_full * block_size], block_size, n_full,
                          seed)


class KVPool:
    """cache_dtype: torch.bfloat16 or torch.float8_e4m3fn (OCP fp8 — halves
    KV bytes and doubles effective decode-attention bandwidth; the gfx950
    kernels convert with the hardware v_cvt_*_fp8 ops, hip_common.h).

    fp8 limitation (documented, by design): K/V are stored as UNSCALED
    e4m3 — magnitudes >448 saturate and tiny values lose precision. This
    matches the bf16->e4m3 direct-cast mode (vLLM's fp8 KV without
    calibration scales); K/V activations of the supported model families
    are well inside e4m3 range at bf16 training scale, and kv_cache_dtype
    defaults to "auto" (=compute dtype). Outlier-heavy checkpoints should
    keep bf16 KV; per-layer scale support would require threading a scale
    through store (kv_cache.hip) and the attention dequant path."""

    def __init__(self, config: ModelConfig, num_blocks: int,
                 device: torch.device, dtype: torch.dtype = torch.bfloat16,
                 block_size: int = BLOCK_SIZE,
                 cache_dtype: torch.dtype = None,
                 ipc_alloc: bool = False):
        self.cfg = config
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.device = device
        self.dtype = dtype
        self.cache_dtype = cache_dtype or dtype
        shape = (config.num_layers, 2, num_blocks, config.num_kv_heads,
                 block_size, config.head_dim)
        self.ipc_backed = False
        if ipc_alloc:
            # dedicated hipMalloc (outside the caching allocator) so the
            # pool is IPC-shareable for the peer-pull transfer transport:
            # an IPC handle of a caching-allocator suballocation maps the
            # containing arena, not the pool, on the peer side
            from .. import ops as ops_mod
            like = torch.empty(0, dtype=self.cache_dtype, device=device)
            self.tensor = ops_mod.hip_ops().ipc_alloc_tensor(
                list(shape), like)
            self.ipc_backed = True
        else:
            self.tensor = torch.zeros(shape, dtype=self.cache_dtype,
                                      device=device)

    def layer(self, li: int):
        return self.tensor[li, 0], self.tensor[li, 1]

    @property
    def block_bytes(self) -> int:
        """Bytes of one block across all layers and K+V."""
        c = self.cfg
        return (c.num_layers * 2 * c.num_kv_heads * self.block_size *
                c.head_dim * self.tensor.element_size())

    @staticmethod
    def blocks_for_budget(config: ModelConfig, budget_bytes: int,
                          block_size: int = BLOCK_SIZE,
                          dtype_bytes: int = 2) -> int:
        per_block = (config.num_layers * 2 * config.num_kv_heads *
                     block_size * config.head_dim * dtype_bytes)
        return max(16, budget_bytes // per_block)


class BlockManager:
    """Refcounted free-list allocator + per-sequence block tables with
    content-addressed full-block prefix caching."""

    def __init__(self, num_blocks: int, block_size: int = BLOCK_SIZE,
                 prefix_caching: bool = True):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.prefix_caching = prefix_caching
        # all blocks start free and content-less; LRU order: oldest first
        self._free_lru: "OrderedDict[int, None]" = OrderedDict(
            (b, None) for b in range(num_blocks))
        self._refcnt = [0] * num_blocks
        self._hash_of: List[Optional[int]] = [None] * num_blocks
        self._by_hash: Dict[int, int] = {}
        self.tables: Dict[str, List[int]] = {}
        self.seq_lens: Dict[str, int] = {}
        # stats
        self.cached_tokens_total = 0
        self.queried_tokens_total = 0
        # KV event stream for the router's precise prefix index
        # (replaces vLLM's ZMQ KV events, precise_prefix_cache.go:655-671)
        self._ev_stored: List[int] = []
        self._ev_evicted: List[int] = []

    @property
    def free_blocks(self) -> int:
        return len(self._free_lru)

    @property
    def usage(self) -> float:
        return 1.0 - len(self._free_lru) / max(1, self.num_blocks)

    @property
    def hit_rate(self) -> float:
        q = self.queried_tokens_total
        return self.cached_tokens_total / q if q else 0.0

    # ---- free-list internals -------------------------------------------
    def _pop_free(self) -> int:
        blk, _ = self._free_lru.popitem(last=False)
        h = self._hash_of[blk]
        if h is not None:                      # evict stale cached content
            if self._by_hash.get(h) == blk:
                del self._by_hash[h]
                self._ev_evicted.append(h)
            self._hash_of[blk] = None
        self._refcnt[blk] = 1
        return blk

    def _release(self, blk: int) -> None:
        self._refcnt[blk] -= 1
        if self._refcnt[blk] == 0:
            # most-recently-freed = most likely reused: push to LRU tail
            self._free_lru[blk] = None

    # ---- prefix caching -------------------------------------------------
    def match_prefix(self, hashes: np.ndarray, prompt_len: int) -> int:
        """Longest cached full-block prefix, capped so at least one prompt
        token is always recomputed (its logits seed generation)."""
        if not self.prefix_caching:
            return 0
        max_blocks = min(len(hashes), (prompt_len - 1) // self.block_size)
        n = 0
        while n < max_blocks and int(hashes[n]) in self._by_hash:
            n += 1
        return n * self.block_size

    def allocate_prompt(self, seq_id: str, hashes: np.ndarray,
                        prompt_len: int) -> int:
        """Build the sequence's table reusing cached prefix blocks.
        Returns the number of cached (skippable) prompt tokens."""
        matched = self.match_prefix(hashes, prompt_len)
        table = self.tables.setdefault(seq_id, [])
        assert not table, "allocate_prompt on an existing sequence"
        for i in range(matched // self.block_size):
            blk = self._by_hash[int(hashes[i])]
            if self._refcnt[blk] == 0:          # resurrect from free LRU
                del self._free_lru[blk]
            self._refcnt[blk] += 1
            table.append(blk)
        self.seq_lens[seq_id] = 0
        self.queried_tokens_total += prompt_len
        self.cached_tokens_total += matched
        return matched

    def register_block(self, seq_id: str, block_idx: int, h: int) -> None:
        """Publish a fully-written block's content hash (prefill only —
        shared blocks are immutable: writers always own fresh blocks)."""
        if not self.prefix_caching:
            return
        blk = self.tables[seq_id][block_idx]
        if h in self._by_hash or self._hash_of[blk] is not None:
            return                              # first writer wins
        self._by_hash[h] = blk
        self._hash_of[blk] = h
        self._ev_stored.append(h)

    # ---- allocation ------------------------------------------------------
    def can_allocate(self, n_tokens: int,
                     seq_id: Optional[str] = None) -> bool:
        """Can the pool grow `seq_id` (or a fresh sequence) to n_tokens?
        Blocks the sequence already holds — including prefix-cache hits
        adopted at admission — count toward the requirement; pricing them
        as fresh allocations deadlocks a waiting sequence whose cached
        prefix is most of the pool (found by the engine-lifecycle
        property fuzz)."""
        need = (n_tokens + self.block_size - 1) // self.block_size
        if seq_id is not None:
            need -= len(self.tables.get(seq_id, ()))
        return need <= len(self._free_lru)

    def allocate(self, seq_id: str, n_tokens: int) -> bool:
        """Ensure the sequence has capacity for n_tokens total."""
        table = self.tables.setdefault(seq_id, [])
        need = (n_tokens + self.block_size - 1) // self.block_size
        if need - len(table) > len(self._free_lru):
            return False
        while len(table) < need:
            table.append(self._pop_free())
        self.seq_lens.setdefault(seq_id, 0)
        return True

    def append_token_slot(self, seq_id: str) -> Optional[int]:
        """Allocate room for one more token; returns its global slot."""
        cur = self.seq_lens[seq_id]
        if not self.allocate(seq_id, cur + 1):
            return None
        table = self.tables[seq_id]
        blk = table[cur // self.block_size]
        self.seq_lens[seq_id] = cur + 1
        return blk * self.block_size + cur % self.block_size

    def slots_for_range(self, seq_id: str, start: int, end: int) -> List[int]:
        table = self.tables[seq_id]
        return [table[p // self.block_size] * self.block_size +
                p % self.block_size for p in range(start, end)]

    def set_seq_len(self, seq_id: str, n: int) -> None:
        self.seq_lens[seq_id] = n

    def free(self, seq_id: str) -> None:
        for blk in self.tables.pop(seq_id, []):
            self._release(blk)
        self.seq_lens.pop(seq_id, None)

    def adopt(self, seq_id: str, blocks: List[int], seq_len: int) -> None:
        """Adopt externally-filled blocks (xGMI transfer receive path)."""
        self.tables[seq_id] = list(blocks)
        self.seq_lens[seq_id] = seq_len

    def take_blocks(self, n: int) -> Optional[List[int]]:
        if n > len(self._free_lru):
            return None
        return [self._pop_free() for _ in range(n)]

    def release_blocks(self, blocks: List[int]) -> None:
        for blk in blocks:
            self._release(blk)

    def drain_events(self):
        """(stored, evicted) content-hash batches since the last drain."""
        s, e = self._ev_stored, self._ev_evicted
        self._ev_stored, self._ev_evicted = [], []
        return s, e
