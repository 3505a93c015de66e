"""GPU prefix-cache index — device-resident hash table driven by the gfx950
kernels (csrc/hip/prefix_kernels.hip), mirroring the C++ PrefixIndex's LRU
policy decisions.

The host C++ index stays the source of truth for LRU/eviction bookkeeping
(cheap, and eviction policy is inherently sequential); this wrapper mirrors
its inserts/evictions into the device table so the batched admission-queue
match runs entirely on-GPU: one `hash_prompts` launch hashes every queued
request, one `match_longest` launch probes the table for all of them
(SURVEY.md §2.6 MI355X mapping).
"""
from typing import Optional

import numpy as np
import torch

from .. import _router_core as rc
from .dispatch import hip_ops


def _i64(u: int) -> int:
    """Reinterpret a python uint64 as the int64 the binding expects."""
    u &= (1 << 64) - 1
    return u - (1 << 64) if u >= (1 << 63) else u


class GpuPrefixIndex:
    def __init__(self, device, capacity_pow2: int = 1 << 21,
                 lru_capacity_per_endpoint: int = 131072):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        self.device = torch.device(device)
        self.ext = hip_ops()
        self.keys = torch.zeros(capacity_pow2, dtype=torch.uint64,
                                device=self.device)
        self.masks = torch.zeros(capacity_pow2, dtype=torch.uint64,
                                 device=self.device)
        # host-side twin drives LRU decisions; evictions mirror to device
        self.host = rc.PrefixIndex(lru_capacity_per_endpoint)

    def add(self, endpoint: int, hashes: np.ndarray) -> None:
        h = torch.from_numpy(np.ascontiguousarray(hashes, dtype=np.uint64))
        self.ext.table_update(self.keys, self.masks, h.to(self.device),
                              endpoint, False)

    def remove(self, endpoint: int, hashes: np.ndarray) -> None:
        h = torch.from_numpy(np.ascontiguousarray(hashes, dtype=np.uint64))
        self.ext.table_update(self.keys, self.masks, h.to(self.device),
                              endpoint, True)

    def hash_prompts_batch(self, token_lists, block_tokens: int,
                           max_blocks: int, seed0: int):
        """Hash a whole admission batch in one launch.
        Returns (hashes [R, max_blocks] uint64 on device, counts [R])."""
        lens = [len(t) for t in token_lists]
        flat = np.concatenate([np.asarray(t, dtype=np.int32)
                               for t in token_lists]) if token_lists else \
            np.zeros(0, dtype=np.int32)
        offsets = np.zeros(len(lens) + 1, dtype=np.int64)
        offsets[1:] = np.cumsum(lens)
        return self.ext.hash_prompts(
            torch.from_numpy(flat).to(self.device),
            torch.from_numpy(offsets).to(self.device),
            block_tokens, max_blocks, _i64(seed0))

    def match_batch(self, hashes: torch.Tensor, counts: torch.Tensor,
                    n_endpoints: int) -> torch.Tensor:
        """[R, n_endpoints] consecutive matched blocks, one launch."""
        return self.ext.match_longest(self.keys, self.masks, hashes, counts,
                                      n_endpoints)
