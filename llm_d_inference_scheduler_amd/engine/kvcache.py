"""Paged KV pool + block allocator (per-GPU worker engine).

Pool layout [L, 2, NB, KVH, BS, D] bf16 — one (layer, K/V, block) is a
contiguous MFMA-tile-aligned chunk consumed zero-repack by the decode
attention kernel and the xGMI transfer engine (kv_cache.hip
header). Sizing targets the 288 GB HBM3E budget: Llama-3-8B bf16 KV is
128 KB/token -> ~1.9M tokens of residency beside the 16 GB of weights.
"""
from typing import Dict, List, Optional

import torch

from ..models.configs import ModelConfig

BLOCK_SIZE = 16  # tokens per KV block (reference default, types.go:92)


class KVPool:
    def __init__(self, config: ModelConfig, num_blocks: int,
                 device: torch.device, dtype: torch.dtype = torch.bfloat16,
                 block_size: int = BLOCK_SIZE):
        self.cfg = config
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.device = device
        self.dtype = dtype
        self.tensor = torch.zeros(
            (config.num_layers, 2, num_blocks, config.num_kv_heads,
             block_size, config.head_dim), dtype=dtype, device=device)

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
    """Free-list allocator + per-sequence block tables."""

    def __init__(self, num_blocks: int, block_size: int = BLOCK_SIZE):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self.tables: Dict[str, List[int]] = {}
        self.seq_lens: Dict[str, int] = {}

    @property
    def free_blocks(self) -> int:
        return len(self._free)

    @property
    def usage(self) -> float:
        return 1.0 - len(self._free) / max(1, self.num_blocks)

    def can_allocate(self, n_tokens: int) -> bool:
        need = (n_tokens + self.block_size - 1) // self.block_size
        return need <= len(self._free)

    def allocate(self, seq_id: str, n_tokens: int) -> bool:
        """Ensure the sequence has capacity for n_tokens total."""
        table = self.tables.setdefault(seq_id, [])
        need = (n_tokens + self.block_size - 1) // self.block_size
        while len(table) < need:
            if not self._free:
                return False
            table.append(self._free.pop())
        self.seq_lens[seq_id] = max(self.seq_lens.get(seq_id, 0), 0)
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
            self._free.append(blk)
        self.seq_lens.pop(seq_id, None)

    def adopt(self, seq_id: str, blocks: List[int], seq_len: int) -> None:
        """Adopt externally-filled blocks (xGMI transfer receive path)."""
        self.tables[seq_id] = list(blocks)
        self.seq_lens[seq_id] = seq_len

    def take_blocks(self, n: int) -> Optional[List[int]]:
        if n > len(self._free):
            return None
        return [self._free.pop() for _ in range(n)]

    def release_blocks(self, blocks: List[int]) -> None:
        self._free.extend(blocks)
