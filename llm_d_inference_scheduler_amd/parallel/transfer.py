"""xGMI KV/embedding transfer engine.

This is the MI355X-native replacement for the data movement the reference
only *triggers* (NIXL/UCX/RDMA `kv_transfer_params`, connector_nixlv2.go;
SURVEY.md §5.8): prefill-role and decode-role GPUs exchange whole KV blocks
directly over the node's xGMI links.

Mechanism: the gfx950 gather kernel (kv_cache.hip) packs the selected
blocks — already MFMA-tile-aligned, so the decode kernel consumes them with
zero repack — into one contiguous staging buffer; a single RCCL
send/recv (torch.distributed P2P over the "nccl"=RCCL backend) moves it
point-to-point (any P->D pair has a direct ~153 GB/s xGMI path); the
receiver's scatter kernel drops the blocks into its own pool. Transfers run
on a dedicated HIP stream so decode compute overlaps the copy; the
`kv_transfer_params` JSON contract becomes the in-process block-handle
messages carried by the mailbox (engine id -> rank, block IDs -> pool
indices). CPU/gloo path exists for hermetic multi-process tests.
"""
import time
from typing import List, Optional

import torch
import torch.distributed as dist

from .. import ops
from ..metrics import prom


class KVTransferEngine:
    def __init__(self, pool_tensor: torch.Tensor, rank: int,
                 group: Optional[object] = None,
                 device: Optional[torch.device] = None):
        self.pool = pool_tensor  # [L, 2, NB, KVH, BS, D]
        self.rank = rank
        self.group = group
        self.device = device or pool_tensor.device
        self.on_gpu = self.pool.is_cuda
        self.stream = torch.cuda.Stream(self.device) if self.on_gpu else None

    def _staging(self, n_blocks: int) -> torch.Tensor:
        L = self.pool.shape[0]
        return torch.empty((n_blocks, L, 2) + tuple(self.pool.shape[3:]),
                           dtype=self.pool.dtype, device=self.device)

    @staticmethod
    def _wire(t: torch.Tensor) -> torch.Tensor:
        """RCCL has no fp8 dtype: ship fp8 staging buffers as uint8."""
        return t.view(torch.uint8) if t.dtype == torch.float8_e4m3fn else t

    def send_blocks(self, dst_rank: int, block_ids: List[int]) -> None:
        t0 = time.monotonic()
        ids = torch.tensor(block_ids, dtype=torch.int32, device=self.device)
        staging = self._staging(len(block_ids))
        if self.on_gpu:
            with torch.cuda.stream(self.stream):
                ops.move_blocks(self.pool, staging, ids, is_scatter=False)
                dist.send(self._wire(staging), dst=dst_rank, group=self.group)
            self.stream.synchronize()
        else:
            ops.move_blocks(self.pool, staging, ids, is_scatter=False)
            dist.send(self._wire(staging), dst=dst_rank, group=self.group)
        nbytes = staging.numel() * staging.element_size()
        prom.xgmi_kv_transfer_bytes.labels("send").inc(nbytes)
        prom.xgmi_kv_transfer_seconds.observe(time.monotonic() - t0)

    def recv_blocks(self, src_rank: int, block_ids: List[int]) -> None:
        t0 = time.monotonic()
        ids = torch.tensor(block_ids, dtype=torch.int32, device=self.device)
        staging = self._staging(len(block_ids))
        if self.on_gpu:
            with torch.cuda.stream(self.stream):
                dist.recv(self._wire(staging), src=src_rank, group=self.group)
                ops.move_blocks(self.pool, staging, ids, is_scatter=True)
            self.stream.synchronize()
        else:
            dist.recv(self._wire(staging), src=src_rank, group=self.group)
            ops.move_blocks(self.pool, staging, ids, is_scatter=True)
        nbytes = staging.numel() * staging.element_size()
        prom.xgmi_kv_transfer_bytes.labels("recv").inc(nbytes)
        prom.xgmi_kv_transfer_seconds.observe(time.monotonic() - t0)

    # ---- embeddings (encode -> prefill hand-off, E/P/D) ----
    def send_tensor(self, dst_rank: int, t: torch.Tensor) -> None:
        dist.send(t.contiguous(), dst=dst_rank, group=self.group)
        prom.xgmi_kv_transfer_bytes.labels("send").inc(
            t.numel() * t.element_size())

    def recv_tensor(self, src_rank: int, shape, dtype) -> torch.Tensor:
        t = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(t, src=src_rank, group=self.group)
        prom.xgmi_kv_transfer_bytes.labels("recv").inc(
            t.numel() * t.element_size())
        return t

    def local_copy(self, src_engine: "KVTransferEngine",
                   src_blocks: List[int], dst_blocks: List[int]) -> None:
        """Same-GPU role pair (prefill-decode combined): direct pool copy."""
        src = torch.tensor(src_blocks, dtype=torch.long)
        dst = torch.tensor(dst_blocks, dtype=torch.long)
        self.pool[:, :, dst] = src_engine.pool[:, :, src]
