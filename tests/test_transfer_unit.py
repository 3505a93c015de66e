"""KVTransferEngine unit semantics (parallel/transfer.py): handle
lifecycle, byte accounting, CPU-synchronous completion — no process group
(the world>1 paths are covered by the gloo tests in test_node.py)."""
import torch

from llm_d_inference_scheduler_amd.parallel.transfer import (KVTransferEngine,
                                                             TransferHandle)


def make_engine():
    pool = torch.zeros(2, 2, 16, 2, 4, 8)   # [L,2,NB,KVH,BS,D]
    return KVTransferEngine(pool, rank=0)


class TestTransferHandles:
    def test_handle_fires_once(self):
        fired = []
        h = TransferHandle(None, lambda: fired.append(1), 10, "send")
        assert h.poll() is True
        assert h.poll() is True
        assert fired == [1]

    def test_block_nbytes_matches_layout(self):
        eng = make_engine()
        # 2 layers * 2(K/V) * 2 heads * 4 rows * 8 dims * 4B fp32 = 1024/blk
        assert eng._block_nbytes(1) == 2 * 2 * 2 * 4 * 8 * 4
        assert eng._block_nbytes(3) == 3 * eng._block_nbytes(1)

    def test_cpu_transport_forced_rccl(self):
        eng = make_engine()
        assert eng.transport == "rccl"     # peer needs a GPU pool
        assert eng.inflight_bytes == 0

    def test_wire_view_fp8(self):
        t8 = torch.zeros(4, dtype=torch.float8_e4m3fn)
        assert KVTransferEngine._wire(t8).dtype == torch.uint8
        tb = torch.zeros(4, dtype=torch.bfloat16)
        assert KVTransferEngine._wire(tb).dtype == torch.bfloat16

    def test_local_copy_moves_blocks(self):
        a, b = make_engine(), make_engine()
        a.pool[:, :, 3] = 7.0
        b.local_copy(a, [3], [5])
        assert torch.equal(b.pool[:, :, 5], a.pool[:, :, 3])
        assert float(b.pool[:, :, 4].abs().sum()) == 0.0

    def test_staging_shape(self):
        eng = make_engine()
        s = eng._staging(3)
        assert tuple(s.shape) == (3, 2, 2, 2, 4, 8)
