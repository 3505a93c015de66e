"""NodeTopology spec parsing — the GPU-role layout the driver's pd/epd
runs depend on (parallel/topology.py; role labels mirror
filter/bylabel/roles.go values)."""
import pytest

from llm_d_inference_scheduler_amd.datalayer.endpoint import Role
from llm_d_inference_scheduler_amd.parallel.topology import NodeTopology


class TestTopologyParse:
    def test_mono_all_decode(self):
        t = NodeTopology.parse("mono", 8)
        assert t.ranks_with(Role.DECODE) == list(range(8))
        assert t.ranks_with(Role.PREFILL) == []

    def test_pd_combined(self):
        t = NodeTopology.parse("pd-combined", 4)
        assert t.ranks_with(Role.DECODE) == list(range(4))
        assert t.ranks_with(Role.PREFILL) == list(range(4))
        assert t.ranks[0].role_label == "prefill-decode"

    def test_pd_2p6d_is_driver_8gpu_shape(self):
        t = NodeTopology.parse("pd:2p6d", 8)
        assert t.ranks_with(Role.PREFILL) == [0, 1]
        assert t.ranks_with(Role.DECODE) == [2, 3, 4, 5, 6, 7]
        assert t.ranks[0].role_label == "prefill"
        assert t.ranks[7].role_label == "decode"

    def test_epd_1e2p5d(self):
        t = NodeTopology.parse("epd:1e2p5d", 8)
        assert t.ranks_with(Role.ENCODE) == [0]
        assert t.ranks_with(Role.PREFILL) == [1, 2]
        assert t.ranks_with(Role.DECODE) == [3, 4, 5, 6, 7]
        assert t.ranks[0].role_label == "encode"

    def test_count_mismatch_rejected(self):
        with pytest.raises(ValueError, match="wants 3 ranks"):
            NodeTopology.parse("pd:1p2d", 8)

    def test_garbage_rejected(self):
        for bad in ("pd:", "xyz", "pd:1x1d", "epd:pd"):
            with pytest.raises(ValueError):
                NodeTopology.parse(bad, 2)

    def test_case_and_whitespace_tolerant(self):
        t = NodeTopology.parse("  PD:1P1D ", 2)
        assert t.ranks_with(Role.PREFILL) == [0]

    def test_repeated_segments_accumulate(self):
        t = NodeTopology.parse("pd:1p1d1p1d", 4)
        # segments accumulate by kind: 2 prefill then 2 decode
        assert t.ranks_with(Role.PREFILL) == [0, 1]
        assert t.ranks_with(Role.DECODE) == [2, 3]
