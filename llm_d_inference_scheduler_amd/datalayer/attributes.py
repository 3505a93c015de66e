"""Typed endpoint/request attributes (parity: pkg/epp/framework/plugins/datalayer/attribute/*).

Reference attribute keys: prefix.PrefixCacheMatchInfo, concurrency.InFlightLoad,
latency.LatencyPredictionInfo.
"""
import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

PREFIX_CACHE_MATCH_INFO = "prefix.PrefixCacheMatchInfo"
IN_FLIGHT_LOAD = "concurrency.InFlightLoad"
LATENCY_PREDICTION_INFO = "latency.LatencyPredictionInfo"


@dataclass
class PrefixCacheMatchInfo:
    """Per-request, per-endpoint prefix match (scorer/prefix/plugin.go:96-124)."""
    match_blocks: Dict[str, int] = field(default_factory=dict)  # endpoint name -> blocks
    total_blocks: int = 0
    block_size_tokens: int = 16

    def ratio(self, endpoint_name: str) -> float:
        if self.total_blocks <= 0:
            return 0.0
        return self.match_blocks.get(endpoint_name, 0) / self.total_blocks


class InFlightLoad:
    """Per-endpoint in-flight request/token counters
    (dataproducer/inflightload, attribute/concurrency)."""

    def __init__(self):
        self._lock = threading.Lock()
        self.requests = 0
        self.tokens = 0

    def add(self, requests: int, tokens: int) -> None:
        with self._lock:
            self.requests += requests
            self.tokens += tokens

    def snapshot(self):
        with self._lock:
            return self.requests, self.tokens


@dataclass
class LatencyPredictionInfo:
    """Per-request, per-endpoint predicted TTFT/TPOT + SLO headroom
    (attribute/latency; dataproducer/predictedlatency)."""
    predicted_ttft_ms: Dict[str, float] = field(default_factory=dict)
    predicted_tpot_ms: Dict[str, float] = field(default_factory=dict)
    ttft_headroom_ms: Dict[str, float] = field(default_factory=dict)
    tpot_headroom_ms: Dict[str, float] = field(default_factory=dict)
    ttft_slo_ms: Optional[float] = None
    tpot_slo_ms: Optional[float] = None
