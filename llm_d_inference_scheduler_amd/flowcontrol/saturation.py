"""Saturation detectors
(parity: pkg/epp/framework/plugins/flowcontrol/saturationdetector/{utilization,concurrency}).

utilization-detector: per-endpoint roofline max(queue/qThresh, kv/kvThresh);
pool = average; stale metrics score fully saturated ("Fail-Open Fallback" /
stale handling per the utilization README). Defaults: queue depth 5,
kv-cache utilization 0.8 (BASELINE.md row "Default saturation thresholds").
"""
import time
from typing import List

from ..datalayer.attributes import IN_FLIGHT_LOAD
from ..datalayer.endpoint import Endpoint
from ..metrics import prom


class SaturationDetector:
    type_name = "saturation"

    def saturation(self, endpoints: List[Endpoint]) -> float:
        raise NotImplementedError

    def is_saturated(self, endpoints: List[Endpoint]) -> bool:
        s = self.saturation(endpoints)
        prom.saturation_gauge.set(s)
        return s >= 1.0


class UtilizationSaturationDetector(SaturationDetector):
    type_name = "utilization-detector"

    def __init__(self, queue_threshold: float = 5.0,
                 kv_threshold: float = 0.8,
                 staleness_s: float = 0.5):
        self.queue_threshold = queue_threshold
        self.kv_threshold = kv_threshold
        self.staleness_s = staleness_s

    def endpoint_saturation(self, ep: Endpoint) -> float:
        m = ep.metrics
        if m.update_time and (time.monotonic() - m.update_time) > self.staleness_s:
            return 1.0  # stale metrics treated as 100% saturated
        return max(m.waiting_queue_size / self.queue_threshold,
                   m.kv_cache_usage / self.kv_threshold)

    def saturation(self, endpoints: List[Endpoint]) -> float:
        if not endpoints:
            return 1.0
        return sum(self.endpoint_saturation(ep) for ep in endpoints) / len(endpoints)

    # Also usable as a scheduling Filter with fail-open (utilization README):
    def filter(self, ctx, endpoints: List[Endpoint]) -> List[Endpoint]:
        keep = [ep for ep in endpoints if self.endpoint_saturation(ep) < 1.0]
        return keep or endpoints  # fail-open fallback


class ConcurrencySaturationDetector(SaturationDetector):
    """Aggregate in-flight fraction (saturationdetector/concurrency)."""
    type_name = "concurrency-detector"

    def __init__(self, max_inflight_per_endpoint: int = 256):
        self.max_inflight = max_inflight_per_endpoint

    def saturation(self, endpoints: List[Endpoint]) -> float:
        if not endpoints:
            return 1.0
        total = 0
        for ep in endpoints:
            load = ep.get_attribute(IN_FLIGHT_LOAD)
            total += load.snapshot()[0] if load else 0
        return total / (self.max_inflight * len(endpoints))
