"""Admitter plugins (parity: latency-slo admitter (169 LoC) and
probabilistic-admitter (162 LoC) under dataproducer/predictedlatency consumers,
SURVEY.md §2.9)."""
import random
from typing import Tuple

from ..datalayer.attributes import LATENCY_PREDICTION_INFO
from .interface import Admitter
from .registry import register_plugin


@register_plugin("latency-slo-admitter", aliases=["latency-slo"])
class LatencySLOAdmitter(Admitter):
    """Reject sheddable requests when no endpoint has positive predicted
    TTFT headroom; critical (priority >= threshold) always admitted."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.critical_priority = int(params.get("criticalPriority", 0))

    def admit(self, ctx, endpoints) -> Tuple[bool, str]:
        if ctx.request.priority >= self.critical_priority:
            return True, ""
        lat = ctx.attributes.get(LATENCY_PREDICTION_INFO)
        if lat is None or not lat.ttft_headroom_ms:
            return True, ""  # fail open without predictions
        if any(lat.ttft_headroom_ms.get(ep.name, -1) > 0 for ep in endpoints):
            return True, ""
        return False, "no endpoint within TTFT SLO"


@register_plugin("probabilistic-admitter")
class ProbabilisticAdmitter(Admitter):
    """Admit sheddable load with probability shrinking as headroom shrinks."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._rng = random.Random(params.get("seed", 0xAD))
        self.floor = float(params.get("floor", 0.05))

    def admit(self, ctx, endpoints) -> Tuple[bool, str]:
        if ctx.request.priority >= 0:
            return True, ""
        lat = ctx.attributes.get(LATENCY_PREDICTION_INFO)
        if lat is None or not lat.ttft_headroom_ms or lat.ttft_slo_ms is None:
            return True, ""
        best = max((lat.ttft_headroom_ms.get(ep.name, 0.0)
                    for ep in endpoints), default=0.0)
        p = max(self.floor, min(1.0, 0.5 + best / (2 * lat.ttft_slo_ms)))
        if self._rng.random() < p:
            return True, ""
        return False, "probabilistically shed under load"
