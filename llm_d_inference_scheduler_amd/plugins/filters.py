"""Filter plugins (parity: pkg/epp/framework/plugins/scheduling/filter/*).

Role filters key on the `llm-d.ai/role` label exactly like
filter/bylabel/roles.go:9-48; on this node the label lives on the GPU-role
endpoint instead of a pod.
"""
from typing import Dict, List

from ..datalayer.endpoint import CONTEXT_LENGTH_RANGE_LABEL, Endpoint, Role
from ..datalayer.attributes import LATENCY_PREDICTION_INFO, PREFIX_CACHE_MATCH_INFO
from ..scheduling.types import SchedulingContext
from .interface import Filter
from .registry import register_plugin


class _RoleFilter(Filter):
    role: Role = Role.DECODE

    def filter(self, ctx, endpoints: List[Endpoint]) -> List[Endpoint]:
        r = int(self.role)
        return [ep for ep in endpoints if ep.metadata._mask() & r]


@register_plugin("decode-filter", aliases=["decode_filter"])
class DecodeFilter(_RoleFilter):
    role = Role.DECODE


@register_plugin("prefill-filter", aliases=["prefill_filter"])
class PrefillFilter(_RoleFilter):
    role = Role.PREFILL


@register_plugin("encode-filter", aliases=["encode_filter"])
class EncodeFilter(_RoleFilter):
    role = Role.ENCODE


@register_plugin("label-selector-filter", aliases=["by-label-selector"])
class LabelSelectorFilter(Filter):
    """K8s-style label selector: parameters.selector = {key: value} with
    set-style values "a|b" allowed (filter/bylabel)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.selector: Dict[str, str] = params.get("selector", {})

    def filter(self, ctx, endpoints: List[Endpoint]) -> List[Endpoint]:
        out = []
        for ep in endpoints:
            ok = True
            for k, v in self.selector.items():
                allowed = str(v).split("|")
                if ep.metadata.labels.get(k) not in allowed:
                    ok = False
                    break
            if ok:
                out.append(ep)
        return out


@register_plugin("prefix-cache-affinity-filter")
class PrefixCacheAffinityFilter(Filter):
    """Sticky-set pre-filter (filter/prefixcacheaffinity): when some
    endpoints have a strong prefix match, keep only those; back off (keep
    all) when the predicted TTFT of the sticky set exceeds the SLO gate."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.min_ratio = float(params.get("minMatchRatio", 0.5))
        self.ttft_backoff_ms = params.get("ttftBackoffMs")

    def filter(self, ctx: SchedulingContext,
               endpoints: List[Endpoint]) -> List[Endpoint]:
        info = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        if info is None or info.total_blocks == 0:
            return endpoints
        sticky = [ep for ep in endpoints
                  if info.ratio(ep.name) >= self.min_ratio]
        if not sticky:
            return endpoints
        if self.ttft_backoff_ms is not None:
            lat = ctx.attributes.get(LATENCY_PREDICTION_INFO)
            if lat is not None:
                preds = [lat.predicted_ttft_ms.get(ep.name) for ep in sticky]
                preds = [p for p in preds if p is not None]
                if preds and min(preds) > float(self.ttft_backoff_ms):
                    return endpoints  # TTFT back-off gate: fail open
        return sticky


@register_plugin("slo-headroom-tier-filter")
class SLOHeadroomTierFilter(Filter):
    """Positive/negative headroom tiers with epsilon exploration
    (filter/sloheadroomtier): prefer endpoints whose predicted latency
    leaves positive SLO headroom; with probability epsilon keep all."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.epsilon = float(params.get("epsilon", 0.01))
        import random
        self._rng = random.Random(params.get("seed", 0xD15A))

    def filter(self, ctx: SchedulingContext,
               endpoints: List[Endpoint]) -> List[Endpoint]:
        lat = ctx.attributes.get(LATENCY_PREDICTION_INFO)
        if lat is None or not lat.ttft_headroom_ms:
            return endpoints
        if self._rng.random() < self.epsilon:
            return endpoints  # exploration
        positive = [ep for ep in endpoints
                    if lat.ttft_headroom_ms.get(ep.name, -1.0) > 0
                    and lat.tpot_headroom_ms.get(ep.name, 0.0) >= 0]
        return positive or endpoints  # fail open to the negative tier


@register_plugin("header-based-testing-filter")
class HeaderBasedTestingFilter(Filter):
    """Conformance/test-only filter (test/filter/): keep the endpoint named
    by the `test-epp-endpoint-selection` request header."""

    HEADER = "test-epp-endpoint-selection"

    def filter(self, ctx: SchedulingContext,
               endpoints: List[Endpoint]) -> List[Endpoint]:
        want = ctx.request.headers.get(self.HEADER)
        if not want:
            return endpoints
        names = set(want.split(","))
        return [ep for ep in endpoints
                if ep.name in names or ep.metadata.address in names]


@register_plugin("utilization-detector")
class UtilizationDetectorPlugin(Filter):
    """The utilization saturation detector registered as a plugin: usable
    from EndpointPickerConfig both as the flow-control saturation signal
    and as a scheduling Filter with fail-open fallback
    (saturationdetector/utilization/README.md)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        from ..flowcontrol.saturation import UtilizationSaturationDetector
        self.detector = UtilizationSaturationDetector(
            queue_threshold=float(params.get("queueDepthThreshold", 5.0)),
            kv_threshold=float(params.get("kvCacheUtilThreshold", 0.8)),
            staleness_s=float(params.get("metricsStalenessSeconds", 0.5)))

    def filter(self, ctx, endpoints):
        return self.detector.filter(ctx, endpoints)

    def is_saturated(self, endpoints) -> bool:
        return self.detector.is_saturated(endpoints)


@register_plugin("concurrency-detector")
class ConcurrencyDetectorPlugin(Filter):
    """Aggregate in-flight-fraction saturation detector
    (saturationdetector/concurrency); passes endpoints through unchanged
    when used in a profile."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        from ..flowcontrol.saturation import ConcurrencySaturationDetector
        self.detector = ConcurrencySaturationDetector(
            max_inflight_per_endpoint=int(params.get(
                "maxInflightPerEndpoint", 256)))

    def filter(self, ctx, endpoints):
        return endpoints

    def is_saturated(self, endpoints) -> bool:
        return self.detector.is_saturated(endpoints)


@register_plugin("destination-endpoint-served-verifier")
class DestinationEndpointServedVerifier(Filter):
    """Conformance plugin (reference
    .../requestcontrol/test/responsereceived, registered in the production
    runner for IGW conformance, runner.go:496-499): verifies that the
    endpoint which SERVED the response (`x-gateway-destination-endpoint-
    served` response header) matches the scheduler's pick; counts
    mismatches for the conformance harness. As a Filter it passes
    endpoints through unchanged."""

    SERVED_HEADER = "x-gateway-destination-endpoint-served"

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.checked = 0
        self.mismatches = 0

    def filter(self, ctx, endpoints):
        return endpoints

    def response_received(self, ctx, target, headers) -> None:
        served = (headers or {}).get(self.SERVED_HEADER, "")
        if not served:
            return
        self.checked += 1
        if target is not None and served not in (target.name,
                                                 target.metadata.address):
            self.mismatches += 1
