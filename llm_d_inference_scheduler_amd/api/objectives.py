"""InferenceObjective (parity: apix/v1alpha2/inferenceobjective_types.go:58-79).

`spec.priority` — higher = more critical; requests with priority < 0 are
sheddable under saturation (requestcontrol/admission.go). On the single-node
build objectives are local config objects keyed by name, attached to a
request via the `x-gateway-inference-objective` header or request field.
"""
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class InferenceObjective:
    name: str
    priority: int = 0
    pool_ref: str = "node-pool"
    # SLO targets consumed by the latency-aware plugins (the reference keeps
    # them in per-request headers; a named objective may carry defaults)
    ttft_slo_ms: Optional[float] = None
    tpot_slo_ms: Optional[float] = None
    labels: dict = field(default_factory=dict)

    @property
    def sheddable(self) -> bool:
        return self.priority < 0
