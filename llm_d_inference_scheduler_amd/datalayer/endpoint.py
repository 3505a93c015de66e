"""Endpoint model (parity: pkg/epp/framework/interface/datalayer/endpoint.go:25-100).

An endpoint = one GPU-role worker on the node (the reference's model-server
pod). `Endpoint = Metadata + Metrics + AttributeMap` with atomic swap of the
metrics snapshot (reference uses an atomic pointer, endpoint.go:80 — here a
single reference assignment, atomic under the GIL, plus a lock for the
attribute map).
"""
import enum
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

ROLE_LABEL = "llm-d.ai/role"
CONTEXT_LENGTH_RANGE_LABEL = "llm-d.ai/context-length-range"


class Role(enum.IntFlag):
    """GPU role bits (reference label values, filter/bylabel/roles.go:9-48)."""
    NONE = 0
    DECODE = 1
    PREFILL = 2
    ENCODE = 4


_ROLE_VALUES = {
    "decode": Role.DECODE,
    "prefill": Role.PREFILL,
    "encode": Role.ENCODE,
    "prefill-decode": Role.PREFILL | Role.DECODE,
    "encode-prefill": Role.ENCODE | Role.PREFILL,
    "encode-prefill-decode": Role.ENCODE | Role.PREFILL | Role.DECODE,
    "both": Role.PREFILL | Role.DECODE,  # deprecated alias (roles.go)
}


def role_mask(label_value: str) -> Role:
    return _ROLE_VALUES.get(label_value, Role.NONE)


@dataclass
class Metrics:
    """Worker metrics snapshot (parity: datalayer extractor Metrics struct;
    vLLM-compatible metric names in extractor/metrics/mapping.go)."""
    waiting_queue_size: int = 0
    running_requests_size: int = 0
    kv_cache_usage: float = 0.0          # [0,1]
    active_models: Dict[str, int] = field(default_factory=dict)   # LoRA -> count
    waiting_models: Dict[str, int] = field(default_factory=dict)
    max_active_models: int = 0
    cache_block_size: int = 16
    cache_num_blocks: int = 0
    update_time: float = 0.0

    def clone(self) -> "Metrics":
        return Metrics(self.waiting_queue_size, self.running_requests_size,
                       self.kv_cache_usage, dict(self.active_models),
                       dict(self.waiting_models), self.max_active_models,
                       self.cache_block_size, self.cache_num_blocks,
                       self.update_time)


@dataclass
class EndpointMetadata:
    name: str                   # e.g. "gpu0"
    index: int                  # dense endpoint index (C++ core bitmask bit)
    address: str                # "rank:port"-style locator; rank for in-node
    rank: int = 0               # torch.distributed rank of the worker
    labels: Dict[str, str] = field(default_factory=dict)
    # role-mask cache keyed on the label VALUE (labels are mutable; the
    # string parse in role_mask was 4 us x ~10 calls per routed request —
    # profiles/router_tax.json)
    _role_lbl: str = field(default="", repr=False, compare=False)
    _role_mask: int = field(default=0, repr=False, compare=False)

    def _mask(self) -> int:
        lbl = self.labels.get(ROLE_LABEL, "decode")
        if lbl != self._role_lbl:
            self._role_mask = int(role_mask(lbl))
            self._role_lbl = lbl
        return self._role_mask

    @property
    def roles(self) -> Role:
        return Role(self._mask())


class Endpoint:
    def __init__(self, metadata: EndpointMetadata):
        self.metadata = metadata
        self._metrics = Metrics()
        self._attrs: Dict[str, Any] = {}
        self._attr_lock = threading.Lock()

    # -- metrics (atomic snapshot swap) --
    @property
    def metrics(self) -> Metrics:
        return self._metrics

    def update_metrics(self, m: Metrics) -> None:
        m.update_time = time.monotonic()
        self._metrics = m  # single ref assignment = atomic swap

    # -- attribute map --
    def put_attribute(self, key: str, value: Any) -> None:
        with self._attr_lock:
            self._attrs[key] = value

    def get_attribute(self, key: str, default: Any = None) -> Any:
        with self._attr_lock:
            return self._attrs.get(key, default)

    def attribute_keys(self) -> List[str]:
        with self._attr_lock:
            return list(self._attrs.keys())

    # -- helpers --
    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def index(self) -> int:
        return self.metadata.index

    @property
    def roles(self) -> Role:
        return self.metadata.roles

    def has_role(self, role: Role) -> bool:
        # plain int AND — IntFlag.__and__ re-constructs an enum member per
        # call and showed up at 40 us/request in the router profile
        return bool(self.metadata._mask() & int(role))

    def __repr__(self) -> str:
        return f"Endpoint({self.metadata.name}, roles={self.metadata.roles!r})"
