"""Flow registry — the flow-control control plane
(parity: pkg/epp/flowcontrol/registry/{registry,shard,managedqueue}.go).

Owns band configuration, per-shard topology and stats. Bands are walked
high-priority-first by the shard processors; flows (fairness ids) are
created on first use and GC'd when drained (registry.go:239 GC)."""
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .policies import (FAIRNESS_POLICIES, ORDERING_POLICIES, FairnessPolicy,
                       FlowQueue, OrderingPolicy, StaticUsageLimit)


@dataclass
class BandConfig:
    priority: int
    fairness: str = "roundrobin"
    ordering: str = "fcfs"
    max_bytes: Optional[int] = None     # band capacity
    max_items: Optional[int] = None
    usage_limit: Optional[StaticUsageLimit] = None


@dataclass
class RegistryStats:
    enqueued: int = 0
    dispatched: int = 0
    rejected: int = 0
    evicted: int = 0


class Band:
    """One priority band inside a shard: flows -> FlowQueue."""

    def __init__(self, config: BandConfig):
        self.config = config
        self.ordering: OrderingPolicy = ORDERING_POLICIES[config.ordering]()
        self.fairness: FairnessPolicy = FAIRNESS_POLICIES[config.fairness]()
        self.flows: Dict[str, FlowQueue] = {}

    def flow(self, key: str) -> FlowQueue:
        q = self.flows.get(key)
        if q is None:
            q = FlowQueue(self.ordering)
            self.flows[key] = q
        return q

    def gc(self) -> None:
        self.flows = {k: q for k, q in self.flows.items() if len(q)}

    @property
    def total_len(self) -> int:
        return sum(len(q) for q in self.flows.values())

    @property
    def total_bytes(self) -> int:
        return sum(q.bytes for q in self.flows.values())

    def over_capacity(self, extra_bytes: int, extra_items: int = 1) -> bool:
        c = self.config
        if c.max_items is not None and self.total_len + extra_items > c.max_items:
            return True
        if c.max_bytes is not None and self.total_bytes + extra_bytes > c.max_bytes:
            return True
        return False


class FlowRegistry:
    def __init__(self, bands: Optional[List[BandConfig]] = None,
                 num_shards: int = 1,
                 global_max_bytes: Optional[int] = None,
                 global_max_items: Optional[int] = None):
        if not bands:
            bands = [BandConfig(priority=0)]
        # walked high -> low (processor.go:322 dispatchCycle)
        self.band_configs = sorted(bands, key=lambda b: -b.priority)
        self.num_shards = max(1, num_shards)
        self.global_max_bytes = global_max_bytes
        self.global_max_items = global_max_items
        self.stats = RegistryStats()
        self._lock = threading.Lock()

    def band_for_priority(self, priority: int) -> BandConfig:
        # the closest configured band at or below the request's priority,
        # else the lowest band
        for bc in self.band_configs:
            if priority >= bc.priority:
                return bc
        return self.band_configs[-1]

    def make_shard_bands(self) -> Dict[int, Band]:
        return {bc.priority: Band(bc) for bc in self.band_configs}

    def record(self, what: str, n: int = 1) -> None:
        with self._lock:
            setattr(self.stats, what, getattr(self.stats, what) + n)
