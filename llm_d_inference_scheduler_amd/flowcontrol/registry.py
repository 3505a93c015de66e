"""Flow registry — the flow-control control plane
(parity: pkg/epp/flowcontrol/registry/{registry,shard,managedqueue,leasing,
connection}.go).

Owns band configuration, per-shard topology, the flow-lifecycle table and
stats. Bands are walked high-priority-first by the shard processors; flows
(fairness ids) are created on first use, held alive by connection leases
(one per in-queue request, registry.go leasing), and garbage-collected only
once leaseless AND idle past `flow_idle_ttl_s` (registry.go:239 GC) — an
empty-but-leased flow keeps its fairness position instead of being reset by
every momentary drain. Global capacity is partitioned across shards
(ceil-divide) so the configured limits hold regardless of shard count."""
import math
import threading
import time
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

    def gc(self, keep=None) -> None:
        """Drop drained flow queues — except those `keep` says to retain
        (active leases / recently active), which hold their fairness
        position across momentary drains."""
        self.flows = {k: q for k, q in self.flows.items()
                      if len(q) or (keep is not None and keep(k))}

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


@dataclass
class FlowState:
    """Lifecycle record of one flow (fairness id) — registry-level, shared
    across shards (registry.go flow instances + leasing.go)."""
    key: str
    leases: int = 0
    total_enqueued: int = 0
    last_active: float = field(default_factory=time.monotonic)


class FlowRegistry:
    def __init__(self, bands: Optional[List[BandConfig]] = None,
                 num_shards: int = 1,
                 global_max_bytes: Optional[int] = None,
                 global_max_items: Optional[int] = None,
                 flow_idle_ttl_s: float = 30.0):
        if not bands:
            bands = [BandConfig(priority=0)]
        # walked high -> low (processor.go:322 dispatchCycle)
        self.band_configs = sorted(bands, key=lambda b: -b.priority)
        self.num_shards = max(1, num_shards)
        self.global_max_bytes = global_max_bytes
        self.global_max_items = global_max_items
        self.flow_idle_ttl_s = flow_idle_ttl_s
        self.stats = RegistryStats()
        self.flows: Dict[str, FlowState] = {}
        self._lock = threading.Lock()

    # ---- capacity partitioning (shard.go): ceil-divide the global caps
    # over shards so the configured totals hold for any shard count ----
    @property
    def shard_max_items(self) -> Optional[int]:
        if self.global_max_items is None:
            return None
        return math.ceil(self.global_max_items / self.num_shards)

    @property
    def shard_max_bytes(self) -> Optional[int]:
        if self.global_max_bytes is None:
            return None
        return math.ceil(self.global_max_bytes / self.num_shards)

    # ---- flow lifecycle: leases + idle GC (leasing.go, connection.go) ----
    def open_connection(self, key: str) -> FlowState:
        """A caller entering the queue path leases the flow: it cannot be
        GC'd (and its fairness position cannot reset) while leased."""
        with self._lock:
            st = self.flows.get(key)
            if st is None:
                st = self.flows[key] = FlowState(key)
            st.leases += 1
            st.last_active = time.monotonic()
            return st

    def close_connection(self, key: str) -> None:
        with self._lock:
            st = self.flows.get(key)
            if st is not None:
                st.leases = max(0, st.leases - 1)
                st.last_active = time.monotonic()

    def touch_flow(self, key: str) -> None:
        with self._lock:
            st = self.flows.get(key)
            if st is None:
                st = self.flows[key] = FlowState(key)
            st.total_enqueued += 1
            st.last_active = time.monotonic()

    def keep_flow(self, key: str) -> bool:
        """Shard GC predicate for empty flow queues."""
        st = self.flows.get(key)
        if st is None:
            return False
        return st.leases > 0 or \
            (time.monotonic() - st.last_active) < self.flow_idle_ttl_s

    def gc_flows(self) -> int:
        """Registry sweep: drop lifecycle records of leaseless flows idle
        past the TTL (registry.go:239). Returns flows collected."""
        now = time.monotonic()
        with self._lock:
            dead = [k for k, st in self.flows.items()
                    if st.leases == 0 and
                    (now - st.last_active) >= self.flow_idle_ttl_s]
            for k in dead:
                del self.flows[k]
            return len(dead)

    # ---- band/shard helpers ----
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
