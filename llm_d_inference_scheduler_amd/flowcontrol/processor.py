"""Shard processor (parity: pkg/epp/flowcontrol/controller/internal/processor.go).

The reference runs a single-goroutine actor per shard (enqueue -> capacity
check -> dispatch cycle walking priority bands high->low, fairness between
flows, ordering within a flow; cleanup sweep; evictAll on shutdown). Here
the actor is either a daemon thread (`start()`, server mode) or an external
driver calling `tick()` (bench/step mode); all state transitions happen
under the shard mutex with exactly-once finalization on the item itself.
"""
import threading
import time
from typing import Callable, Dict, List, Optional

from ..metrics import prom
from ..utils.logging import get_logger
from .registry import Band, FlowRegistry
from .types import FlowControlRequest, QueueOutcome

log = get_logger("flowcontrol.processor")

# dispatch_fn returns True if the item was handed off to the scheduler;
# False means "cannot dispatch right now" (saturated) and stops the cycle.
DispatchFn = Callable[[FlowControlRequest], bool]


class ShardProcessor:
    def __init__(self, shard_id: int, registry: FlowRegistry,
                 dispatch_fn: DispatchFn,
                 saturated_fn: Optional[Callable[[], bool]] = None):
        self.shard_id = shard_id
        self.registry = registry
        self.dispatch_fn = dispatch_fn
        self.saturated_fn = saturated_fn or (lambda: False)
        self.bands: Dict[int, Band] = registry.make_shard_bands()
        self._lock = threading.RLock()
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        # draining: no NEW work via JSQ; retires once fully drained
        # (registry shard.go draining lifecycle)
        self.draining = False

    @property
    def drained(self) -> bool:
        return self.draining and self.queued_len == 0

    # ---- stats used by JSQ-bytes shard selection (controller.go) ----
    @property
    def queued_bytes(self) -> int:
        with self._lock:
            return sum(b.total_bytes for b in self.bands.values())

    @property
    def queued_len(self) -> int:
        with self._lock:
            return sum(b.total_len for b in self.bands.values())

    # ---- enqueue (processor.go:218 capacity check + eviction) ----
    def submit(self, item: FlowControlRequest) -> None:
        with self._lock:
            bc = self.registry.band_for_priority(item.priority)
            band = self.bands[bc.priority]
            q = band.flow(item.flow_key)
            if bc.usage_limit is not None and bc.usage_limit.would_exceed(q, item):
                item.finalize(QueueOutcome.REJECTED_CAPACITY,
                              "per-flow usage limit")
                self.registry.record("rejected")
                prom.flow_dispatch_total.labels("rejected_capacity").inc()
                return
            over_band = band.over_capacity(item.byte_size)
            over_global = self._over_global(item.byte_size)
            if over_band or over_global:
                if not self._evict_for(item, band):
                    item.finalize(QueueOutcome.REJECTED_CAPACITY,
                                  "band capacity" if over_band
                                  else "shard capacity")
                    self.registry.record("rejected")
                    prom.flow_dispatch_total.labels("rejected_capacity").inc()
                    return
            q.push(item)
            self.registry.record("enqueued")
            self.registry.touch_flow(item.flow_key)
            prom.flow_queue_size.labels(str(bc.priority)).set(band.total_len)
        self._wake.set()

    def _over_global(self, extra_bytes: int) -> bool:
        # this shard's slice of the partitioned global capacity (shard.go)
        r = self.registry
        max_items, max_bytes = r.shard_max_items, r.shard_max_bytes
        if max_items is not None and \
                sum(b.total_len for b in self.bands.values()) + 1 > max_items:
            return True
        if max_bytes is not None and \
                sum(b.total_bytes for b in self.bands.values()) + extra_bytes > max_bytes:
            return True
        return False

    def _evict_for(self, item: FlowControlRequest, target_band: Band) -> bool:
        """Free capacity for a higher-priority arrival by evicting queued
        items from STRICTLY lower-priority bands, newest victims first
        (flowcontrol/eviction request_evictor semantics). A same-priority
        arrival never displaces — it is rejected instead."""
        freed = 0
        for prio in sorted(self.bands):  # lowest priority first
            band = self.bands[prio]
            if band.config.priority >= item.priority:
                break
            for flow_key in list(band.flows):
                q = band.flows[flow_key]
                while len(q) and freed < item.byte_size:
                    victim = q.peek_victim()
                    if victim is None:
                        break
                    q.remove(victim)
                    victim.finalize(QueueOutcome.EVICTED_DISPLACED,
                                    "displaced by higher-priority arrival")
                    self.registry.record("evicted")
                    prom.flow_dispatch_total.labels("evicted_displaced").inc()
                    freed += victim.byte_size
                if freed >= item.byte_size:
                    break
            if freed >= item.byte_size:
                break
        return freed >= item.byte_size or (
            not target_band.over_capacity(item.byte_size)
            and not self._over_global(item.byte_size))

    # ---- dispatch cycle (processor.go:322) ----
    def tick(self, max_dispatch: int = 1_000_000) -> int:
        """Run one dispatch cycle; returns number dispatched."""
        dispatched = 0
        with self._lock:
            self._sweep_expired()
            while dispatched < max_dispatch:
                if self.saturated_fn():
                    break
                sel = self._select_item()
                if sel is None:
                    break
                q, item = sel
                if not self.dispatch_fn(item):
                    break  # peeked only — item keeps its queue position
                q.remove(item)
                if item.finalize(QueueOutcome.DISPATCHED):
                    self.registry.record("dispatched")
                    prom.flow_dispatch_total.labels("dispatched").inc()
                    prom.flow_queue_duration.observe(
                        (time.monotonic_ns() - item.enqueue_ns) / 1e9)
                    dispatched += 1
            for band in self.bands.values():
                band.gc(self.registry.keep_flow)
        return dispatched

    def _select_item(self):
        """Peek the next dispatchable item (band high->low, fairness across
        flows, ordering within the flow). The item stays queued until the
        dispatch callback accepts it."""
        for bc in self.registry.band_configs:  # high -> low priority
            band = self.bands[bc.priority]
            while True:
                flow_key = band.fairness.select(band.flows)
                if flow_key is None:
                    break
                q = band.flows[flow_key]
                # drop already-finalized heads (evicted/cancelled while queued)
                while True:
                    item = q.peek()
                    if item is None or not item.finalized:
                        break
                    q.pop()
                if item is None:
                    if not len(q):
                        band.flows.pop(flow_key, None)
                    continue
                return q, item
        return None

    def _sweep_expired(self) -> None:
        now = time.monotonic_ns()
        for band in self.bands.values():
            for q in band.flows.values():
                for item in q.items():
                    if item.expired(now):
                        q.remove(item)
                        if item.finalize(QueueOutcome.EVICTED_TTL, "TTL"):
                            self.registry.record("evicted")
                            prom.flow_dispatch_total.labels("evicted_ttl").inc()

    def evict_all(self, outcome: QueueOutcome = QueueOutcome.EVICTED_SHUTDOWN):
        with self._lock:
            for band in self.bands.values():
                for q in band.flows.values():
                    while True:
                        item = q.pop()
                        if item is None:
                            break
                        item.finalize(outcome, "shard shutdown")
                        self.registry.record("evicted")
                band.flows.clear()

    # ---- actor mode ----
    def start(self, interval_s: float = 0.001) -> None:
        if self._thread is not None:
            return
        def run():
            while not self._stop.is_set():
                self._wake.wait(interval_s)
                self._wake.clear()
                self.tick()
        self._thread = threading.Thread(target=run, daemon=True,
                                        name=f"fc-shard-{self.shard_id}")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None
        self.evict_all()
