"""Flow controller (parity: pkg/epp/flowcontrol/controller/controller.go).

Supervises per-shard processors, distributes arrivals via JSQ-bytes
(join-shortest-queue by queued bytes, controller.go:94-150) and offers the
blocking `enqueue_and_wait` contract (controller.go:203) used by the
flow-control admission controller. Each waiting caller holds a flow
CONNECTION LEASE in the registry for the duration of its queue residency
(registry leasing.go), so an active flow is never GC'd mid-request.
Shard topology is elastic: `set_shard_count` adds shards immediately and
removes them by DRAINING (no new JSQ assignments; retired once empty —
shard.go draining lifecycle). `tick()` drives all shards one cycle in
step mode (bench / tests); `start()` runs them as actor threads.
"""
import threading
from typing import Callable, List, Optional

from .processor import DispatchFn, ShardProcessor
from .registry import FlowRegistry
from .types import FlowControlRequest, QueueOutcome


class FlowController:
    def __init__(self, registry: FlowRegistry, dispatch_fn: DispatchFn,
                 saturated_fn: Optional[Callable[[], bool]] = None):
        self.registry = registry
        self.dispatch_fn = dispatch_fn
        self.saturated_fn = saturated_fn
        self._lock = threading.Lock()
        self._running = False
        self.shards: List[ShardProcessor] = [
            ShardProcessor(i, registry, dispatch_fn, saturated_fn)
            for i in range(registry.num_shards)
        ]
        self._next_shard_id = registry.num_shards

    # ---- shard topology ----
    def set_shard_count(self, n: int) -> None:
        """Elastic scale: grow immediately, shrink by draining."""
        n = max(1, n)
        with self._lock:
            active = [s for s in self.shards if not s.draining]
            while len(active) < n:
                s = ShardProcessor(self._next_shard_id, self.registry,
                                   self.dispatch_fn, self.saturated_fn)
                self._next_shard_id += 1
                if self._running:
                    s.start()
                self.shards.append(s)
                active.append(s)
            # drain the newest shards first
            for s in reversed(active[n:]):
                s.draining = True
            self.registry.num_shards = n

    def _reap(self) -> None:
        drained = [s for s in self.shards if s.drained]
        if not drained:
            return
        with self._lock:
            for s in drained:
                if s.drained:           # re-check under lock
                    s.stop()
                    self.shards.remove(s)

    def _pick_shard(self) -> ShardProcessor:
        candidates = [s for s in self.shards if not s.draining]
        return min(candidates, key=lambda s: s.queued_bytes)  # JSQ-bytes

    # ---- request path ----
    def submit(self, item: FlowControlRequest) -> None:
        self._reap()
        self._pick_shard().submit(item)

    def enqueue_and_wait(self, item: FlowControlRequest,
                         timeout: Optional[float] = None) -> QueueOutcome:
        """Blocks the calling thread until the item is finalized
        (dispatched / rejected / evicted), exactly like the reference
        blocks the request goroutine. Holds a flow lease for the wait."""
        self.registry.open_connection(item.flow_key)
        try:
            self.submit(item)
            outcome = item.wait(timeout if timeout is not None
                                else item.ttl_s + 1)
            if outcome is None:
                item.finalize(QueueOutcome.EVICTED_TTL, "wait timeout")
                outcome = item.outcome
            return outcome
        finally:
            self.registry.close_connection(item.flow_key)

    def tick(self) -> int:
        n = sum(s.tick() for s in self.shards)
        self._reap()
        return n

    def start(self) -> None:
        with self._lock:
            self._running = True
            for s in self.shards:
                s.start()

    def stop(self) -> None:
        with self._lock:
            self._running = False
            for s in self.shards:
                s.stop()

    @property
    def queued_len(self) -> int:
        return sum(s.queued_len for s in self.shards)

    @property
    def queued_bytes(self) -> int:
        return sum(s.queued_bytes for s in self.shards)
