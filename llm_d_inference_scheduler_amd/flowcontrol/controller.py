"""Flow controller (parity: pkg/epp/flowcontrol/controller/controller.go).

Supervises per-shard processors, distributes arrivals via JSQ-bytes
(join-shortest-queue by queued bytes, controller.go:94-150) and offers the
blocking `enqueue_and_wait` contract (controller.go:203) used by the
flow-control admission controller. `tick()` drives all shards one cycle in
step mode (bench / tests); `start()` runs them as actor threads.
"""
from typing import Callable, List, Optional

from .processor import DispatchFn, ShardProcessor
from .registry import FlowRegistry
from .types import FlowControlRequest, QueueOutcome


class FlowController:
    def __init__(self, registry: FlowRegistry, dispatch_fn: DispatchFn,
                 saturated_fn: Optional[Callable[[], bool]] = None):
        self.registry = registry
        self.shards: List[ShardProcessor] = [
            ShardProcessor(i, registry, dispatch_fn, saturated_fn)
            for i in range(registry.num_shards)
        ]

    def _pick_shard(self) -> ShardProcessor:
        return min(self.shards, key=lambda s: s.queued_bytes)  # JSQ-bytes

    def submit(self, item: FlowControlRequest) -> None:
        self._pick_shard().submit(item)

    def enqueue_and_wait(self, item: FlowControlRequest,
                         timeout: Optional[float] = None) -> QueueOutcome:
        """Blocks the calling thread until the item is finalized
        (dispatched / rejected / evicted), exactly like the reference
        blocks the request goroutine."""
        self.submit(item)
        outcome = item.wait(timeout if timeout is not None else item.ttl_s + 1)
        if outcome is None:
            item.finalize(QueueOutcome.EVICTED_TTL, "wait timeout")
            outcome = item.outcome
        return outcome

    def tick(self) -> int:
        return sum(s.tick() for s in self.shards)

    def start(self) -> None:
        for s in self.shards:
            s.start()

    def stop(self) -> None:
        for s in self.shards:
            s.stop()

    @property
    def queued_len(self) -> int:
        return sum(s.queued_len for s in self.shards)

    @property
    def queued_bytes(self) -> int:
        return sum(s.queued_bytes for s in self.shards)
