"""In-flight request eviction (parity: pkg/epp/flowcontrol/eviction/
{request_evictor,queue,evictor}.go + the plugin policies under
framework/plugins/flowcontrol/eviction/).

The reference tracks every DISPATCHED request in an eviction min-heap;
when saturation blocks dispatch, `EvictN` kills the most-evictable
in-flight requests (EvictCh -> ext-proc 429 with
x-request-dropped-reason). Here the tracked "in-flight" request is an
engine-resident sequence; the kill callback aborts it on every rank and
surfaces an error completion to the waiting client.

Policy plugins (same registered type names as the reference):
* ordering `priority-then-time-eviction-order-policy`
  (ordering/priority_time.go): lowest priority first, ties broken by
  NEWEST dispatch (least KV-cache investment lost). Expressed as a
  sort key rather than a Less() comparator — same total order.
* filter `sheddable-eviction-filter` (filtering/sheddable.go): only
  priority < 0 requests are evictable.
"""
import heapq
import itertools
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from ..plugins.interface import Plugin
from ..plugins.registry import register_plugin


@dataclass
class EvictionItem:
    """One dispatched request (interface/flowcontrol/eviction.go:29-46)."""
    request_id: str
    priority: int = 0
    dispatch_time: float = field(default_factory=time.time)
    target: str = ""


class EvictionOrderingPolicy(Plugin):
    def key(self, item: EvictionItem) -> Tuple:
        """Sort key; the smallest key is evicted first."""
        raise NotImplementedError


class EvictionFilterPolicy(Plugin):
    def accept(self, item: EvictionItem) -> bool:
        raise NotImplementedError


@register_plugin("priority-then-time-eviction-order-policy")
class PriorityThenTimeOrdering(EvictionOrderingPolicy):
    """Lowest priority evicted first; ties -> newest dispatch first
    (minimal wasted KV investment, ordering/priority_time.go:57-62)."""

    def key(self, item: EvictionItem) -> Tuple:
        return (item.priority, -item.dispatch_time)


@register_plugin("sheddable-eviction-filter")
class SheddableFilter(EvictionFilterPolicy):
    """Only sheddable (priority < 0) requests enter the eviction queue
    (requtil.IsSheddable convention)."""

    def accept(self, item: EvictionItem) -> bool:
        return item.priority < 0


class RequestEvictor:
    """Tracks in-flight requests; evicts the most-evictable on demand
    (request_evictor.go). Untrack is idempotent (the context-done safety
    net in the reference maps to our unconditional completion-path
    untrack)."""

    def __init__(self,
                 ordering: Optional[EvictionOrderingPolicy] = None,
                 filter_policy: Optional[EvictionFilterPolicy] = None):
        self.ordering = ordering or PriorityThenTimeOrdering("")
        self.filter = filter_policy or SheddableFilter("")
        self._heap: List[Tuple[Tuple, int, EvictionItem]] = []
        self._live: Dict[str, EvictionItem] = {}       # evictable only
        self._all: set = set()                         # every in-flight id
        self._tie = itertools.count()

    def track(self, item: EvictionItem) -> None:
        self._all.add(item.request_id)
        if not self.filter.accept(item):
            return
        self._live[item.request_id] = item
        heapq.heappush(self._heap,
                       (self.ordering.key(item), next(self._tie), item))

    def untrack(self, request_id: str) -> None:
        """Idempotent: safe from both the completion path and the
        eviction path."""
        self._all.discard(request_id)
        self._live.pop(request_id, None)    # heap entry becomes a tombstone

    def evict_n(self, n: int,
                evict_fn: Callable[[EvictionItem], None]) -> List[str]:
        """Pop up to n most-evictable live items and run evict_fn on each.
        Returns the evicted request ids."""
        out: List[str] = []
        while len(out) < n and self._heap:
            _, _, item = heapq.heappop(self._heap)
            if self._live.pop(item.request_id, None) is None:
                continue                    # tombstone
            self._all.discard(item.request_id)
            evict_fn(item)
            out.append(item.request_id)
        return out

    @property
    def stats(self) -> Tuple[int, int]:
        """(in_flight, evictable) — request_evictor.go:175."""
        return len(self._all), len(self._live)
