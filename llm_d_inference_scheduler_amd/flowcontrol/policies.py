"""Intra-flow ordering + inter-flow fairness policies
(parity: pkg/epp/framework/plugins/flowcontrol/{ordering,fairness}/*
and the SafeQueue factory pkg/epp/flowcontrol/framework/plugins/queue).

Ordering policies supply the dispatch key (lower = first) and the queue
capability they need: FIFO keys ride the C++ ListQueue, keyed policies the
C++ MaxMinHeap — the capability-based factory of queue/factory.go:30-72.
"""
import itertools
from typing import Dict, List, Optional

from .. import _router_core as rc
from .types import FlowControlRequest


class OrderingPolicy:
    type_name = "ordering"
    needs_heap = False

    def key(self, item: FlowControlRequest) -> float:
        raise NotImplementedError


class FCFSOrdering(OrderingPolicy):
    """First-come-first-served (ordering/fcfs)."""
    type_name = "fcfs"
    needs_heap = False

    def key(self, item):
        return float(item.enqueue_ns)


class EDFOrdering(OrderingPolicy):
    """Earliest deadline first (ordering/edf)."""
    type_name = "edf"
    needs_heap = True

    def key(self, item):
        if item.deadline_ns is not None:
            return float(item.deadline_ns)
        return float(item.enqueue_ns + int(item.ttl_s * 1e9))


class SLODeadlineOrdering(OrderingPolicy):
    """Deadline from the request's TTFT SLO (ordering/slodeadline)."""
    type_name = "slodeadline"
    needs_heap = True

    def key(self, item):
        slo_ms = item.request.ttft_slo_ms
        if slo_ms is not None:
            return float(item.request.arrival_ns + slo_ms * 1e6)
        return EDFOrdering.key(self, item)


ORDERING_POLICIES = {c.type_name: c for c in
                     (FCFSOrdering, EDFOrdering, SLODeadlineOrdering)}


class FlowQueue:
    """One (band, flow) queue: C++ ListQueue or MaxMinHeap + item map
    (managedqueue.go len/bytes stats come free from the C++ side)."""

    def __init__(self, ordering: OrderingPolicy):
        self.ordering = ordering
        self._q = rc.MaxMinHeap() if ordering.needs_heap else rc.ListQueue()
        self._items: Dict[int, FlowControlRequest] = {}

    def push(self, item: FlowControlRequest) -> None:
        self._items[item.item_id] = item
        self._q.push(item.item_id, self.ordering.key(item), item.byte_size)

    def peek(self) -> Optional[FlowControlRequest]:
        item_id = self._q.peek()
        return self._items.get(item_id) if item_id is not None else None

    def pop(self) -> Optional[FlowControlRequest]:
        popped = self._q.pop()
        if popped is None:
            return None
        return self._items.pop(popped[0], None)

    def remove(self, item: FlowControlRequest) -> bool:
        if self._q.remove(item.item_id) is None:
            return False
        self._items.pop(item.item_id, None)
        return True

    def peek_victim(self) -> Optional[FlowControlRequest]:
        """Default eviction victim: newest (FIFO tail) / worst key (heap max)."""
        item_id = (self._q.peek_max() if isinstance(self._q, rc.MaxMinHeap)
                   else self._q.peek_tail())
        return self._items.get(item_id) if item_id is not None else None

    def __len__(self) -> int:
        return len(self._q)

    @property
    def bytes(self) -> int:
        return self._q.bytes

    def items(self) -> List[FlowControlRequest]:
        return list(self._items.values())


class FairnessPolicy:
    """Inter-flow selection within one priority band."""
    type_name = "fairness"

    def select(self, flows: Dict[str, FlowQueue]) -> Optional[str]:
        raise NotImplementedError


class RoundRobinFairness(FairnessPolicy):
    """Cycle across non-empty flows (fairness/roundrobin)."""
    type_name = "roundrobin"

    def __init__(self):
        self._cycle_pos = 0

    def select(self, flows):
        names = sorted(k for k, q in flows.items() if len(q))
        if not names:
            return None
        self._cycle_pos = (self._cycle_pos + 1) % len(names)
        return names[self._cycle_pos]


class GlobalStrictFairness(FairnessPolicy):
    """Strict global order: the flow whose head has the smallest ordering
    key wins (fairness/globalstrict)."""
    type_name = "globalstrict"

    def select(self, flows):
        best_name, best_key = None, None
        for name, q in flows.items():
            head = q.peek()
            if head is None:
                continue
            key = q.ordering.key(head)
            if best_key is None or key < best_key:
                best_name, best_key = name, key
        return best_name


FAIRNESS_POLICIES = {c.type_name: c for c in
                     (RoundRobinFairness, GlobalStrictFairness)}


class StaticUsageLimit:
    """Per-flow static caps (usagelimits/static)."""

    def __init__(self, max_items: Optional[int] = None,
                 max_bytes: Optional[int] = None):
        self.max_items = max_items
        self.max_bytes = max_bytes

    def would_exceed(self, q: FlowQueue, item: FlowControlRequest) -> bool:
        if self.max_items is not None and len(q) + 1 > self.max_items:
            return True
        if self.max_bytes is not None and q.bytes + item.byte_size > self.max_bytes:
            return True
        return False
