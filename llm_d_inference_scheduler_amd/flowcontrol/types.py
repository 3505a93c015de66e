"""Flow-control types (parity: pkg/epp/flowcontrol/types/{outcomes,errors}.go
and controller/internal/item.go FlowItem)."""
import enum
import itertools
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Optional

from ..scheduling.types import LLMRequest


class QueueOutcome(enum.Enum):
    DISPATCHED = "dispatched"
    REJECTED_CAPACITY = "rejected_capacity"
    REJECTED_OTHER = "rejected_other"
    EVICTED_TTL = "evicted_ttl"
    EVICTED_CONTEXT_CANCELLED = "evicted_context"
    EVICTED_SATURATION = "evicted_saturation"
    EVICTED_DISPLACED = "evicted_displaced"
    EVICTED_SHUTDOWN = "evicted_shutdown"


_id_counter = itertools.count(1)


@dataclass
class FlowControlRequest:
    """One queued request (types.FlowControlRequest). `flow_key` is the
    fairness id; `priority` selects the band (higher = more critical)."""
    request: LLMRequest
    flow_key: str = ""
    priority: int = 0
    byte_size: int = 0
    ttl_s: float = 30.0
    deadline_ns: Optional[int] = None  # EDF/SLO ordering key
    enqueue_ns: int = 0
    item_id: int = field(default_factory=lambda: next(_id_counter))

    # finalization handle — atomic once-only (item.go: "FlowItem uses atomic
    # operations"); here a lock + flag with an Event for blocking waiters.
    _final_lock: threading.Lock = field(default_factory=threading.Lock,
                                        repr=False)
    _done: threading.Event = field(default_factory=threading.Event, repr=False)
    outcome: Optional[QueueOutcome] = None
    detail: str = ""

    def __post_init__(self):
        if not self.enqueue_ns:
            self.enqueue_ns = time.monotonic_ns()
        if not self.flow_key:
            self.flow_key = self.request.fairness_id or "default"
        if not self.byte_size:
            self.byte_size = max(1, self.request.prompt_len_chars)

    def finalize(self, outcome: QueueOutcome, detail: str = "") -> bool:
        """Returns True only for the finalizing caller (exactly-once)."""
        with self._final_lock:
            if self.outcome is not None:
                return False
            self.outcome = outcome
            self.detail = detail
        self._done.set()
        return True

    @property
    def finalized(self) -> bool:
        return self.outcome is not None

    def wait(self, timeout: Optional[float] = None) -> Optional[QueueOutcome]:
        self._done.wait(timeout)
        return self.outcome

    def expired(self, now_ns: int) -> bool:
        return (now_ns - self.enqueue_ns) > self.ttl_s * 1e9
