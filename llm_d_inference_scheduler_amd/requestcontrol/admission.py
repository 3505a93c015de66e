"""Admission controllers (parity: pkg/epp/requestcontrol/admission.go).

* LegacyAdmissionController (admission.go:109): shed sheddable requests
  (priority < 0) when the pool is saturated.
* FlowControlAdmissionController (admission.go:149): wrap the request into
  a FlowControlRequest and block in FlowController.enqueue_and_wait; the
  queue outcome translates to admit/deny (outcome translation :216).
"""
from typing import List, Optional, Tuple

from ..datalayer.endpoint import Endpoint
from ..flowcontrol.controller import FlowController
from ..flowcontrol.saturation import SaturationDetector
from ..flowcontrol.types import FlowControlRequest, QueueOutcome
from ..scheduling.types import SchedulingContext

# canonical x-request-dropped-reason values (pkg/common/error)
REASON_SATURATED = "saturated"
REASON_QUEUE_CAPACITY = "queue_capacity"
REASON_QUEUE_TIMEOUT = "queue_timeout"
REASON_EVICTED = "evicted"


class AdmissionDenied(Exception):
    def __init__(self, reason: str, detail: str = "", status: int = 429):
        super().__init__(detail or reason)
        self.reason = reason
        self.status = status


class AdmissionController:
    def admit(self, ctx: SchedulingContext,
              endpoints: List[Endpoint]) -> None:
        """Raises AdmissionDenied to shed the request."""
        raise NotImplementedError


class LegacyAdmissionController(AdmissionController):
    def __init__(self, detector: SaturationDetector):
        self.detector = detector

    def admit(self, ctx, endpoints) -> None:
        if ctx.request.priority < 0 and self.detector.is_saturated(endpoints):
            raise AdmissionDenied(REASON_SATURATED,
                                  "sheddable request shed: pool saturated")


_OUTCOME_TO_REASON = {
    QueueOutcome.REJECTED_CAPACITY: (REASON_QUEUE_CAPACITY, 429),
    QueueOutcome.REJECTED_OTHER: (REASON_EVICTED, 429),
    QueueOutcome.EVICTED_TTL: (REASON_QUEUE_TIMEOUT, 429),
    QueueOutcome.EVICTED_CONTEXT_CANCELLED: ("client_cancelled", 499),
    QueueOutcome.EVICTED_SATURATION: (REASON_SATURATED, 429),
    QueueOutcome.EVICTED_DISPLACED: (REASON_EVICTED, 429),
    QueueOutcome.EVICTED_SHUTDOWN: ("shutting_down", 503),
}


class FlowControlAdmissionController(AdmissionController):
    def __init__(self, controller: FlowController,
                 default_ttl_s: float = 30.0):
        self.controller = controller
        self.default_ttl_s = default_ttl_s

    def admit(self, ctx, endpoints) -> None:
        req = ctx.request
        deadline_ns = None
        if req.ttft_slo_ms is not None:
            deadline_ns = req.arrival_ns + int(req.ttft_slo_ms * 1e6)
        item = FlowControlRequest(request=req, priority=req.priority,
                                  ttl_s=self.default_ttl_s,
                                  deadline_ns=deadline_ns)
        ctx.state["flow_item"] = item
        outcome = self.controller.enqueue_and_wait(item)
        if outcome == QueueOutcome.DISPATCHED:
            return
        reason, status = _OUTCOME_TO_REASON.get(
            outcome, (REASON_EVICTED, 429))
        raise AdmissionDenied(reason, item.detail, status)
