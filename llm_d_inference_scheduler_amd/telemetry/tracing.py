"""Lightweight tracing (parity: pkg/telemetry/tracing.go + pkg/common/observability/tracing).

The reference initializes an OTel OTLP exporter and wraps the stream
(`gateway.request`, server.go:179), director orchestration, profile-handler
decisions and sidecar stages (incl. true_ttft_ms) in spans. There is no
network egress here, so spans are recorded in-process (ring buffer +
optional JSONL export) with the same span/attribute names; an OTLP exporter
can be attached later without touching call sites.
"""
import json
import threading
import time
from collections import deque
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class Span:
    name: str
    start_ns: int
    end_ns: int = 0
    attributes: Dict[str, Any] = field(default_factory=dict)
    parent: Optional[str] = None        # parent span NAME (in-process view)
    trace_id: str = ""
    span_id: str = ""
    parent_span_id: str = ""

    def set_attribute(self, key: str, value: Any) -> None:
        self.attributes[key] = value

    @property
    def duration_ms(self) -> float:
        return (self.end_ns - self.start_ns) / 1e6


class Tracer:
    """Env config mirrors the reference's OTEL_* driven init
    (pkg/telemetry/tracing.go): LLMD_TRACING=0 disables span recording;
    LLMD_TRACE_EXPORT=<path> appends finished spans as JSONL at shutdown
    (the OTLP-exporter stand-in — there is no network egress here)."""

    def __init__(self, service_name: str, capacity: int = 4096):
        import atexit
        import os
        self.service_name = service_name
        self._spans: deque = deque(maxlen=capacity)
        self._lock = threading.Lock()
        self._local = threading.local()
        self.otlp = None
        self.enabled = os.environ.get("LLMD_TRACING", "1") != "0"
        export = os.environ.get("LLMD_TRACE_EXPORT", "")
        if export and self.enabled:
            atexit.register(self.export_jsonl, export)
        # OTLP/HTTP+JSON export (the reference's OTLP exporter,
        # pkg/telemetry/tracing.go, env-driven): standard OTEL endpoint
        # variable; spans batch on a daemon thread to <endpoint>/v1/traces
        self.otlp = OtlpHttpExporter.from_env(service_name)
        if self.otlp is not None and self.enabled:
            atexit.register(self.otlp.flush)

    @contextmanager
    def span(self, name: str, **attrs):
        if not self.enabled:
            yield _NOOP_SPAN
            return
        parent = getattr(self._local, "current", None)
        import os
        s = Span(name=name, start_ns=time.monotonic_ns(),
                 attributes=dict(attrs),
                 parent=parent.name if parent else None,
                 trace_id=parent.trace_id if parent else
                 os.urandom(16).hex(),
                 span_id=os.urandom(8).hex(),
                 parent_span_id=parent.span_id if parent else "")
        prev = parent
        self._local.current = s
        try:
            yield s
        finally:
            s.end_ns = time.monotonic_ns()
            self._local.current = prev
            with self._lock:
                self._spans.append(s)
            if self.otlp is not None:
                self.otlp.enqueue(s)

    def finished_spans(self, name: Optional[str] = None) -> List[Span]:
        with self._lock:
            spans = list(self._spans)
        if name is not None:
            spans = [s for s in spans if s.name == name]
        return spans

    def export_jsonl(self, path: str) -> None:
        with self._lock:
            spans = list(self._spans)
        with open(path, "w") as f:
            for s in spans:
                f.write(json.dumps({
                    "name": s.name, "start_ns": s.start_ns, "end_ns": s.end_ns,
                    "duration_ms": s.duration_ms, "parent": s.parent,
                    "trace_id": s.trace_id,
                    "attributes": s.attributes}) + "\n")


class OtlpHttpExporter:
    """Minimal OTLP/HTTP+JSON trace exporter (the protocol's official JSON
    encoding of ExportTraceServiceRequest). Batches spans and POSTs to
    `<OTEL_EXPORTER_OTLP_ENDPOINT>/v1/traces` on a daemon thread; drops on
    transport failure (tracing must never block serving)."""

    def __init__(self, endpoint: str, service_name: str,
                 batch: int = 256, interval_s: float = 2.0, client=None):
        self.endpoint = endpoint.rstrip("/")
        self.service_name = service_name
        self.batch = batch
        self.interval_s = interval_s
        self._buf: List[Span] = []
        self._lock = threading.Lock()
        self._client = client
        self.sent = 0
        self.dropped = 0
        self._timer: Optional[threading.Timer] = None

    @classmethod
    def from_env(cls, service_name: str):
        import os
        ep = os.environ.get("OTEL_EXPORTER_OTLP_ENDPOINT", "")
        if not ep:
            return None
        return cls(ep, service_name)

    def enqueue(self, span: Span) -> None:
        with self._lock:
            self._buf.append(span)
            full = len(self._buf) >= self.batch
        if full:
            self.flush()
        elif self._timer is None:
            t = threading.Timer(self.interval_s, self.flush)
            t.daemon = True
            self._timer = t
            t.start()

    @staticmethod
    def _attr(k: str, v: Any) -> Dict[str, Any]:
        if isinstance(v, bool):
            val = {"boolValue": v}
        elif isinstance(v, int):
            val = {"intValue": str(v)}
        elif isinstance(v, float):
            val = {"doubleValue": v}
        else:
            val = {"stringValue": str(v)}
        return {"key": k, "value": val}

    def _payload(self, spans: List[Span]) -> Dict[str, Any]:
        return {"resourceSpans": [{
            "resource": {"attributes": [
                self._attr("service.name", self.service_name)]},
            "scopeSpans": [{
                "scope": {"name": "llm_d_inference_scheduler_amd"},
                "spans": [{
                    "traceId": s.trace_id or "0" * 32,
                    "spanId": s.span_id or "0" * 16,
                    **({"parentSpanId": s.parent_span_id}
                       if s.parent_span_id else {}),
                    "name": s.name,
                    "kind": 1,  # SPAN_KIND_INTERNAL
                    "startTimeUnixNano": str(s.start_ns),
                    "endTimeUnixNano": str(s.end_ns),
                    "attributes": [self._attr(k, v)
                                   for k, v in s.attributes.items()],
                } for s in spans],
            }],
        }]}

    def flush(self) -> None:
        with self._lock:
            spans, self._buf = self._buf, []
            if self._timer is not None:
                self._timer.cancel()
                self._timer = None
        if not spans:
            return
        try:
            client = self._client
            if client is None:
                import httpx
                client = self._client = httpx.Client(timeout=2.0)
            r = client.post(self.endpoint + "/v1/traces",
                            json=self._payload(spans))
            if r.status_code < 300:
                self.sent += len(spans)
            else:
                self.dropped += len(spans)
        except Exception:
            self.dropped += len(spans)


class _NoopSpan:
    def set_attribute(self, key, value):
        pass


_NOOP_SPAN = _NoopSpan()
_tracer: Optional[Tracer] = None


def init_tracing(service_name: str = "llm-d-inference-scheduler-amd",
                 enabled: bool = True) -> Tracer:
    global _tracer
    _tracer = Tracer(service_name)
    _tracer.enabled = enabled
    return _tracer


def get_tracer() -> Tracer:
    global _tracer
    if _tracer is None:
        _tracer = Tracer("llm-d-inference-scheduler-amd")
    return _tracer


def parse_traceparent(header: Optional[str]):
    """W3C traceparent `00-<trace_id>-<parent_span_id>-<flags>` (the
    context the reference extracts from Envoy headers via otel
    propagation, handlers/request.go). Returns (trace_id, parent_span_id)
    or None on any malformation."""
    if not header:
        return None
    parts = header.strip().split("-")
    if len(parts) != 4:
        return None
    version, trace_id, span_id, _flags = parts
    if len(trace_id) != 32 or len(span_id) != 16 or len(version) != 2:
        return None
    try:
        int(trace_id, 16), int(span_id, 16), int(version, 16)
    except ValueError:
        return None
    if trace_id == "0" * 32 or span_id == "0" * 16:
        return None
    return trace_id, span_id
