"""Datalayer runtime + per-endpoint collectors
(parity: pkg/epp/datalayer/{runtime,collector,factory}.go).

Reference: a registry of Sources x Extractors with a per-endpoint Collector
goroutine ticking every 50 ms (runtime.go:52) and a 1 s collection timeout.
Here a DataSource is any object with `collect(endpoint) -> Metrics` — the
in-process source reads the worker engine's snapshot directly (or the
cross-rank mailbox state), the HTTP source (server mode) scrapes the
worker's /metrics endpoint with vLLM-compatible metric names.

Two drive modes:
  * threaded: one daemon thread per endpoint with the refresh ticker
    (the reference shape), started by `start()`;
  * stepped: `collect_all_now()` — used by the bench loop and tests, where
    the engine step IS the clock.
"""
import threading
import time
from typing import Callable, Dict, List, Optional

from .endpoint import Endpoint, Metrics
from ..utils.logging import get_logger

log = get_logger("datalayer.runtime")

DEFAULT_REFRESH_INTERVAL_S = 0.05  # 50 ms (runtime.go:52)


class DataSource:
    """Source interface (datalayer source plugins)."""

    def collect(self, endpoint: Endpoint) -> Optional[Metrics]:  # pragma: no cover
        raise NotImplementedError


class CallableSource(DataSource):
    """In-process source: pulls a Metrics snapshot from a callable keyed by
    endpoint name (the worker shim registers its snapshot fn)."""

    def __init__(self):
        self._fns: Dict[str, Callable[[], Metrics]] = {}

    def register(self, endpoint_name: str, fn: Callable[[], Metrics]) -> None:
        self._fns[endpoint_name] = fn

    def collect(self, endpoint: Endpoint) -> Optional[Metrics]:
        fn = self._fns.get(endpoint.name)
        return fn() if fn else None


class Collector:
    """Per-endpoint collection loop (collector.go:87)."""

    def __init__(self, endpoint: Endpoint, sources: List[DataSource],
                 interval_s: float = DEFAULT_REFRESH_INTERVAL_S):
        self.endpoint = endpoint
        self.sources = sources
        self.interval_s = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def collect_once(self) -> None:
        from ..metrics import prom
        for src in self.sources:
            try:
                m = src.collect(self.endpoint)
            except Exception as e:  # stale metrics treated as saturated later
                log.v(4).info("collect failed", endpoint=self.endpoint.name, err=str(e))
                prom.datalayer_poll_errors.labels(
                    type(src).__name__).inc()
                continue
            if m is not None:
                self.endpoint.update_metrics(m)

    def start(self) -> None:
        if self._thread is not None:
            return
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name=f"collector-{self.endpoint.name}")
        self._thread.start()

    def _run(self) -> None:
        while not self._stop.is_set():
            self.collect_once()
            self._stop.wait(self.interval_s)

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None


class DataLayerRuntime:
    """Registry of sources + collector lifecycle (runtime.go:36)."""

    def __init__(self, interval_s: float = DEFAULT_REFRESH_INTERVAL_S):
        self.sources: List[DataSource] = []
        self.interval_s = interval_s
        self._collectors: Dict[str, Collector] = {}
        self._started = False

    def add_source(self, src: DataSource) -> None:
        self.sources.append(src)

    def track(self, endpoint: Endpoint) -> None:
        if endpoint.name in self._collectors:
            return
        c = Collector(endpoint, self.sources, self.interval_s)
        self._collectors[endpoint.name] = c
        if self._started:
            c.start()

    def untrack(self, endpoint_name: str) -> None:
        c = self._collectors.pop(endpoint_name, None)
        if c:
            c.stop()

    def start(self) -> None:
        self._started = True
        for c in self._collectors.values():
            c.start()

    def stop(self) -> None:
        self._started = False
        for c in self._collectors.values():
            c.stop()

    def collect_all_now(self) -> None:
        for c in self._collectors.values():
            c.collect_once()

    def bind_datastore(self, datastore) -> None:
        """Track endpoints as the datastore adds/removes them (k8s_bind.go)."""
        def on_event(kind: str, ep: Endpoint):
            if kind == "add":
                self.track(ep)
            else:
                self.untrack(ep.name)
        datastore.on_endpoint_event(on_event)
        for ep in datastore.endpoints():
            self.track(ep)
