"""Remote-node endpoints: route requests to ANOTHER node's front door.

Reference analog: a multi-pool deployment where the router steers traffic
to model servers outside its own pool (apix InferencePoolImport /
cross-pool endpoints). Here the remote "model server" is a peer node's
front door, which already speaks both halves of the contract:
  * metrics: its /metrics renders the vLLM-compatible families, scraped
    by `datalayer/extractor.HttpMetricsSource` (options.go:121-125);
  * execution: the internal enqueue API (`/internal/v1/enqueue`,
    server/openai_app.py) accepts a tokenized request and returns the
    completion.

A remote endpoint is a normal datastore Endpoint whose
`llm-d.ai/remote-url` label carries the peer's base URL — every filter,
scorer, and picker treats it exactly like a local GPU-role endpoint; only
dispatch differs (`NodeRunner._emit_assignment` hands the request to the
RemoteForwarder instead of the engine mailbox). Forwarded requests are
non-streaming end-to-end; tokens surface in one completion.
"""
import queue
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Callable, Dict, List, Optional

from ..datalayer.datastore import make_endpoint
from ..datalayer.endpoint import Endpoint
from ..handlers.parsers import Usage
from ..scheduling.types import LLMRequest
from ..utils.logging import get_logger

log = get_logger("node.remote")

REMOTE_URL_LABEL = "llm-d.ai/remote-url"
ENQUEUE_PATH = "/internal/v1/enqueue"
CANCEL_PATH = "/internal/v1/cancel"

# transport(url, payload_dict) -> response_dict; injectable for tests
Transport = Callable[[str, Dict], Dict]
# stream_transport(url, payload_dict) -> iterator of event dicts
# ({"tokens": [...]} chunks, then a final {"done": True, ...} record)
StreamTransport = Callable[[str, Dict], object]


def remote_endpoint(name: str, index: int, url: str,
                    role: str = "decode") -> Endpoint:
    """Datastore endpoint for a peer node's front door at `url`."""
    base = url.rstrip("/")
    return make_endpoint(name, index, role=role, address=base,
                         labels={REMOTE_URL_LABEL: base,
                                 "metrics_url": base + "/metrics"})


def _default_transport(url: str, payload: Dict) -> Dict:
    import httpx
    r = httpx.post(url, json=payload, timeout=300.0)
    r.raise_for_status()
    return r.json()


def _default_stream_transport(url: str, payload: Dict):
    """SSE relay: iterate `data:` events from the peer's streaming
    enqueue route (decode.go SSE re-emission, here across nodes)."""
    import json as json_mod

    import httpx
    with httpx.stream("POST", url, json=payload, timeout=300.0) as r:
        r.raise_for_status()
        for line in r.iter_lines():
            line = line.strip()
            if not line.startswith("data:"):
                continue
            body = line[5:].strip()
            if body == "[DONE]":
                break
            yield json_mod.loads(body)


class RemoteForwarder:
    """Forwards scheduled requests to remote-node endpoints on a small
    thread pool; completed responses drain into the router step as
    Completion objects (mirrors the sidecar's role of carrying the
    request to the remote engine and relaying the response)."""

    def __init__(self, transport: Optional[Transport] = None,
                 max_workers: int = 8,
                 stream_transport: Optional[StreamTransport] = None):
        self.transport = transport or _default_transport
        self.stream_transport = stream_transport or _default_stream_transport
        self.max_workers = max_workers
        self._pool: Optional[ThreadPoolExecutor] = None
        self._done: "queue.Queue" = queue.Queue()
        self._tokens: "queue.Queue" = queue.Queue()  # (req_id, [tokens])
        self._lock = threading.Lock()
        self.inflight = 0

    def _ensure_pool(self) -> ThreadPoolExecutor:
        if self._pool is None:
            self._pool = ThreadPoolExecutor(max_workers=self.max_workers,
                                            thread_name_prefix="remote-fwd")
        return self._pool

    def forward(self, req: LLMRequest, base_url: str) -> None:
        payload = {
            "request_id": req.request_id,
            "model": req.target_model or req.model,
            "prompt": req.prompt,
            "prompt_tokens": req.prompt_tokens,
            "max_tokens": req.max_tokens,
            "temperature": req.temperature,
            "stop_token_ids": req.stop_token_ids,
            "priority": req.priority,
            "stream": bool(req.streaming),
        }
        with self._lock:
            self.inflight += 1
        runner = self._run_stream if req.streaming else self._run
        self._ensure_pool().submit(runner, req.request_id,
                                   base_url + ENQUEUE_PATH, payload)

    def _run(self, request_id: str, url: str, payload: Dict) -> None:
        from .runner import Completion
        try:
            resp = self.transport(url, payload)
            u = resp.get("usage", {})
            comp = Completion(
                request_id=request_id,
                usage=Usage(prompt_tokens=u.get("prompt_tokens", 0),
                            completion_tokens=u.get("completion_tokens", 0),
                            cached_tokens=u.get("cached_tokens", 0),
                            ttft_ms=u.get("ttft_ms"),
                            e2e_ms=u.get("e2e_ms")),
                tokens=resp.get("tokens", []),
                finish_reason=resp.get("finish_reason", "length"),
                error=resp.get("error", ""))
        except Exception as e:
            log.error("remote forward failed", url=url, err=str(e))
            comp = Completion(request_id=request_id, usage=Usage(),
                              error=f"remote_error: {e}")
        with self._lock:
            self.inflight -= 1
        self._done.put(comp)

    def _run_stream(self, request_id: str, url: str, payload: Dict) -> None:
        """Cross-node SSE token relay: per-token chunks surface as router
        token events while the generation runs on the peer; the final
        event closes out as a normal Completion."""
        from .runner import Completion
        final = None
        try:
            for ev in self.stream_transport(url, payload):
                if ev.get("done"):
                    final = ev
                    break
                toks = ev.get("tokens") or []
                if toks:
                    self._tokens.put((request_id, list(toks)))
            if final is None:
                raise RuntimeError("stream ended without a final event")
            u = final.get("usage", {})
            comp = Completion(
                request_id=request_id,
                usage=Usage(prompt_tokens=u.get("prompt_tokens", 0),
                            completion_tokens=u.get("completion_tokens", 0),
                            cached_tokens=u.get("cached_tokens", 0),
                            ttft_ms=u.get("ttft_ms"),
                            e2e_ms=u.get("e2e_ms")),
                tokens=final.get("tokens", []),
                finish_reason=final.get("finish_reason", "length"),
                error=final.get("error", ""))
        except Exception as e:
            log.error("remote stream failed", url=url, err=str(e))
            comp = Completion(request_id=request_id, usage=Usage(),
                              error=f"remote_error: {e}")
        with self._lock:
            self.inflight -= 1
        self._done.put(comp)

    def drain_tokens(self) -> List:
        out = []
        while True:
            try:
                out.append(self._tokens.get_nowait())
            except queue.Empty:
                return out

    def cancel(self, request_id: str, base_url: str) -> None:
        """Best-effort remote abort: tells the peer to unwind the
        forwarded request (client died on this side). The in-flight
        forward still resolves — its completion is discarded by the
        router since the decision entry is gone."""
        def _go():
            try:
                self.transport(base_url + CANCEL_PATH,
                               {"request_id": request_id})
            except Exception as e:
                log.v(4).info("remote cancel failed", err=str(e))
        self._ensure_pool().submit(_go)

    def drain(self) -> List:
        out = []
        while True:
            try:
                out.append(self._done.get_nowait())
            except queue.Empty:
                return out

    def shutdown(self) -> None:
        if self._pool is not None:
            self._pool.shutdown(wait=False)
