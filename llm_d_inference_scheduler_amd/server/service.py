"""NodeService — bridges the lockstep NodeRunner loop to concurrent callers
(the FastAPI front door / tests).

Runs the node step loop on a daemon thread; `submit()` returns a handle
whose `wait()` blocks until completion and whose `token_queue` yields
streamed tokens. This is the single-node stand-in for the reference's
Envoy->EPP->pod round trip: the ext-proc decision and the worker execution
happen in-process.
"""
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..node.runner import Completion, NodeRunner
from ..scheduling.types import LLMRequest


@dataclass
class RequestHandle:
    request_id: str
    token_queue: "queue.Queue" = field(default_factory=queue.Queue)
    _done = None

    def __post_init__(self):
        self._done = threading.Event()
        self.completion: Optional[Completion] = None

    def wait(self, timeout: Optional[float] = None) -> Optional[Completion]:
        self._done.wait(timeout)
        return self.completion

    def finish(self, completion: Completion) -> None:
        self.completion = completion
        self.token_queue.put(None)  # stream sentinel
        self._done.set()


class NodeService:
    def __init__(self, node: NodeRunner, step_interval_s: float = 0.0):
        assert node.is_router, "NodeService runs on the router rank"
        self.node = node
        self.step_interval_s = step_interval_s
        self._handles: Dict[str, RequestHandle] = {}
        self._cancels: "queue.Queue" = queue.Queue()
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def submit(self, req: LLMRequest) -> RequestHandle:
        handle = RequestHandle(request_id=req.request_id)
        with self._lock:
            self._handles[req.request_id] = handle
        self.node.submit(req)
        return handle

    def cancel(self, request_id: str) -> None:
        """Client went away (stream death): finish the handle immediately
        and unwind the request across the node on the loop thread."""
        with self._lock:
            h = self._handles.pop(request_id, None)
        if h is not None:
            from ..handlers.parsers import Usage
            h.finish(Completion(request_id=request_id, usage=Usage(),
                                error="canceled"))
        self._cancels.put(request_id)

    def step_once(self) -> None:
        while True:
            try:
                rid = self._cancels.get_nowait()
            except queue.Empty:
                break
            self.node.cancel(rid)
        self.node.step()
        for req_id, toks in self.node.drain_token_events():
            with self._lock:
                h = self._handles.get(req_id)
            if h is not None:
                for t in toks:
                    h.token_queue.put(t)
        for c in self.node.drain_completions():
            with self._lock:
                h = self._handles.pop(c.request_id, None)
            if h is not None:
                h.finish(c)

    def start(self) -> None:
        if self._thread is not None:
            return

        def loop():
            while not self._stop.is_set():
                try:
                    self.step_once()
                except Exception as e:  # engine failure: fail pending reqs
                    from ..handlers.parsers import Usage
                    with self._lock:
                        handles = list(self._handles.values())
                        self._handles.clear()
                    for h in handles:
                        h.finish(Completion(request_id=h.request_id,
                                            usage=Usage(),
                                            error=f"internal: {e}"))
                if self.step_interval_s:
                    time.sleep(self.step_interval_s)
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="node-step-loop")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
            self._thread = None
        self.node.shutdown()
