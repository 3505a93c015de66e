"""Datastore (parity: pkg/epp/datastore/datastore.go:62-97 + modelrewritestore.go).

Thread-safe store of the endpoint pool, InferenceObjectives and
InferenceModelRewrites. On the single-node build these are populated from
local config objects (`NodePoolConfig`) instead of Kubernetes reconcilers;
the watch/reconcile surface is a callback list (`on_endpoint_event`) that
plays the role of the reference's endpoint lifecycle notifications
(datalayer/source/notifications) driving e.g. prefix-table subscribers.
"""
import threading
from typing import Callable, Dict, List, Optional

from .endpoint import Endpoint, EndpointMetadata
from ..api.objectives import InferenceObjective
from ..api.modelrewrite import InferenceModelRewrite


class Datastore:
    def __init__(self):
        self._lock = threading.RLock()
        self._endpoints: Dict[str, Endpoint] = {}
        self._objectives: Dict[str, InferenceObjective] = {}
        self._rewrites: List[InferenceModelRewrite] = []
        self._pool_ready = False
        self._listeners: List[Callable[[str, Endpoint], None]] = []

    # -- pool / endpoints --
    def set_pool_ready(self, ready: bool = True) -> None:
        with self._lock:
            self._pool_ready = ready

    def pool_ready(self) -> bool:
        with self._lock:
            return self._pool_ready

    def add_endpoint(self, ep: Endpoint) -> None:
        with self._lock:
            self._endpoints[ep.name] = ep
            listeners = list(self._listeners)
        for fn in listeners:
            fn("add", ep)

    def remove_endpoint(self, name: str) -> Optional[Endpoint]:
        with self._lock:
            ep = self._endpoints.pop(name, None)
            listeners = list(self._listeners)
        if ep is not None:
            for fn in listeners:
                fn("remove", ep)
        return ep

    def endpoints(self) -> List[Endpoint]:
        with self._lock:
            return list(self._endpoints.values())

    def get_endpoint(self, name: str) -> Optional[Endpoint]:
        with self._lock:
            return self._endpoints.get(name)

    def on_endpoint_event(self, fn: Callable[[str, Endpoint], None]) -> None:
        with self._lock:
            self._listeners.append(fn)

    # -- objectives --
    def put_objective(self, obj: InferenceObjective) -> None:
        with self._lock:
            self._objectives[obj.name] = obj

    def get_objective(self, name: str) -> Optional[InferenceObjective]:
        with self._lock:
            return self._objectives.get(name)

    def delete_objective(self, name: str) -> None:
        with self._lock:
            self._objectives.pop(name, None)

    # -- model rewrites --
    def put_model_rewrite(self, rw: InferenceModelRewrite) -> None:
        with self._lock:
            self._rewrites = [r for r in self._rewrites if r.name != rw.name]
            self._rewrites.append(rw)
            # oldest-resource tie-break: stable order by creation seq
            self._rewrites.sort(key=lambda r: r.creation_seq)

    def model_rewrites(self) -> List[InferenceModelRewrite]:
        with self._lock:
            return list(self._rewrites)

    def delete_model_rewrite(self, name: str) -> None:
        with self._lock:
            self._rewrites = [r for r in self._rewrites if r.name != name]


def make_endpoint(name: str, index: int, rank: int = 0,
                  role: str = "decode", labels: Optional[Dict[str, str]] = None,
                  address: str = "") -> Endpoint:
    lab = dict(labels or {})
    lab.setdefault("llm-d.ai/role", role)
    return Endpoint(EndpointMetadata(name=name, index=index,
                                     address=address or f"rank:{rank}",
                                     rank=rank, labels=lab))
