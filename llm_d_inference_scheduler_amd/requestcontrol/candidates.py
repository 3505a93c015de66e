"""Endpoint candidate resolution (parity: pkg/epp/requestcontrol/candidates.go).

Resolves the Envoy subset-hint metadata
(`x-gateway-destination-endpoint-subset`, metadata/consts.go:24) against the
datastore, with a short-TTL cached wrapper (50 ms, runner.go:349) so the
flow-control saturation checks don't hammer the datastore."""
import time
from typing import List, Optional

from ..datalayer.datastore import Datastore
from ..datalayer.endpoint import Endpoint

SUBSET_HINT_HEADER = "x-gateway-destination-endpoint-subset"


class EndpointCandidates:
    def __init__(self, datastore: Datastore, cache_ttl_s: float = 0.05):
        self.datastore = datastore
        self.cache_ttl_s = cache_ttl_s
        self._cache: List[Endpoint] = []
        self._cache_time = 0.0

    def all(self) -> List[Endpoint]:
        now = time.monotonic()
        if now - self._cache_time > self.cache_ttl_s:
            self._cache = self.datastore.endpoints()
            self._cache_time = now
        return self._cache

    def locate(self, subset_hint: Optional[List[str]] = None) -> List[Endpoint]:
        eps = self.all()
        if not subset_hint:
            return list(eps)
        allowed = set(subset_hint)
        return [ep for ep in eps
                if ep.name in allowed or ep.metadata.address in allowed]
