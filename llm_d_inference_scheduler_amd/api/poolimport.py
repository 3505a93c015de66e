"""InferencePoolImport — declare a REMOTE pool of serving endpoints.

Parity: the reference's `InferencePoolImport` CRD surface
(/root/reference/config/crd/bases poolimports; apix) lets an EPP route to
pools imported from elsewhere. Re-grounded on this node: an import names a
set of peer front-door URLs; applying it materializes them as remote
endpoints (node/remote.py) in the datastore — scraped through
`HttpMetricsSource` like any vLLM-compatible worker, scored/filtered by
the same plugins, dispatched over the internal enqueue API (streaming or
not). Deleting an import removes its endpoints (and their collectors via
the datastore's endpoint-event hooks).
"""
from dataclasses import dataclass, field
from typing import Dict, List


@dataclass
class ImportedEndpoint:
    url: str                   # peer front-door base URL
    role: str = "decode"       # llm-d.ai/role label value


@dataclass
class InferencePoolImport:
    name: str
    endpoints: List[ImportedEndpoint] = field(default_factory=list)


class PoolImportManager:
    """Applies/deletes imports against a datastore; endpoint indices are
    allocated above the local ranks so C++ core bitmasks never collide."""

    def __init__(self, datastore, index_base: int = 64):
        self.datastore = datastore
        self.index_base = index_base
        self._applied: Dict[str, List[str]] = {}   # import -> endpoint names
        self._next_index = index_base

    def apply(self, imp: InferencePoolImport) -> List[str]:
        from ..node.remote import remote_endpoint
        self.delete(imp.name)
        names = []
        for i, spec in enumerate(imp.endpoints):
            name = f"import-{imp.name}-{i}"
            ep = remote_endpoint(name, self._next_index, spec.url,
                                 role=spec.role)
            self._next_index += 1
            self.datastore.add_endpoint(ep)
            names.append(name)
        self._applied[imp.name] = names
        return names

    def delete(self, name: str) -> None:
        for ep_name in self._applied.pop(name, []):
            self.datastore.remove_endpoint(ep_name)

    def imports(self) -> List[str]:
        return list(self._applied)
