"""Exact KV-block index fed by engine KV events.

Reference parity: the `precise-prefix-cache-scorer` keeps a
`kvblock.Index` fresh via ZMQ subscriptions to per-pod vLLM KV events
(scorer/preciseprefixcache/precise_prefix_cache.go:197-258, 622-691) plus
2s-TTL *speculative* entries added at PreRequest to cover the
routing->KV-event blind spot (:42, 533-604).

MI355X-native re-grounding: the "pods" are this node's engines, whose
BlockManagers emit real stored/evicted content hashes (engine/kvcache.py);
events ride the per-step control-plane mailbox instead of ZMQ, and the
index lives in-process on the router rank.
"""
import time
from typing import Dict, Iterable, List, Sequence, Set, Tuple

SPECULATIVE_TTL_S = 2.0


class KVBlockIndex:
    def __init__(self, speculative_ttl_s: float = SPECULATIVE_TTL_S):
        self._by_hash: Dict[int, Set[str]] = {}
        # (hash, endpoint) -> expiry wall time
        self._spec: Dict[Tuple[int, str], float] = {}
        self.ttl = speculative_ttl_s
        self.events_applied = 0

    def apply_events(self, endpoint: str, stored: Iterable[int],
                     evicted: Iterable[int]) -> None:
        """Apply one engine's block store/evict event batch."""
        for h in stored:
            self._by_hash.setdefault(int(h), set()).add(endpoint)
            self._spec.pop((int(h), endpoint), None)
            self.events_applied += 1
        for h in evicted:
            eps = self._by_hash.get(int(h))
            if eps is not None:
                eps.discard(endpoint)
                if not eps:
                    del self._by_hash[int(h)]
            self.events_applied += 1

    def add_speculative(self, endpoint: str, hashes: Sequence[int],
                        now: float = None) -> None:
        """Routing-time entries with TTL: the picked engine will hold these
        blocks a beat later; cover the blind spot without trusting it."""
        exp = (now if now is not None else time.time()) + self.ttl
        for h in hashes:
            self._spec[(int(h), endpoint)] = exp

    def remove_endpoint(self, endpoint: str) -> None:
        for h in list(self._by_hash):
            self._by_hash[h].discard(endpoint)
            if not self._by_hash[h]:
                del self._by_hash[h]
        for key in [k for k in self._spec if k[1] == endpoint]:
            del self._spec[key]

    def _holds(self, h: int, endpoint: str, now: float) -> bool:
        if endpoint in self._by_hash.get(h, ()):  # confirmed
            return True
        exp = self._spec.get((h, endpoint))
        return exp is not None and exp > now

    def match_longest(self, hashes: Sequence[int],
                      endpoints: List[str]) -> Dict[str, int]:
        """Per endpoint: number of leading blocks it holds (confirmed or
        live-speculative)."""
        now = time.time()
        out: Dict[str, int] = {}
        for ep in endpoints:
            n = 0
            for h in hashes:
                if not self._holds(int(h), ep, now):
                    break
                n += 1
            out[ep] = n
        return out

    def sweep(self, now: float = None) -> None:
        """Drop expired speculative entries (reference :228-243)."""
        t = now if now is not None else time.time()
        for key in [k for k, exp in self._spec.items() if exp <= t]:
            del self._spec[key]

    @property
    def size(self) -> int:
        return len(self._by_hash)
