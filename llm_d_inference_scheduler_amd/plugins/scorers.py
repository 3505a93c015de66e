"""Scorer plugins (parity: pkg/epp/framework/plugins/scheduling/scorer/*).

All return endpoint-name -> [0,1]; the core weights and clamps. Scorers with
a `native_spec` run inside the C++ ProfileRunner hot loop (scoring.h) — the
python `score` method is the same formula and is used for parity tests and
for profiles mixing non-native plugins.
"""
import math
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

from ..datalayer.attributes import (IN_FLIGHT_LOAD, LATENCY_PREDICTION_INFO,
                                    PREFIX_CACHE_MATCH_INFO)
from ..datalayer.endpoint import CONTEXT_LENGTH_RANGE_LABEL, Endpoint
from ..scheduling.types import SchedulingContext
from .interface import Scorer
from .registry import register_plugin

# ScorerKind mirror of csrc/router/scoring.h
SC_QUEUE, SC_KV_UTIL, SC_PREFIX, SC_RUNNING = 0, 1, 2, 3
SC_LOAD_AWARE, SC_TOKEN_LOAD, SC_ACTIVE_REQUEST = 4, 5, 6


def _minmax_inverted(vals: Dict[str, float]) -> Dict[str, float]:
    if not vals:
        return {}
    lo, hi = min(vals.values()), max(vals.values())
    span = hi - lo
    if span <= 0:
        return {k: 1.0 for k in vals}
    return {k: 1.0 - (v - lo) / span for k, v in vals.items()}


@register_plugin("queue-scorer", aliases=["queue"])
class QueueScorer(Scorer):
    """Min-max normalized waiting-queue depth, inverted (scorer/queuedepth)."""

    def native_spec(self):
        return (SC_QUEUE, 0.0)

    def score(self, ctx, endpoints):
        return _minmax_inverted(
            {ep.name: float(ep.metrics.waiting_queue_size) for ep in endpoints})


@register_plugin("kv-cache-utilization-scorer", aliases=["kv-cache-scorer"])
class KVCacheUtilizationScorer(Scorer):
    """1 - kvCacheUsagePercent (scorer/kvcacheutilization)."""

    def native_spec(self):
        return (SC_KV_UTIL, 0.0)

    def score(self, ctx, endpoints):
        return {ep.name: 1.0 - ep.metrics.kv_cache_usage for ep in endpoints}


@register_plugin("prefix-cache-scorer", aliases=["prefix-scorer"])
class PrefixCacheScorer(Scorer):
    """MatchBlocks/TotalBlocks from PrefixCacheMatchInfo
    (scorer/prefix/plugin.go:96-124). Consumes the approx-prefix producer."""

    def native_spec(self):
        return (SC_PREFIX, 0.0)

    def score(self, ctx: SchedulingContext, endpoints):
        info = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        if info is None:
            return {ep.name: 0.0 for ep in endpoints}
        return {ep.name: info.ratio(ep.name) for ep in endpoints}


@register_plugin("running-requests-size-scorer", aliases=["running-requests-scorer"])
class RunningRequestsScorer(Scorer):
    def native_spec(self):
        return (SC_RUNNING, 0.0)

    def score(self, ctx, endpoints):
        return _minmax_inverted(
            {ep.name: float(ep.metrics.running_requests_size) for ep in endpoints})


@register_plugin("load-aware-scorer")
class LoadAwareScorer(Scorer):
    """0.5 if queue empty -> 0 at threshold (default 128) (scorer/loadaware)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.threshold = float(params.get("threshold", 128))

    def native_spec(self):
        return (SC_LOAD_AWARE, self.threshold)

    def score(self, ctx, endpoints):
        return {ep.name: max(0.0, 0.5 * (1.0 - ep.metrics.waiting_queue_size
                                         / self.threshold))
                for ep in endpoints}


@register_plugin("token-load-scorer")
class TokenLoadScorer(Scorer):
    """1 - inflightTokens/threshold from InFlightLoad (scorer/tokenload)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.threshold = float(params.get("threshold", 1_000_000))

    def native_spec(self):
        return (SC_TOKEN_LOAD, self.threshold)

    def score(self, ctx, endpoints):
        out = {}
        for ep in endpoints:
            load = ep.get_attribute(IN_FLIGHT_LOAD)
            toks = load.snapshot()[1] if load else 0
            out[ep.name] = max(0.0, 1.0 - toks / self.threshold)
        return out


@register_plugin("active-request-scorer")
class ActiveRequestScorer(Scorer):
    """Router-tracked in-flight counts (scorer/activerequest, active_request.go:
    139-168): idle endpoints (count <= idleThreshold) pin 1.0; busy endpoints
    scale (max-count)/max * maxBusyScore — the gap steers toward idle pods.
    requestTimeout is deprecated upstream and ignored here too (tracking is
    the inflight-load-producer's job)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.idle_threshold = max(0, int(params.get("idleThreshold", 0)))
        mbs = float(params.get("maxBusyScore", 1.0))
        self.max_busy_score = mbs if 0.0 < mbs <= 1.0 else 1.0

    def native_spec(self):
        return (SC_ACTIVE_REQUEST, float(self.idle_threshold),
                self.max_busy_score)

    def score(self, ctx, endpoints):
        vals = {}
        for ep in endpoints:
            load = ep.get_attribute(IN_FLIGHT_LOAD)
            vals[ep.name] = float(load.snapshot()[0]) if load else 0.0
        hi = max(vals.values(), default=0.0)
        out = {}
        for name, c in vals.items():
            if c <= self.idle_threshold:
                out[name] = 1.0
            else:
                out[name] = ((hi - c) / hi * self.max_busy_score
                             if hi > 0 else 1.0)
        return out


@register_plugin("lora-affinity-scorer")
class LoraAffinityScorer(Scorer):
    """1.0 active / 0.8 capacity / 0.6 waiting / 0.0 full (scorer/loraaffinity)."""

    def score(self, ctx: SchedulingContext, endpoints):
        model = ctx.request.target_model
        out = {}
        for ep in endpoints:
            m = ep.metrics
            if model in m.active_models:
                out[ep.name] = 1.0
            elif m.max_active_models and \
                    len(m.active_models) < m.max_active_models:
                out[ep.name] = 0.8
            elif model in m.waiting_models:
                out[ep.name] = 0.6
            else:
                out[ep.name] = 0.0
        return out


@register_plugin("session-affinity-scorer")
class SessionAffinityScorer(Scorer):
    """Sticky by session id (scorer/sessionaffinity): remembered endpoint
    scores 1.0, everything else 0."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._sessions: "OrderedDict[str, str]" = OrderedDict()
        self.capacity = int(params.get("capacity", 65536))

    def score(self, ctx: SchedulingContext, endpoints):
        sid = ctx.request.session_id or ctx.request.headers.get("session-id", "")
        if not sid:
            return {ep.name: 0.0 for ep in endpoints}
        target = self._sessions.get(sid)
        return {ep.name: (1.0 if ep.name == target else 0.0)
                for ep in endpoints}

    def remember(self, session_id: str, endpoint_name: str) -> None:
        if not session_id:
            return
        self._sessions[session_id] = endpoint_name
        self._sessions.move_to_end(session_id)
        while len(self._sessions) > self.capacity:
            self._sessions.popitem(last=False)

    def pre_request(self, ctx, result, target) -> None:
        if target is not None:
            sid = ctx.request.session_id or \
                ctx.request.headers.get("session-id", "")
            self.remember(sid, target.name)


@register_plugin("no-hit-lru-scorer")
class NoHitLRUScorer(Scorer):
    """Cold requests (no prefix hit) -> least-recently-routed endpoint;
    warm requests -> 0.5 neutral (scorer/nohitlru)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._last_routed: Dict[str, float] = {}
        self._clock = 0.0

    def score(self, ctx: SchedulingContext, endpoints):
        info = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        warm = info is not None and info.total_blocks > 0 and \
            any(v > 0 for v in info.match_blocks.values())
        if warm:
            return {ep.name: 0.5 for ep in endpoints}
        ages = {ep.name: self._last_routed.get(ep.name, -1.0)
                for ep in endpoints}
        return _minmax_inverted(ages)  # least-recent = lowest stamp -> 1.0

    def touch(self, endpoint_name: str) -> None:
        self._clock += 1.0
        self._last_routed[endpoint_name] = self._clock

    def pre_request(self, ctx, result, target) -> None:
        if target is not None:
            self.touch(target.name)


@register_plugin("context-length-aware-scorer", aliases=["context-length-aware"])
class ContextLengthAwareScorer(Scorer):
    """Pod label `llm-d.ai/context-length-range` "min-max" vs token count:
    in-range (0.3,1.0], out-of-range [0,0.3) (scorer/contextlengthaware)."""

    def score(self, ctx: SchedulingContext, endpoints):
        n_tokens = len(ctx.request.prompt_tokens or []) or \
            max(1, ctx.request.prompt_len_chars // 4)
        out = {}
        for ep in endpoints:
            rng = ep.metadata.labels.get(CONTEXT_LENGTH_RANGE_LABEL)
            if not rng or "-" not in rng:
                out[ep.name] = 0.5
                continue
            lo_s, hi_s = rng.split("-", 1)
            try:
                lo, hi = int(lo_s), int(hi_s)
            except ValueError:
                out[ep.name] = 0.5
                continue
            if lo <= n_tokens <= hi:
                # tighter fit -> higher score, in (0.3, 1.0]
                span = max(1, hi - lo)
                out[ep.name] = 1.0 - 0.7 * min(1.0, (hi - n_tokens) / span) * 0.99
            else:
                dist = (lo - n_tokens) if n_tokens < lo else (n_tokens - hi)
                out[ep.name] = max(0.0, 0.3 - 0.3 * min(1.0, dist / max(1, hi)))
        return out


@register_plugin("latency-scorer")
class LatencyScorer(Scorer):
    """Predicted TTFT/TPOT headroom vs SLO (scorer/latency)."""

    def score(self, ctx: SchedulingContext, endpoints):
        lat = ctx.attributes.get(LATENCY_PREDICTION_INFO)
        if lat is None or not lat.ttft_headroom_ms:
            return {ep.name: 0.5 for ep in endpoints}
        out = {}
        for ep in endpoints:
            h_ttft = lat.ttft_headroom_ms.get(ep.name)
            h_tpot = lat.tpot_headroom_ms.get(ep.name, 0.0)
            if h_ttft is None:
                out[ep.name] = 0.5
                continue
            h = min(h_ttft, h_tpot if h_tpot is not None else h_ttft)
            # squash headroom (ms) into [0,1]; 0 headroom -> 0.5
            out[ep.name] = 1.0 / (1.0 + math.exp(-h / 100.0))
        return out


@register_plugin("precise-prefix-cache-scorer")
class PrecisePrefixCacheScorer(Scorer):
    """Exact KV-block index scorer (scorer/preciseprefixcache/
    precise_prefix_cache.go). The reference subscribes to per-pod vLLM KV
    events over ZMQ; here the engines' BlockManagers emit real content
    hashes over the node mailbox into a KVBlockIndex (datalayer/kvblock.py),
    plus 2s-TTL speculative entries written at PreRequest (:533-604).

    Hashing uses the ENGINE's chained-block scheme (engine/kvcache.py
    block_hashes) so scorer lookups match event hashes bit-for-bit.
    """

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        from ..datalayer.kvblock import KVBlockIndex
        self.block_size = int(params.get("blockSize", 16))
        self.index = KVBlockIndex(
            float(params.get("speculativeTTLSeconds", 2.0)))

    def _hashes(self, ctx: SchedulingContext):
        cached = getattr(ctx, "_precise_hashes", None)
        if cached is not None:
            return cached
        from ..engine.kvcache import block_hashes
        toks = ctx.request.prompt_tokens or []
        h = [int(x) for x in block_hashes(toks, self.block_size)]
        ctx._precise_hashes = h
        return h

    def score(self, ctx: SchedulingContext, endpoints):
        hashes = self._hashes(ctx)
        if not hashes:
            return {ep.name: 0.0 for ep in endpoints}
        matched = self.index.match_longest(hashes,
                                           [ep.name for ep in endpoints])
        total = len(hashes)
        return {name: n / total for name, n in matched.items()}

    def pre_request(self, ctx: SchedulingContext, result, target) -> None:
        if target is None:
            return
        hashes = self._hashes(ctx)
        if hashes:
            self.index.add_speculative(target.name, hashes)

    def apply_events(self, endpoint_name: str, stored, evicted) -> None:
        self.index.apply_events(endpoint_name, stored, evicted)
        self.index.sweep()

    def remove_endpoint(self, ep: Endpoint) -> None:
        self.index.remove_endpoint(ep.name)
