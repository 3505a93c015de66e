"""Scheduler core (parity: pkg/epp/scheduling/scheduler.go:54-102,
scheduler_profile.go:117-192, weighted_scorer.go, scheduler_config.go).

`Scheduler.schedule` loops ProfileHandler.pick_profiles -> run each returned
SchedulerProfile -> ProfileHandler.process_results. A profile run is
filters (sequential) -> weighted scorers (clamped to [0,1]) -> one picker.

MI355X-native hot path: when every scorer in the profile advertises a native
spec, the weighted-score + pick stage executes in the C++ core
(`_router_core.ProfileRunner`) over dense endpoint-snapshot arrays; python
scorer plugins contribute through the pre-weighted `extra` array. Tests
assert python/native parity (tests/test_scheduler.py).
"""
import random
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from .. import _router_core as rc
from ..datalayer.endpoint import Endpoint

if False:  # import-cycle break: plugins.interface imports scheduling.types
    from ..plugins.interface import Filter, Picker, ProfileHandler, Scorer  # noqa
from typing import Any
Filter = Picker = ProfileHandler = Scorer = Any  # structural typing only

from ..telemetry import get_tracer
from ..utils.logging import get_logger
from .types import LLMRequest, ProfileRunResult, SchedulingContext, SchedulingResult
from ..datalayer.attributes import PREFIX_CACHE_MATCH_INFO
from ..metrics import prom

log = get_logger("scheduler")

SC_PREFIX_KIND = 2  # mirrors csrc/router/scoring.h ScorerKind::SC_PREFIX


@dataclass
class SchedulerProfile:
    name: str
    filters: List[Filter] = field(default_factory=list)
    scorers: List[Tuple[Scorer, float]] = field(default_factory=list)
    picker: Optional[Picker] = None
    max_endpoints: int = 1

    def __post_init__(self):
        self._runner = rc.ProfileRunner(hash(self.name) & 0x7FFFFFFF)

    def run(self, ctx: SchedulingContext,
            endpoints: List[Endpoint]) -> ProfileRunResult:
        result = ProfileRunResult(profile_name=self.name)
        eps = list(endpoints)
        for f in self.filters:
            eps = f.filter(ctx, eps)
            if not eps:
                return result
        native_specs: List[Tuple[int, float, float, float]] = []
        python_scorers: List[Tuple[Scorer, float]] = []
        for scorer, weight in self.scorers:
            spec = scorer.native_spec()
            if spec is not None:
                p2 = float(spec[2]) if len(spec) > 2 else 0.0
                native_specs.append((spec[0], float(weight),
                                     float(spec[1]), p2))
            else:
                python_scorers.append((scorer, weight))

        extra = np.zeros(len(eps), dtype=np.float32)
        for scorer, weight in python_scorers:
            with prom.L(prom.plugin_latency, scorer.name).time():
                smap = scorer.score(ctx, eps)
            for i, ep in enumerate(eps):
                v = min(1.0, max(0.0, smap.get(ep.name, 0.0)))
                extra[i] += weight * v

        picker_kind = getattr(self.picker, "native_kind", None)
        use_native_pick = picker_kind is not None

        n = len(eps)
        snap = _snapshot_arrays(eps)
        match_blocks, total_blocks = _prefix_arrays(ctx, eps, native_specs)
        picks_idx, scores = self._runner.run(
            snap["roles"], snap["queue"], snap["running"], snap["kv"],
            snap["tokens"], snap["active"],
            0, None, native_specs, match_blocks, total_blocks,
            extra if len(python_scorers) else None,
            picker_kind if use_native_pick else 0,
            self.max_endpoints if use_native_pick else n)

        result.scores = {eps[i].name: float(scores[i]) for i in range(n)}
        if use_native_pick:
            result.picks = [eps[i] for i in picks_idx]
        else:
            result.picks = self.picker.pick(ctx, result.scores, eps,
                                            self.max_endpoints)
        return result


def _snapshot_arrays(eps: List[Endpoint]) -> Dict[str, np.ndarray]:
    n = len(eps)
    snap = {
        "roles": np.zeros(n, dtype=np.uint8),
        "queue": np.zeros(n, dtype=np.float32),
        "running": np.zeros(n, dtype=np.float32),
        "kv": np.zeros(n, dtype=np.float32),
        "tokens": np.zeros(n, dtype=np.float32),
        "active": np.zeros(n, dtype=np.float32),
    }
    from ..datalayer.attributes import IN_FLIGHT_LOAD
    for i, ep in enumerate(eps):
        m = ep.metrics
        snap["roles"][i] = ep.metadata._mask() & 0xFF
        snap["queue"][i] = m.waiting_queue_size
        snap["running"][i] = m.running_requests_size
        snap["kv"][i] = m.kv_cache_usage
        load = ep.get_attribute(IN_FLIGHT_LOAD)
        if load is not None:
            reqs, toks = load.snapshot()
            snap["tokens"][i] = toks
            snap["active"][i] = reqs
    return snap


def _prefix_arrays(ctx, eps, native_specs):
    if not any(sp[0] == SC_PREFIX_KIND for sp in native_specs):
        return None, 0
    info = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
    if info is None:
        return np.zeros(len(eps), dtype=np.int32), 0
    arr = np.array([info.match_blocks.get(ep.name, 0) for ep in eps],
                   dtype=np.int32)
    return arr, info.total_blocks


@dataclass
class SchedulerConfig:
    profiles: Dict[str, SchedulerProfile] = field(default_factory=dict)
    profile_handler: Optional[ProfileHandler] = None


class Scheduler:
    """Scheduler.Schedule loop (scheduler.go:54)."""

    def __init__(self, config: SchedulerConfig):
        self.config = config
        if config.profile_handler is None:
            raise ValueError("SchedulerConfig requires a profile handler")

    def schedule(self, ctx: SchedulingContext,
                 endpoints: List[Endpoint]) -> SchedulingResult:
        tracer = get_tracer()
        with tracer.span("scheduler.schedule",
                         request_id=ctx.request.request_id) as span:
            handler = self.config.profile_handler
            results: Dict[str, ProfileRunResult] = {}
            while True:
                names = handler.pick_profiles(ctx, self.config.profiles, results)
                names = [n for n in names if n not in results]
                if not names:
                    break
                for name in names:
                    profile = self.config.profiles.get(name)
                    if profile is None:
                        raise KeyError(f"unknown scheduling profile {name!r}")
                    with prom.plugin_latency.labels(f"profile:{name}").time():
                        results[name] = profile.run(ctx, endpoints)
            primary = handler.process_results(ctx, results)
            span.set_attribute("profiles", list(results))
            span.set_attribute("primary", primary)
            return SchedulingResult(profile_results=results,
                                    primary_profile=primary)
