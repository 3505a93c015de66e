"""Picker plugins (parity: pkg/epp/framework/plugins/scheduling/picker/*).

All honor maxNumOfEndpoints (default 1, picker/maxscore/picker.go:47-60).
Each declares `native_kind` so the C++ ProfileRunner performs the pick in
the hot loop; the python implementations exist for parity tests and custom
profiles.
"""
import math
import random
from typing import Dict, List

from ..datalayer.endpoint import Endpoint
from .interface import Picker
from .registry import register_plugin


@register_plugin("max-score-picker")
class MaxScorePicker(Picker):
    native_kind = 0  # PK_MAX_SCORE

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._rng = random.Random(params.get("seed", 0x5EED))

    def pick(self, ctx, scored: Dict[str, float], endpoints: List[Endpoint],
             max_endpoints: int) -> List[Endpoint]:
        keyed = [((scored.get(ep.name, 0.0), self._rng.random()), ep)
                 for ep in endpoints]
        keyed.sort(key=lambda t: t[0], reverse=True)  # ties random
        return [ep for _, ep in keyed[:max(1, max_endpoints)]]


@register_plugin("random-picker")
class RandomPicker(Picker):
    native_kind = 1  # PK_RANDOM

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._rng = random.Random(params.get("seed", 0xA11CE))

    def pick(self, ctx, scored, endpoints, max_endpoints):
        eps = list(endpoints)
        self._rng.shuffle(eps)
        return eps[:max(1, max_endpoints)]


@register_plugin("weighted-random-picker")
class WeightedRandomPicker(Picker):
    """A-Res reservoir sampling: key = U^(1/score); uniform fallback when
    all scores <= 0 (picker/weightedrandom/README.md)."""
    native_kind = 2  # PK_WEIGHTED_RANDOM

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self._rng = random.Random(params.get("seed", 0xBEEF))

    def pick(self, ctx, scored, endpoints, max_endpoints):
        any_pos = any(scored.get(ep.name, 0.0) > 0 for ep in endpoints)
        keyed = []
        for ep in endpoints:
            u = self._rng.random()
            s = scored.get(ep.name, 0.0)
            if any_pos:
                key = math.pow(u, 1.0 / s) if s > 0 else -1.0
            else:
                key = u
            keyed.append((key, ep))
        keyed.sort(key=lambda t: t[0], reverse=True)
        return [ep for _, ep in keyed[:max(1, max_endpoints)]]
