"""InferenceModelRewrite (parity: apix/v1alpha2/inferencemodelrewrite_types.go:47-118).

Ordered match rules -> weighted targets. Exact model match beats generic
(empty) match; among several matching resources the oldest wins
(creation_seq tie-break in the datastore). The director rewrites the
request's model before scheduling and rewrites the response's model name
back (handlers/server.go:471 rewriteModelName).
"""
import random
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class RewriteTarget:
    model_rewrite: str
    weight: int = 1


@dataclass
class RewriteRule:
    # match: exact model name, or "" meaning any model (generic)
    model: str = ""
    targets: List[RewriteTarget] = field(default_factory=list)


@dataclass
class InferenceModelRewrite:
    name: str
    rules: List[RewriteRule] = field(default_factory=list)
    creation_seq: int = 0  # oldest-resource tie-break stand-in

    def match(self, model: str) -> Optional[RewriteRule]:
        generic = None
        for rule in self.rules:
            if rule.model == model:
                return rule
            if rule.model == "" and generic is None:
                generic = rule
        return generic


def select_weighted_target(rule: RewriteRule, rng: random.Random) -> Optional[str]:
    """Weighted selection among rule targets (director.go:317 selectWeightedModel)."""
    if not rule.targets:
        return None
    total = sum(max(0, t.weight) for t in rule.targets)
    if total <= 0:
        return rule.targets[0].model_rewrite
    pick = rng.uniform(0, total)
    acc = 0.0
    for t in rule.targets:
        acc += max(0, t.weight)
        if pick <= acc:
            return t.model_rewrite
    return rule.targets[-1].model_rewrite
