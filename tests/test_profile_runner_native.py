"""Native ProfileRunner edge semantics (csrc/router/scoring.h).

Pins the behaviors the reference specifies in scheduler_profile.go:117-202
and framework/plugins/scheduling/picker/weightedrandom: score clamping,
uniform fallback when every weighted score is <= 0, threshold-param scorers,
and filter-mask intersection.
"""
import collections

import numpy as np
import pytest

from llm_d_inference_scheduler_amd import _router_core as rc

SC_QUEUE, SC_KV, SC_PREFIX, SC_RUNNING, SC_LOAD, SC_TOKLOAD, SC_ACTIVE = range(7)
PK_MAX, PK_RANDOM, PK_WEIGHTED = range(3)


def run(runner, n=4, roles=None, queue=None, kv=None, tokens=None,
        active=None, running=None, role_filter=0, mask=None, scorers=(),
        match=None, total=0, extra=None, picker=PK_MAX, k=1):
    z = np.zeros(n, dtype=np.float32)
    return runner.run(
        np.full(n, 1, dtype=np.uint8) if roles is None else roles,
        z if queue is None else queue, z if running is None else running,
        z if kv is None else kv, z if tokens is None else tokens,
        z if active is None else active,
        role_filter, mask, list(scorers), match, total, extra, picker, k)


class TestWeightedRandom:
    def test_uniform_fallback_when_all_zero(self):
        """All weighted scores 0 -> uniform pick, never an empty result
        (weightedrandom fallback)."""
        r = rc.ProfileRunner(7)
        counts = collections.Counter()
        for _ in range(400):
            picks, scores = run(r, n=4, scorers=[(SC_KV, 1.0, 0.0, 0.0)],
                                kv=np.ones(4, dtype=np.float32),
                                picker=PK_WEIGHTED)
            assert len(picks) == 1
            counts[int(picks[0])] += 1
        assert all(scores[i] == 0.0 for i in range(4))
        assert len(counts) == 4          # every endpoint reachable
        assert max(counts.values()) < 250

    def test_proportional_to_score(self):
        """2x score -> picked roughly 2x as often (A-Res keys)."""
        r = rc.ProfileRunner(11)
        kv = np.array([0.2, 0.6], dtype=np.float32)   # scores 0.8 / 0.4
        counts = collections.Counter()
        for _ in range(3000):
            picks, _ = run(r, n=2, scorers=[(SC_KV, 1.0, 0.0, 0.0)], kv=kv,
                           picker=PK_WEIGHTED)
            counts[int(picks[0])] += 1
        ratio = counts[0] / max(1, counts[1])
        assert 1.5 < ratio < 2.7

    def test_zero_scored_sorted_after_positive(self):
        r = rc.ProfileRunner(3)
        kv = np.array([1.0, 0.3, 1.0], dtype=np.float32)  # scores 0,0.7,0
        picks, _ = run(r, n=3, scorers=[(SC_KV, 1.0, 0.0, 0.0)], kv=kv,
                       picker=PK_WEIGHTED, k=3)
        assert int(picks[0]) == 1


class TestThresholdScorers:
    def test_load_aware_param(self):
        """score = max(0, 0.5*(1 - queue/threshold)) with param threshold."""
        r = rc.ProfileRunner(1)
        q = np.array([0.0, 5.0, 10.0, 20.0], dtype=np.float32)
        _, scores = run(r, n=4, scorers=[(SC_LOAD, 1.0, 10.0, 0.0)], queue=q)
        np.testing.assert_allclose(scores, [0.5, 0.25, 0.0, 0.0], atol=1e-6)

    def test_token_load_param(self):
        r = rc.ProfileRunner(1)
        t = np.array([0.0, 500.0, 1000.0, 4000.0], dtype=np.float32)
        _, scores = run(r, n=4, scorers=[(SC_TOKLOAD, 2.0, 1000.0, 0.0)], tokens=t)
        np.testing.assert_allclose(scores, [2.0, 1.0, 0.0, 0.0], atol=1e-6)


class TestFiltersAndPrefix:
    def test_role_and_mask_intersection(self):
        roles = np.array([1, 2, 3, 1], dtype=np.uint8)   # decode bit = 1
        mask = np.array([1, 1, 1, 0], dtype=np.uint8)
        r = rc.ProfileRunner(1)
        picks, scores = run(r, n=4, roles=roles, role_filter=1, mask=mask,
                            scorers=[(SC_KV, 1.0, 0.0, 0.0)], k=4)
        # only endpoints 0 and 2 survive (decode role AND mask)
        assert sorted(int(i) for i in picks) == [0, 2]
        assert scores[1] == -1.0 and scores[3] == -1.0

    def test_prefix_ratio_and_zero_total(self):
        r = rc.ProfileRunner(1)
        match = np.array([3, 1, 0], dtype=np.int32)
        _, scores = run(r, n=3, scorers=[(SC_PREFIX, 1.0, 0.0, 0.0)],
                        match=match, total=4)
        np.testing.assert_allclose(scores, [0.75, 0.25, 0.0], atol=1e-6)
        _, scores = run(r, n=3, scorers=[(SC_PREFIX, 1.0, 0.0, 0.0)],
                        match=match, total=0)   # no blocks -> all zero, no div0
        np.testing.assert_allclose(scores, [0.0, 0.0, 0.0], atol=1e-6)

    def test_minmax_single_endpoint(self):
        """Degenerate span -> score 1.0 (not NaN)."""
        r = rc.ProfileRunner(1)
        _, scores = run(r, n=1, scorers=[(SC_QUEUE, 1.0, 0.0, 0.0)],
                        queue=np.array([42.0], dtype=np.float32))
        assert scores[0] == 1.0

    def test_all_filtered_returns_empty(self):
        r = rc.ProfileRunner(1)
        picks, scores = run(r, n=2, roles=np.array([2, 2], dtype=np.uint8),
                            role_filter=1, scorers=[(SC_KV, 1.0, 0.0, 0.0)])
        assert len(picks) == 0 and all(s == -1.0 for s in scores)


class TestActiveRequestIdleBusy:
    """active_request.go:139-168 semantics through BOTH paths: idle pins
    1.0, busy scales (max-c)/max * maxBusyScore."""

    def test_native_formula(self):
        r = rc.ProfileRunner(1)
        active = np.array([0.0, 2.0, 4.0, 8.0], dtype=np.float32)
        # idleThreshold=2, maxBusyScore=0.5
        _, scores = run(r, n=4, active=active,
                        scorers=[(SC_ACTIVE, 1.0, 2.0, 0.5)])
        np.testing.assert_allclose(
            scores, [1.0, 1.0, (8 - 4) / 8 * 0.5, 0.0], atol=1e-6)

    def test_defaults_max_normalized(self):
        r = rc.ProfileRunner(1)
        active = np.array([2.0, 4.0], dtype=np.float32)
        _, scores = run(r, n=2, active=active,
                        scorers=[(SC_ACTIVE, 1.0, 0.0, 0.0)])
        np.testing.assert_allclose(scores, [0.5, 0.0], atol=1e-6)

    def test_python_native_parity(self):
        from llm_d_inference_scheduler_amd.datalayer.attributes import \
            IN_FLIGHT_LOAD, InFlightLoad
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        from llm_d_inference_scheduler_amd.plugins.scorers import \
            ActiveRequestScorer
        sc = ActiveRequestScorer("a", idleThreshold=1, maxBusyScore=0.8)
        eps = []
        for i, cnt in enumerate([0, 1, 3, 6]):
            ep = make_endpoint(f"e{i}", i)
            load = InFlightLoad()
            load.add(cnt, cnt * 10)
            ep.put_attribute(IN_FLIGHT_LOAD, load)
            eps.append(ep)
        py = sc.score(None, eps)
        r = rc.ProfileRunner(1)
        active = np.array([0, 1, 3, 6], dtype=np.float32)
        k, p1, p2 = sc.native_spec()
        _, native = run(r, n=4, active=active, scorers=[(k, 1.0, p1, p2)])
        for i in range(4):
            assert abs(py[f"e{i}"] - native[i]) < 1e-6, (i, py, native)
