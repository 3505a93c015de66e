"""Scheduler core: profile run, native/python scorer parity, profile handlers."""
import numpy as np
import pytest

from llm_d_inference_scheduler_amd.datalayer.attributes import (
    PREFIX_CACHE_MATCH_INFO, PrefixCacheMatchInfo)
from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
from llm_d_inference_scheduler_amd.plugins import register_all_plugins
from llm_d_inference_scheduler_amd.plugins.registry import global_registry
from llm_d_inference_scheduler_amd.plugins.profile_handlers import (
    DisaggProfileHandler, PrefixBasedPDDecider, SingleProfileHandler)
from llm_d_inference_scheduler_amd.scheduling.scheduler import (
    Scheduler, SchedulerConfig, SchedulerProfile)
from llm_d_inference_scheduler_amd.scheduling.types import (
    LLMRequest, SchedulingContext)

register_all_plugins()


def mk(type_name, **params):
    return global_registry.instantiate(type_name, **params)


def set_metrics(ep, queue=0, running=0, kv=0.0):
    m = Metrics(waiting_queue_size=queue, running_requests_size=running,
                kv_cache_usage=kv)
    ep.update_metrics(m)


@pytest.fixture
def ctx(request_factory):
    return SchedulingContext(request=request_factory())


class TestProfileRun:
    def test_native_pick_least_loaded(self, endpoints, ctx):
        for i, ep in enumerate(endpoints):
            set_metrics(ep, queue=i * 2, kv=i * 0.2)
        prof = SchedulerProfile(
            name="p", scorers=[(mk("queue-scorer"), 1.0),
                               (mk("kv-cache-utilization-scorer"), 1.0)],
            picker=mk("max-score-picker"))
        res = prof.run(ctx, endpoints)
        assert res.target.name == "gpu0"
        assert res.scores["gpu0"] == pytest.approx(2.0)

    def test_python_native_parity(self, endpoints, ctx):
        """Python scorer formulas == native formulas for the same state."""
        for i, ep in enumerate(endpoints):
            set_metrics(ep, queue=[3, 0, 7, 2][i], running=[1, 5, 2, 0][i],
                        kv=[0.1, 0.9, 0.4, 0.0][i])
        ctx.attributes[PREFIX_CACHE_MATCH_INFO] = PrefixCacheMatchInfo(
            match_blocks={"gpu0": 2, "gpu2": 6}, total_blocks=8)
        scorers = ["queue-scorer", "kv-cache-utilization-scorer",
                   "prefix-cache-scorer", "running-requests-size-scorer",
                   "load-aware-scorer"]
        weights = [1.0, 2.0, 3.0, 0.5, 1.5]
        plugins = [mk(s) for s in scorers]
        # native run
        prof = SchedulerProfile(name="n",
                                scorers=list(zip(plugins, weights)),
                                picker=mk("max-score-picker"))
        res_native = prof.run(ctx, endpoints)
        # python-side expected
        expected = {ep.name: 0.0 for ep in endpoints}
        for plugin, w in zip(plugins, weights):
            smap = plugin.score(ctx, endpoints)
            for name, v in smap.items():
                expected[name] += w * min(1.0, max(0.0, v))
        for name in expected:
            assert res_native.scores[name] == pytest.approx(expected[name],
                                                            abs=1e-5)

    def test_filter_chain(self, endpoints, ctx):
        prof = SchedulerProfile(name="p", filters=[mk("prefill-filter")],
                                scorers=[(mk("queue-scorer"), 1.0)],
                                picker=mk("max-score-picker"))
        res = prof.run(ctx, endpoints)
        assert res.target.name == "gpu3"  # only prefill-decode endpoint

    def test_empty_filter_result(self, endpoints, ctx):
        prof = SchedulerProfile(name="p", filters=[mk("encode-filter")],
                                picker=mk("max-score-picker"))
        res = prof.run(ctx, endpoints)
        assert res.target is None

    def test_max_endpoints(self, endpoints, ctx):
        prof = SchedulerProfile(name="p",
                                scorers=[(mk("queue-scorer"), 1.0)],
                                picker=mk("max-score-picker"),
                                max_endpoints=3)
        res = prof.run(ctx, endpoints)
        assert len(res.picks) == 3

    def test_python_scorer_in_profile(self, endpoints, ctx):
        """A non-native scorer flows through the `extra` array."""
        sess = mk("session-affinity-scorer")
        sess.remember("alice", "gpu2")
        ctx.request.session_id = "alice"
        prof = SchedulerProfile(name="p", scorers=[(sess, 5.0)],
                                picker=mk("max-score-picker"))
        res = prof.run(ctx, endpoints)
        assert res.target.name == "gpu2"
        assert res.scores["gpu2"] == pytest.approx(5.0)


class TestSchedulerLoop:
    def test_single_profile(self, endpoints, ctx):
        prof = SchedulerProfile(name="default",
                                scorers=[(mk("queue-scorer"), 1.0)],
                                picker=mk("max-score-picker"))
        sched = Scheduler(SchedulerConfig(
            profiles={"default": prof},
            profile_handler=SingleProfileHandler()))
        res = sched.schedule(ctx, endpoints)
        assert res.primary_profile == "default"
        assert res.primary.target is not None

    def _disagg_scheduler(self, non_cached_tokens=8):
        decode = SchedulerProfile(name="decode", filters=[mk("decode-filter")],
                                  scorers=[(mk("queue-scorer"), 1.0),
                                           (mk("prefix-cache-scorer"), 2.0)],
                                  picker=mk("max-score-picker"))
        prefill = SchedulerProfile(name="prefill",
                                   filters=[mk("prefill-filter")],
                                   scorers=[(mk("queue-scorer"), 1.0)],
                                   picker=mk("max-score-picker"))
        handler = DisaggProfileHandler(
            pdDecider=PrefixBasedPDDecider(nonCachedTokens=non_cached_tokens))
        return Scheduler(SchedulerConfig(
            profiles={"decode": decode, "prefill": prefill},
            profile_handler=handler))

    def test_disagg_fires_on_long_uncached_prompt(self, endpoints,
                                                  request_factory):
        req = request_factory()
        req.prompt_tokens = list(range(1000))
        ctx = SchedulingContext(request=req)
        sched = self._disagg_scheduler(non_cached_tokens=512)
        res = sched.schedule(ctx, endpoints)
        assert res.primary_profile == "decode"
        assert "prefill" in res.profile_results
        assert res.profile_results["prefill"].target.name == "gpu3"
        # PreRequest publishes the prefiller header
        handler = sched.config.profile_handler
        handler.pre_request(ctx, res, res.primary.target)
        assert "x-prefiller-host-port" in req.headers

    def test_disagg_skipped_on_short_prompt(self, endpoints, request_factory):
        req = request_factory()
        req.prompt_tokens = list(range(100))
        ctx = SchedulingContext(request=req)
        sched = self._disagg_scheduler(non_cached_tokens=512)
        res = sched.schedule(ctx, endpoints)
        assert "prefill" not in res.profile_results

    def test_disagg_skipped_when_cached(self, endpoints, request_factory):
        req = request_factory()
        req.prompt_tokens = list(range(1000))
        ctx = SchedulingContext(request=req)
        # decode target has ~all blocks cached
        ctx.attributes[PREFIX_CACHE_MATCH_INFO] = PrefixCacheMatchInfo(
            match_blocks={ep: 62 for ep in
                          ("gpu0", "gpu1", "gpu2", "gpu3")},
            total_blocks=62, block_size_tokens=16)
        sched = self._disagg_scheduler(non_cached_tokens=512)
        res = sched.schedule(ctx, endpoints)
        assert "prefill" not in res.profile_results


class TestSaturationDetectorPlugins:
    def test_utilization_filter_fail_open(self):
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
        from llm_d_inference_scheduler_amd.plugins.registry import global_registry
        f = global_registry.instantiate("utilization-detector",
                                 queueDepthThreshold=5,
                                 metricsStalenessSeconds=1e9)
        eps = [make_endpoint(f"gpu{i}", i) for i in range(3)]
        eps[0].update_metrics(Metrics(waiting_queue_size=0))
        eps[1].update_metrics(Metrics(waiting_queue_size=10))   # saturated
        eps[2].update_metrics(Metrics(waiting_queue_size=2))
        kept = f.filter(None, eps)
        assert [e.name for e in kept] == ["gpu0", "gpu2"]
        # all saturated -> fail-open returns everything
        for ep in eps:
            ep.update_metrics(Metrics(waiting_queue_size=50))
        assert len(f.filter(None, eps)) == 3
        assert f.is_saturated(eps)

    def test_concurrency_detector_plugin(self):
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        from llm_d_inference_scheduler_amd.plugins.registry import global_registry
        d = global_registry.instantiate("concurrency-detector",
                                 maxInflightPerEndpoint=1)
        eps = [make_endpoint("gpu0", 0)]
        assert d.filter(None, eps) == eps
        assert not d.is_saturated(eps)


class TestTracerExport:
    def test_env_export_jsonl(self, tmp_path, monkeypatch):
        import importlib
        monkeypatch.setenv("LLMD_TRACE_EXPORT", str(tmp_path / "spans.jsonl"))
        from llm_d_inference_scheduler_amd.telemetry import tracing
        t = tracing.Tracer("test-svc")
        with t.span("gateway.request", model="m") as s:
            s.set_attribute("k", 1)
        t.export_jsonl(str(tmp_path / "spans.jsonl"))
        import json
        lines = (tmp_path / "spans.jsonl").read_text().strip().splitlines()
        rec = json.loads(lines[-1])
        assert rec["name"] == "gateway.request"

    def test_tracing_disabled_env(self, monkeypatch):
        monkeypatch.setenv("LLMD_TRACING", "0")
        from llm_d_inference_scheduler_amd.telemetry.tracing import Tracer
        t = Tracer("svc")
        with t.span("x") as s:
            s.set_attribute("a", 1)   # noop span accepts attributes
        assert not t.finished_spans()


class TestTraceparent:
    def test_parse_valid_and_invalid(self):
        from llm_d_inference_scheduler_amd.telemetry.tracing import \
            parse_traceparent
        tid = "a" * 32
        sid = "b" * 16
        assert parse_traceparent(f"00-{tid}-{sid}-01") == (tid, sid)
        for bad in (None, "", "junk", f"00-{tid}-{sid}", f"00-{'z'*32}-{sid}-01",
                    f"00-{'0'*32}-{sid}-01", f"00-{tid}-{'0'*16}-01",
                    f"0-{tid}-{sid}-01"):
            assert parse_traceparent(bad) is None

    def test_root_span_adopts_remote_context_and_children_inherit(self):
        from llm_d_inference_scheduler_amd.telemetry import (get_tracer,
                                                             init_tracing)
        from llm_d_inference_scheduler_amd.telemetry.tracing import \
            parse_traceparent
        init_tracing(enabled=True)
        tracer = get_tracer()
        tid = "c" * 32
        tp = parse_traceparent(f"00-{tid}-{'d'*16}-01")
        with tracer.span("gateway.request") as root:
            root.trace_id, _ = tp
            with tracer.span("scheduler.schedule"):
                pass
        spans = {s.name: s for s in tracer.finished_spans()}
        assert spans["gateway.request"].trace_id == tid
        assert spans["scheduler.schedule"].trace_id == tid
        assert spans["scheduler.schedule"].parent == "gateway.request"


class TestOtlpExport:
    """OTLP/HTTP+JSON trace export (pkg/telemetry/tracing.go analog),
    env-gated via OTEL_EXPORTER_OTLP_ENDPOINT."""

    def test_payload_shape_and_batching(self):
        from llm_d_inference_scheduler_amd.telemetry.tracing import (
            OtlpHttpExporter, Tracer)

        posts = []

        class FakeResp:
            status_code = 200

        class FakeClient:
            def post(self, url, json=None):
                posts.append((url, json))
                return FakeResp()

        tr = Tracer("svc-test")
        tr.otlp = OtlpHttpExporter("http://collector:4318", "svc-test",
                                   batch=2, client=FakeClient())
        with tr.span("outer", model="m") as outer:
            with tr.span("inner"):
                pass
        tr.otlp.flush()
        assert tr.otlp.sent == 2
        url, body = posts[-1]
        assert url == "http://collector:4318/v1/traces"
        rs = body["resourceSpans"][0]
        svc = rs["resource"]["attributes"][0]
        assert svc["key"] == "service.name"
        spans = rs["scopeSpans"][0]["spans"]
        by_name = {s["name"]: s for s in spans}
        assert set(by_name) <= {"inner", "outer"}
        inner = by_name["inner"]
        assert len(inner["traceId"]) == 32 and len(inner["spanId"]) == 16
        # child carries its parent's span id and trace id
        if "outer" in by_name:
            assert inner["parentSpanId"] == by_name["outer"]["spanId"]
            assert inner["traceId"] == by_name["outer"]["traceId"]
        assert int(inner["endTimeUnixNano"]) >= int(
            inner["startTimeUnixNano"])

    def test_fail_open_on_collector_down(self):
        from llm_d_inference_scheduler_amd.telemetry.tracing import (
            OtlpHttpExporter, Span)

        class DeadClient:
            def post(self, *a, **k):
                raise ConnectionError("collector down")
        ex = OtlpHttpExporter("http://dead:4318", "svc", client=DeadClient())
        ex.enqueue(Span(name="s", start_ns=1, end_ns=2, trace_id="a" * 32,
                        span_id="b" * 16))
        ex.flush()
        assert ex.dropped == 1 and ex.sent == 0


class TestContextLengthAwareScorer:
    """scorer/contextlengthaware parity: in-range (0.3,1.0], out [0,0.3),
    unlabeled neutral 0.5."""

    def _eps(self):
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        short = make_endpoint("short", 0, labels={
            "llm-d.ai/context-length-range": "0-2048"})
        long_ = make_endpoint("long", 1, labels={
            "llm-d.ai/context-length-range": "2048-32768"})
        plain = make_endpoint("plain", 2)
        bad = make_endpoint("bad", 3, labels={
            "llm-d.ai/context-length-range": "oops"})
        return [short, long_, plain, bad]

    def _score(self, n_tokens):
        from llm_d_inference_scheduler_amd.plugins.scorers import \
            ContextLengthAwareScorer
        from llm_d_inference_scheduler_amd.scheduling.types import (
            LLMRequest, SchedulingContext)
        req = LLMRequest(request_id="r", model="m",
                         prompt_tokens=list(range(n_tokens)))
        return ContextLengthAwareScorer("cl").score(
            SchedulingContext(request=req), self._eps())

    def test_short_prompt_prefers_short_range(self):
        s = self._score(256)
        assert s["short"] > 0.3          # in range
        assert s["long"] < 0.3           # out of range
        assert s["plain"] == 0.5         # unlabeled neutral
        assert s["bad"] == 0.5           # malformed label neutral

    def test_long_prompt_prefers_long_range(self):
        s = self._score(8000)
        assert s["long"] > 0.3 and s["short"] < 0.3

    def test_tighter_fit_scores_higher(self):
        # near the top of the short range beats the bottom of it
        hi = self._score(2000)["short"]
        lo = self._score(64)["short"]
        assert hi > lo > 0.3


class TestLatencySLOAdmitter:
    def _ctx(self, priority=0, headroom=None):
        from llm_d_inference_scheduler_amd.datalayer.attributes import (
            LATENCY_PREDICTION_INFO, LatencyPredictionInfo)
        from llm_d_inference_scheduler_amd.scheduling.types import (
            LLMRequest, SchedulingContext)
        req = LLMRequest(request_id="r", model="m", priority=priority)
        ctx = SchedulingContext(request=req)
        if headroom is not None:
            info = LatencyPredictionInfo()
            info.ttft_headroom_ms.update(headroom)
            ctx.attributes[LATENCY_PREDICTION_INFO] = info
        return ctx

    def _eps(self):
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        return [make_endpoint("a", 0), make_endpoint("b", 1)]

    def test_critical_always_admitted(self):
        from llm_d_inference_scheduler_amd.plugins.admitters import \
            LatencySLOAdmitter
        adm = LatencySLOAdmitter("slo")
        ok, _ = adm.admit(self._ctx(priority=5,
                                    headroom={"a": -100, "b": -100}),
                          self._eps())
        assert ok

    def test_sheddable_rejected_with_no_headroom(self):
        from llm_d_inference_scheduler_amd.plugins.admitters import \
            LatencySLOAdmitter
        adm = LatencySLOAdmitter("slo")
        ok, reason = adm.admit(self._ctx(priority=-1,
                                         headroom={"a": -5, "b": -1}),
                               self._eps())
        assert not ok and "SLO" in reason

    def test_admitted_when_any_endpoint_has_headroom(self):
        from llm_d_inference_scheduler_amd.plugins.admitters import \
            LatencySLOAdmitter
        adm = LatencySLOAdmitter("slo")
        ok, _ = adm.admit(self._ctx(priority=-1,
                                    headroom={"a": -5, "b": 40}),
                          self._eps())
        assert ok

    def test_fail_open_without_predictions(self):
        from llm_d_inference_scheduler_amd.plugins.admitters import \
            LatencySLOAdmitter
        adm = LatencySLOAdmitter("slo")
        ok, _ = adm.admit(self._ctx(priority=-1), self._eps())
        assert ok
