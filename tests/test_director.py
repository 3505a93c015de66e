"""Director orchestration: rewrite -> objective -> admission -> producers ->
schedule -> prepare; response hooks; prefix routing affinity end-to-end."""
import pytest

from llm_d_inference_scheduler_amd.api.modelrewrite import (
    InferenceModelRewrite, RewriteRule, RewriteTarget)
from llm_d_inference_scheduler_amd.api.objectives import InferenceObjective
from llm_d_inference_scheduler_amd.config import load_config
from llm_d_inference_scheduler_amd.datalayer.datastore import (Datastore,
                                                               make_endpoint)
from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
from llm_d_inference_scheduler_amd.flowcontrol import (
    FlowController, FlowRegistry, UtilizationSaturationDetector)
from llm_d_inference_scheduler_amd.handlers.parsers import Usage
from llm_d_inference_scheduler_amd.requestcontrol import (
    AdmissionDenied, Director, EndpointCandidates, LegacyAdmissionController)
from llm_d_inference_scheduler_amd.scheduling.scheduler import Scheduler

CONFIG = """
plugins:
  - type: queue-scorer
  - type: prefix-cache-scorer
  - type: kv-cache-utilization-scorer
  - type: inflight-load-producer
  - type: max-score-picker
schedulingProfiles:
  - name: default
    plugins:
      - pluginRef: prefix-cache-scorer
        weight: 3
      - pluginRef: queue-scorer
        weight: 1
      - pluginRef: kv-cache-utilization-scorer
        weight: 1
      - pluginRef: max-score-picker
"""


@pytest.fixture
def stack():
    cfg = load_config(CONFIG)
    ds = Datastore()
    for i in range(4):
        ep = make_endpoint(f"gpu{i}", i, rank=i, role="decode")
        ep.update_metrics(Metrics())
        ds.add_endpoint(ep)
    detector = UtilizationSaturationDetector()
    director = Director(
        datastore=ds,
        scheduler=Scheduler(cfg.scheduler_config),
        admission=LegacyAdmissionController(detector),
        candidates=EndpointCandidates(ds, cache_ttl_s=0),
        config=cfg.request_control)
    return ds, director


class TestDirector:
    def test_basic_routing(self, stack, request_factory):
        ds, director = stack
        d = director.handle_request(request_factory())
        assert d.target is not None
        assert d.request.headers["x-gateway-destination-endpoint"] == \
            d.target.metadata.address
        assert d.epp_latency_ms < 100

    def test_prefix_affinity_convergence(self, stack, request_factory):
        """Same long prompt repeatedly -> same endpoint (prefix cache
        affinity); different prompt can go elsewhere."""
        ds, director = stack
        prompt_a = "the quick brown fox " * 100
        first = director.handle_request(request_factory(prompt=prompt_a))
        for _ in range(5):
            d = director.handle_request(request_factory(prompt=prompt_a))
            assert d.target.name == first.target.name
        info = d.ctx.attributes["prefix.PrefixCacheMatchInfo"]
        assert info.match_blocks[first.target.name] > 0

    def test_model_rewrite(self, stack, request_factory):
        ds, director = stack
        ds.put_model_rewrite(InferenceModelRewrite(
            name="rw", rules=[RewriteRule(
                model="llama-3-8b",
                targets=[RewriteTarget("llama-3-8b-instruct", weight=1)])]))
        d = director.handle_request(request_factory())
        assert d.request.target_model == "llama-3-8b-instruct"

    def test_objective_priority(self, stack, request_factory):
        ds, director = stack
        ds.put_objective(InferenceObjective(name="batch", priority=-5,
                                            ttft_slo_ms=5000))
        req = request_factory()
        req.objective_name = "batch"
        d = director.handle_request(req)
        assert d.request.priority == -5
        assert d.request.ttft_slo_ms == 5000

    def test_sheddable_shed_under_saturation(self, stack, request_factory):
        ds, director = stack
        for ep in ds.endpoints():
            ep.update_metrics(Metrics(waiting_queue_size=100,
                                      kv_cache_usage=0.99))
        ds.put_objective(InferenceObjective(name="batch", priority=-1))
        req = request_factory()
        req.objective_name = "batch"
        with pytest.raises(AdmissionDenied) as ei:
            director.handle_request(req)
        assert ei.value.reason == "saturated"
        # critical request still admitted
        d = director.handle_request(request_factory())
        assert d.target is not None

    def test_subset_hint(self, stack, request_factory):
        ds, director = stack
        req = request_factory()
        req.headers["x-gateway-destination-endpoint-subset"] = "gpu2"
        d = director.handle_request(req)
        assert d.target.name == "gpu2"

    def test_no_endpoints_503(self, request_factory):
        cfg = load_config(CONFIG)
        ds = Datastore()
        director = Director(
            datastore=ds, scheduler=Scheduler(cfg.scheduler_config),
            admission=LegacyAdmissionController(
                UtilizationSaturationDetector()),
            candidates=EndpointCandidates(ds, cache_ttl_s=0),
            config=cfg.request_control)
        with pytest.raises(AdmissionDenied) as ei:
            director.handle_request(request_factory())
        assert ei.value.status == 503

    def test_response_complete_updates_inflight(self, stack, request_factory):
        ds, director = stack
        d = director.handle_request(request_factory())
        from llm_d_inference_scheduler_amd.datalayer.attributes import \
            IN_FLIGHT_LOAD
        load = d.target.get_attribute(IN_FLIGHT_LOAD)
        assert load.snapshot()[0] == 1
        director.handle_response_complete(
            d, Usage(prompt_tokens=40, completion_tokens=16, ttft_ms=30,
                     tpot_ms=9))
        assert load.snapshot()[0] == 0


class TestServedVerifier:
    def test_served_header_verification(self):
        from llm_d_inference_scheduler_amd.plugins.registry import \
            global_registry
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        v = global_registry.instantiate("destination-endpoint-served-verifier")
        ep = make_endpoint("gpu1", 1)
        v.response_received(None, ep, {v.SERVED_HEADER: "gpu1"})
        v.response_received(None, ep, {v.SERVED_HEADER: "gpu0"})
        v.response_received(None, ep, {})
        assert (v.checked, v.mismatches) == (2, 1)
