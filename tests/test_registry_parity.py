"""Registry parity against the reference's plugin inventory (SURVEY.md §2.5
/ PARITY.md): every plugin type the reference registers must exist here
under the same name, and every Scorer must produce sane [0,1] outputs on
synthetic endpoints."""
import pytest

from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
from llm_d_inference_scheduler_amd.handlers import parsers  # noqa: F401
from llm_d_inference_scheduler_amd.plugins import register_all_plugins
from llm_d_inference_scheduler_amd.plugins.registry import global_registry

register_all_plugins()
from llm_d_inference_scheduler_amd.scheduling.types import (LLMRequest,
                                                            SchedulingContext)

# the reference's registered plugin type names
# (pkg/epp/framework/plugins/register.go:23-56 + cmd/epp/runner/runner.go:463-515)
REFERENCE_PLUGINS = [
    # profile handlers + deciders
    "single-profile-handler", "disagg-profile-handler", "pd-profile-handler",
    "prefix-based-pd-decider", "always-disagg-pd-decider",
    "always-disagg-multimodal-decider", "data-parallel-profile-handler",
    # filters
    "label-selector-filter", "decode-filter", "prefill-filter",
    "encode-filter", "prefix-cache-affinity-filter",
    "slo-headroom-tier-filter", "header-based-testing-filter",
    # scorers
    "prefix-cache-scorer", "precise-prefix-cache-scorer",
    "kv-cache-utilization-scorer", "queue-scorer",
    "running-requests-size-scorer", "load-aware-scorer", "token-load-scorer",
    "active-request-scorer", "lora-affinity-scorer",
    "session-affinity-scorer", "no-hit-lru-scorer", "context-length-aware",
    "latency-scorer",
    # pickers
    "max-score-picker", "random-picker", "weighted-random-picker",
    # data producers
    "approx-prefix-cache-producer", "token-producer",
    "inflight-load-producer", "predicted-latency-producer",
    # admitters
    "latency-slo", "probabilistic-admitter",
    # parsers
    "openai-parser", "vllm-grpc-parser", "vertexai-parser",
    "passthrough-parser",
    # saturation detectors (flow control + filter form)
    "utilization-detector", "concurrency-detector",
    # conformance/test plugins registered in the production runner
    "destination-endpoint-served-verifier",
    # in-flight eviction policies (framework/plugins/flowcontrol/eviction)
    "priority-then-time-eviction-order-policy", "sheddable-eviction-filter",
]

SCORER_TYPES = [
    "prefix-cache-scorer", "precise-prefix-cache-scorer",
    "kv-cache-utilization-scorer", "queue-scorer",
    "running-requests-size-scorer", "load-aware-scorer", "token-load-scorer",
    "active-request-scorer", "lora-affinity-scorer",
    "session-affinity-scorer", "no-hit-lru-scorer", "context-length-aware",
    "latency-scorer",
]


def test_every_reference_plugin_type_registered():
    known = set(global_registry.known_types())
    missing = [p for p in REFERENCE_PLUGINS if p not in known]
    assert not missing, f"missing reference plugin types: {missing}"


@pytest.mark.parametrize("scorer_type", SCORER_TYPES)
def test_every_scorer_returns_unit_interval(scorer_type):
    sc = global_registry.instantiate(scorer_type)
    eps = []
    for i in range(3):
        ep = make_endpoint(f"gpu{i}", i)
        ep.update_metrics(Metrics(waiting_queue_size=i * 3,
                                  running_requests_size=i,
                                  kv_cache_usage=0.2 * i,
                                  cache_block_size=16,
                                  cache_num_blocks=1000))
        eps.append(ep)
    req = LLMRequest(request_id="r", model="m", prompt="",
                     prompt_tokens=list(range(64)), max_tokens=8,
                     session_id="sess-1")
    ctx = SchedulingContext(request=req)
    scores = sc.score(ctx, eps)
    assert set(scores) == {e.name for e in eps}
    for name, v in scores.items():
        assert 0.0 <= v <= 1.0, (scorer_type, name, v)


class TestMetricFamilyParity:
    """Pin the §5.5 metric families (reference pkg/epp/metrics/metrics.go
    + pkg/metrics docs/metrics.md) so a rename or accidental removal
    fails loudly — the golden-metrics-testdata analog."""

    def test_required_families_registered(self):
        from llm_d_inference_scheduler_amd.metrics import prom
        text = prom.render().decode()
        fams = {l.split()[2] for l in text.splitlines()
                if l.startswith("# TYPE")}
        required = {
            # request accounting (metrics.go:584-754)
            "inference_extension_request_total",
            "inference_extension_request_error_total",
            "inference_extension_request_duration_seconds",
            "inference_extension_request_sizes",
            "inference_extension_input_tokens",
            "inference_extension_output_tokens",
            "inference_extension_cached_tokens",
            "inference_extension_running_requests",
            # TTFT / TPOT (+normalized)
            "inference_extension_time_to_first_token_seconds",
            "inference_extension_normalized_time_per_output_token_seconds",
            # scheduler + per-plugin latency (metrics.go:786-821)
            "inference_extension_scheduler_e2e_duration_seconds",
            "inference_extension_plugin_duration_seconds",
            # prefix cache (metrics.go:826-841)
            "inference_extension_prefix_indexer_size",
            "inference_extension_prefix_indexer_hit_ratio",
            # flow control (metrics.go:848-897)
            "inference_extension_flow_control_queue_size",
            "inference_extension_flow_control_queue_duration_seconds",
            "inference_extension_flow_control_dispatch_total",
            "inference_extension_saturation",
            # model rewrite decisions
            "inference_extension_model_rewrite_total",
            # llm-d families (docs/metrics.md:13-27)
            "llm_d_inference_scheduler_disagg_decision_total",
            # MI355X-native additions
            "llm_d_inference_scheduler_xgmi_kv_transfer_bytes_total",
            "llm_d_inference_scheduler_xgmi_kv_transfer_seconds",
            "llm_d_inference_scheduler_datalayer_poll_errors_total",
            "llm_d_inference_scheduler_datalayer_extract_errors_total",
        }
        missing = required - fams
        assert not missing, f"metric families missing: {sorted(missing)}"
