"""Prometheus metrics (parity: pkg/epp/metrics/metrics.go:567-902 +
pkg/metrics/metrics.go llm_d_* series, docs/metrics.md).

Key families replicated: request counts/sizes/latencies,
TTFT/TPOT, scheduler e2e + per-plugin latency, prefix-cache size & match
ratio, flow-control queue/dispatch/saturation, disagg decisions, rewrites.
"""
from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest)

registry = CollectorRegistry()

NS = "inference_extension"
LLMD = "llm_d_inference_scheduler"

request_total = Counter(
    f"{NS}_request_total", "Routed requests", ["model", "target_model"],
    registry=registry)
request_error_total = Counter(
    f"{NS}_request_error_total", "Request errors", ["model", "reason"],
    registry=registry)
request_duration = Histogram(
    f"{NS}_request_duration_seconds", "End-to-end request latency", ["model"],
    registry=registry,
    buckets=(.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5, 10, 30, 60))
request_sizes = Histogram(
    f"{NS}_request_sizes", "Request body size bytes", ["model"],
    registry=registry, buckets=(64, 256, 1024, 4096, 16384, 65536, 262144))
input_tokens = Histogram(
    f"{NS}_input_tokens", "Prompt tokens", ["model"], registry=registry,
    buckets=(8, 16, 32, 64, 128, 256, 512, 1024, 2048, 4096, 8192, 16384))
output_tokens = Histogram(
    f"{NS}_output_tokens", "Generated tokens", ["model"], registry=registry,
    buckets=(8, 16, 32, 64, 128, 256, 512, 1024, 2048, 4096))
cached_tokens = Histogram(
    f"{NS}_cached_tokens", "Prefix-cached prompt tokens", ["model"],
    registry=registry,
    buckets=(0, 8, 16, 32, 64, 128, 256, 512, 1024, 2048, 4096))
ttft = Histogram(
    f"{NS}_time_to_first_token_seconds", "TTFT", ["model"], registry=registry,
    buckets=(.001, .005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5, 10))
tpot = Histogram(
    f"{NS}_normalized_time_per_output_token_seconds", "TPOT", ["model"],
    registry=registry,
    buckets=(.0005, .001, .0025, .005, .01, .025, .05, .1, .25))
scheduler_e2e = Histogram(
    f"{NS}_scheduler_e2e_duration_seconds", "Scheduler e2e latency",
    registry=registry,
    buckets=(.00005, .0001, .00025, .0005, .001, .0025, .005, .01, .025, .05))
plugin_latency = Histogram(
    f"{NS}_plugin_duration_seconds", "Per-plugin latency", ["plugin"],
    registry=registry,
    buckets=(.00001, .00005, .0001, .00025, .0005, .001, .0025, .005, .01))
running_requests = Gauge(
    f"{NS}_running_requests", "EPP-tracked in-flight requests", ["model"],
    registry=registry)

prefix_index_size = Gauge(
    f"{NS}_prefix_indexer_size", "Prefix index hash entries", registry=registry)
prefix_match_ratio = Histogram(
    f"{NS}_prefix_indexer_hit_ratio", "Prefix match ratio per request",
    registry=registry, buckets=(0, .1, .25, .5, .75, .9, 1))

flow_queue_size = Gauge(
    f"{NS}_flow_control_queue_size", "Queued requests", ["priority"],
    registry=registry)
flow_queue_duration = Histogram(
    f"{NS}_flow_control_queue_duration_seconds", "Time in flow-control queue",
    registry=registry,
    buckets=(.0001, .001, .005, .01, .05, .1, .5, 1, 5, 30))
flow_dispatch_total = Counter(
    f"{NS}_flow_control_dispatch_total", "Flow-control outcomes", ["outcome"],
    registry=registry)
saturation_gauge = Gauge(
    f"{NS}_saturation", "Pool saturation [0,1+]", registry=registry)

disagg_decision_total = Counter(
    f"{LLMD}_disagg_decision_total", "Disaggregation decisions",
    ["decision_type"], registry=registry)
rewrite_decision_total = Counter(
    f"{NS}_model_rewrite_total", "Model rewrite decisions",
    ["model", "target_model"], registry=registry)

datalayer_poll_errors = Counter(
    f"{LLMD}_datalayer_poll_errors_total",
    "Data-source poll errors per source type", ["source_type"],
    registry=registry)
datalayer_extract_errors = Counter(
    f"{LLMD}_datalayer_extract_errors_total",
    "Extract errors per source/extractor type",
    ["source_type", "extractor_type"], registry=registry)

xgmi_kv_transfer_bytes = Counter(
    f"{LLMD}_xgmi_kv_transfer_bytes_total",
    "KV-cache bytes moved over xGMI", ["direction"], registry=registry)
xgmi_kv_transfer_seconds = Histogram(
    f"{LLMD}_xgmi_kv_transfer_seconds", "per-transfer xGMI latency",
    registry=registry,
    buckets=(.0001, .00025, .0005, .001, .0025, .005, .01, .025, .05, .1))


_label_cache = {}


def L(metric, *vals):
    """Memoized metric.labels(*vals): prometheus_client's labels() costs
    ~3 us (lock + validation) per call — measurable on the per-request
    routing path (profiles/router_tax.json). Use for hot-path metrics."""
    key = (id(metric),) + vals
    child = _label_cache.get(key)
    if child is None:
        child = _label_cache[key] = metric.labels(*vals)
    return child


def render() -> bytes:
    return generate_latest(registry)
