"""Prometheus-text metrics extractor (reference extractor_test.go +
backend/metrics/metrics_test.go behaviors): vLLM family mapping, LoRA
latest-series-wins, label matchers, malformed-input leniency, HTTP source
fail-open, and the self-scrape loop through the front door's /metrics."""
import math

import pytest

from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
from llm_d_inference_scheduler_amd.datalayer.extractor import (
    ExtractorSpecs, HttpMetricsSource, MetricSpec, extract_metrics,
    parse_prom_text)

VLLM_SAMPLE = """\
# HELP vllm:num_requests_waiting Number of requests waiting.
# TYPE vllm:num_requests_waiting gauge
vllm:num_requests_waiting{model_name="m"} 7.0
# TYPE vllm:num_requests_running gauge
vllm:num_requests_running{model_name="m"} 12
# TYPE vllm:kv_cache_usage_perc gauge
vllm:kv_cache_usage_perc{model_name="m"} 0.4375
# TYPE vllm:lora_requests_info gauge
vllm:lora_requests_info{running_lora_adapters="a1,a2",waiting_lora_adapters="",max_lora="4"} 100.0
vllm:lora_requests_info{running_lora_adapters="a3",waiting_lora_adapters="a4",max_lora="4"} 200.0
# TYPE vllm:cache_config_info gauge
vllm:cache_config_info{block_size="16",num_gpu_blocks="81920"} 1
"""


class TestParse:
    def test_families_and_labels(self):
        fams = parse_prom_text(VLLM_SAMPLE)
        assert fams["vllm:num_requests_waiting"][0][0] == {"model_name": "m"}
        assert fams["vllm:num_requests_waiting"][0][1] == 7.0
        assert len(fams["vllm:lora_requests_info"]) == 2

    def test_malformed_lines_skipped(self):
        fams = parse_prom_text("garbage {{{\nvllm:x notanumber\n"
                               "ok_metric 3\n\x00binary\n")
        assert fams == {"ok_metric": [({}, 3.0, None)]}

    def test_inf_nan_and_escapes(self):
        fams = parse_prom_text(
            'a{l="x\\"y"} +Inf\nb 1e3\nc NaN\n')
        assert fams["a"][0][0]["l"] == 'x"y'
        assert math.isinf(fams["a"][0][1])
        assert fams["b"][0][1] == 1000.0
        assert math.isnan(fams["c"][0][1])

    def test_spec_matcher(self):
        fams = parse_prom_text('m{role="a"} 1\nm{role="b"} 2\n')
        spec = MetricSpec.parse('m{role=b}')
        assert spec.select(fams) == [({"role": "b"}, 2.0, None)]


class TestExtract:
    def test_full_mapping(self):
        m = extract_metrics(VLLM_SAMPLE)
        assert m.waiting_queue_size == 7
        assert m.running_requests_size == 12
        assert m.kv_cache_usage == pytest.approx(0.4375)
        assert m.cache_block_size == 16
        assert m.cache_num_blocks == 81920

    def test_lora_latest_series_wins(self):
        """vLLM emits lora_requests_info with the timestamp as the value;
        the freshest series is authoritative (metrics.go:242-270)."""
        m = extract_metrics(VLLM_SAMPLE)
        assert set(m.active_models) == {"a3"}
        assert set(m.waiting_models) == {"a4"}
        assert m.max_active_models == 4

    def test_missing_families_leave_defaults(self):
        m = extract_metrics("unrelated_metric 1\n")
        assert m.waiting_queue_size == 0 and m.kv_cache_usage == 0.0
        assert m.update_time > 0

    def test_custom_specs(self):
        text = "my_waiting 3\nmy_kv 0.5\n"
        m = extract_metrics(text, ExtractorSpecs(waiting="my_waiting",
                                                 kv_usage="my_kv"))
        assert m.waiting_queue_size == 3
        assert m.kv_cache_usage == 0.5


class TestHttpSource:
    def test_scrape_updates_endpoint(self):
        src = HttpMetricsSource(fetcher=lambda url: VLLM_SAMPLE)
        ep = make_endpoint("gpu0", 0)
        m = src.collect(ep)
        assert m.running_requests_size == 12

    def test_fetch_failure_fail_open(self):
        def boom(url):
            raise ConnectionError("down")
        src = HttpMetricsSource(fetcher=boom)
        assert src.collect(make_endpoint("gpu0", 0)) is None

    def test_url_from_label_or_address(self):
        src = HttpMetricsSource(fetcher=lambda u: "")
        ep = make_endpoint("gpu0", 0)
        assert src.url_for(ep).endswith("/metrics")
        ep.metadata.labels["metrics_url"] = "http://hostA:9090/metrics"
        assert src.url_for(ep) == "http://hostA:9090/metrics"
