"""OpenAI front-door server over the in-process node (tiny CPU config)."""
import json

import pytest
import torch
from fastapi.testclient import TestClient

from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
from llm_d_inference_scheduler_amd.server import NodeService, build_app


@pytest.fixture(scope="module")
def client():
    cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                     device="cpu", dtype=torch.float32, kv_blocks=256)
    node = NodeRunner(cfg)
    service = NodeService(node)
    service.start()
    app = build_app(service)
    with TestClient(app) as c:
        yield c
    service.stop()


class TestOpenAIServer:
    def test_completions(self, client):
        r = client.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "hello world test prompt",
            "max_tokens": 4})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["object"] == "text_completion"
        assert body["usage"]["completion_tokens"] == 4
        assert body["choices"][0]["text"]

    def test_chat_completions(self, client):
        r = client.post("/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "hi there"}],
            "max_tokens": 3})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["choices"][0]["message"]["role"] == "assistant"
        assert body["usage"]["completion_tokens"] == 3

    def test_streaming_sse(self, client):
        with client.stream("POST", "/v1/completions", json={
                "model": "tiny-llama", "prompt": "stream me a response",
                "max_tokens": 4, "stream": True}) as r:
            assert r.status_code == 200
            events = []
            for line in r.iter_lines():
                if line.startswith("data:"):
                    events.append(line[5:].strip())
        assert events[-1] == "[DONE]"
        chunks = [json.loads(e) for e in events[:-1]]
        text_chunks = [c for c in chunks
                       if c["choices"][0]["text"]]
        assert len(text_chunks) == 4
        assert chunks[-1].get("usage", {}).get("completion_tokens") == 4

    def test_embeddings(self, client):
        r = client.post("/v1/embeddings", json={
            "model": "tiny-llama", "input": "embed this text"})
        assert r.status_code == 200, r.text
        assert r.json()["data"][0]["object"] == "embedding"

    def test_parse_error(self, client):
        r = client.post("/v1/completions", content=b"{broken",
                        headers={"content-type": "application/json"})
        assert r.status_code == 400
        assert r.headers["x-request-dropped-reason"] == "parse_error"

    def test_missing_model(self, client):
        r = client.post("/v1/completions", json={"prompt": "x"})
        assert r.status_code == 400

    def test_metrics_endpoint(self, client):
        r = client.get("/metrics")
        assert r.status_code == 200
        assert b"inference_extension_request_total" in r.content
        assert b"llm_d_inference_scheduler_disagg_decision_total" in r.content

    def test_healthz(self, client):
        r = client.get("/healthz")
        assert r.status_code == 200 and r.json()["ready"]

    def test_models(self, client):
        r = client.get("/v1/models")
        assert r.json()["data"][0]["id"] == "tiny-llama"


class TestHermeticExtras:
    def test_model_rewrite_round_trip(self, client):
        """Weighted InferenceModelRewrite: request under the alias routes to
        the real model; the response carries the CLIENT-facing name
        (rewriteModelName back, server.go:471)."""
        from llm_d_inference_scheduler_amd.api.modelrewrite import (
            InferenceModelRewrite, RewriteRule, RewriteTarget)
        node = client.app.state.service.node
        node.datastore.put_model_rewrite(InferenceModelRewrite(
            name="alias", rules=[RewriteRule(
                model="tiny-alias",
                targets=[RewriteTarget("tiny-llama", weight=1)])]))
        r = client.post("/v1/completions", json={
            "model": "tiny-alias", "prompt": "rewrite me please now",
            "max_tokens": 3})
        assert r.status_code == 200, r.text
        assert r.json()["model"] == "tiny-alias"

    def test_vllm_grpc_content_type(self, client):
        """vLLM gRPC wire bodies route through the mux by content-type."""
        def ev(v):
            out = b""
            while True:
                b7 = v & 0x7F
                v >>= 7
                if v:
                    out += bytes([b7 | 0x80])
                else:
                    return out + bytes([b7])
        msg = bytes([0x0A]) + ev(10) + b"tiny-llama"
        packed = b"".join(ev(t) for t in [9, 8, 7, 6, 5, 4, 3, 2, 1])
        msg += bytes([0x1A]) + ev(len(packed)) + packed
        msg += bytes([0x20]) + ev(3)
        body = bytes([0]) + len(msg).to_bytes(4, "big") + msg
        r = client.post("/vllm.VllmEngine/Generate", content=body,
                        headers={"content-type": "application/grpc"})
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 3


class TestFlowControl429:
    def test_saturated_returns_429_with_reason(self):
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=256,
                         flow_control=True, fc_global_max_items=2)
        node = NodeRunner(cfg)
        service = NodeService(node)
        service.start()
        app = build_app(service)
        # force saturation: the flow queue fills to fc_global_max_items
        # and further arrivals are rejected with 429 + reason header
        node.detector.is_saturated = lambda eps: True
        with TestClient(app) as c:
            import concurrent.futures as cf

            def post(i):
                return c.post("/v1/completions", json={
                    "model": "tiny-llama",
                    "prompt": "x " * 40 + str(i), "max_tokens": 4})
            with cf.ThreadPoolExecutor(max_workers=8) as pool:
                futs = [pool.submit(post, i) for i in range(8)]
                import time as _t
                _t.sleep(1.0)
                node.detector.is_saturated = lambda eps: False  # drain
                rs = [f.result(timeout=60) for f in futs]
            codes = sorted(r.status_code for r in rs)
            rejected = [r for r in rs if r.status_code == 429]
            assert rejected, codes
            assert rejected[0].headers.get("x-request-dropped-reason")
            assert 200 in codes, codes
        service.stop()


class TestCancellation:
    def test_cancel_unwinds_engine(self):
        """Stream-death cleanup: canceling a routed request frees engine
        state everywhere (server.go:246-253 forced-complete analog)."""
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=256)
        node = NodeRunner(cfg)
        service = NodeService(node)
        from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
        req = LLMRequest(request_id="long", model="tiny-llama", prompt="",
                         prompt_tokens=list(range(5, 69)), max_tokens=5000)
        h = service.submit(req)
        for _ in range(6):
            service.step_once()
        assert node.engine.has_work
        service.cancel("long")
        for _ in range(4):
            service.step_once()
        done = h.wait(1.0)
        assert done is not None and done.error == "canceled"
        assert not node.engine.has_work
        assert node.inflight == 0
        assert node.engine.mgr.usage == pytest.approx(0.0)
        node.shutdown()


class TestMetricsGolden:
    def test_metrics_golden_families(self, client):
        """Golden family inventory (reference pkg/epp/metrics/testdata/):
        every documented series renders at /metrics. docs/metrics.md is the
        human-readable copy of this list."""
        # exercise a request so per-model series exist
        client.post("/v1/completions",
                    json={"model": "tiny-llama", "prompt": "golden metrics",
                          "max_tokens": 2})
        body = client.get("/metrics").content.decode()
        ns = "inference_extension"
        llmd = "llm_d_inference_scheduler"
        families = [
            f"{ns}_request_total",
            f"{ns}_request_error_total",
            f"{ns}_request_duration_seconds",
            f"{ns}_request_sizes",
            f"{ns}_input_tokens",
            f"{ns}_output_tokens",
            f"{ns}_cached_tokens",
            f"{ns}_time_to_first_token_seconds",
            f"{ns}_normalized_time_per_output_token_seconds",
            f"{ns}_scheduler_e2e_duration_seconds",
            f"{ns}_plugin_duration_seconds",
            f"{ns}_running_requests",
            f"{ns}_prefix_indexer_size",
            f"{ns}_prefix_indexer_hit_ratio",
            f"{ns}_flow_control_queue_size",
            f"{ns}_flow_control_queue_duration_seconds",
            f"{ns}_flow_control_dispatch_total",
            f"{ns}_saturation",
            f"{ns}_model_rewrite_total",
            f"{llmd}_disagg_decision_total",
            f"{llmd}_datalayer_poll_errors_total",
            f"{llmd}_datalayer_extract_errors_total",
            f"{llmd}_xgmi_kv_transfer_bytes_total",
            f"{llmd}_xgmi_kv_transfer_seconds",
        ]
        missing = [f for f in families if f"# TYPE {f}" not in body]
        assert not missing, f"families absent from /metrics: {missing}"
        # the exercised request populated the labelled counter
        assert 'request_total{model="tiny-llama"' in body


class TestVllmCompatScrape:
    def test_extractor_round_trips_front_door(self, client):
        """Self-scrape loop: the families our /metrics exposes are the
        ones HttpMetricsSource consumes, so node A can treat node B's
        front door as a vLLM-compatible worker (options.go:121-125)."""
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            make_endpoint
        from llm_d_inference_scheduler_amd.datalayer.extractor import \
            HttpMetricsSource
        body = client.get("/metrics").content.decode()
        assert "vllm:num_requests_waiting" in body
        assert "vllm:kv_cache_usage_perc" in body
        src = HttpMetricsSource(
            fetcher=lambda url: client.get("/metrics").content.decode())
        m = src.collect(make_endpoint("peer-node", 0))
        assert m is not None
        assert m.waiting_queue_size >= 0 and 0.0 <= m.kv_cache_usage <= 1.0
        assert m.cache_num_blocks > 0      # cache_config_info present


class TestTokenizerSurface:
    """Remote-tokenizer surface (reference dataproducer/tokenizer/
    {vllm_http,uds}.go): /tokenize + /detokenize routes and the
    HttpTokenizer client with fail-open fallback."""

    def _client(self):
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=64)
        service = NodeService(NodeRunner(cfg))
        service.start()
        return TestClient(build_app(service)), service

    def test_tokenize_route_roundtrip(self):
        c, service = self._client()
        with c:
            r = c.post("/tokenize", json={"model": "tiny-llama",
                                          "prompt": "hello world, again"})
            assert r.status_code == 200
            body = r.json()
            assert body["count"] == len(body["tokens"]) > 0
            r2 = c.post("/detokenize", json={"tokens": body["tokens"]})
            assert r2.status_code == 200 and r2.json()["prompt"]
        service.stop()

    def test_http_tokenizer_against_front_door(self):
        from llm_d_inference_scheduler_amd.models.tokenizer import (
            HashTokenizer, HttpTokenizer)
        c, service = self._client()
        with c:
            tok = HttpTokenizer(model="tiny-llama", client=c,
                                fallback=HashTokenizer(1000))
            ids = tok("the quick brown fox")
            # the server used its own HashTokenizer: results must agree
            assert ids == HashTokenizer()("the quick brown fox")
            assert tok.errors == 0
        service.stop()

    def test_http_tokenizer_fail_open(self):
        from llm_d_inference_scheduler_amd.models.tokenizer import (
            HashTokenizer, HttpTokenizer)

        class DeadClient:
            def post(self, *a, **k):
                raise ConnectionError("worker down")
        tok = HttpTokenizer(model="m", client=DeadClient(),
                            fallback=HashTokenizer(5000))
        ids = tok("still routes")
        assert ids == HashTokenizer(5000)("still routes")
        assert tok.errors == 1

    def test_token_producer_http_mode(self):
        from llm_d_inference_scheduler_amd.plugins.producers import (
            TokenProducer)
        from llm_d_inference_scheduler_amd.models.tokenizer import (
            HttpTokenizer)
        p = TokenProducer("token-producer", mode="http",
                          url="http://127.0.0.1:1", timeoutMs=50)
        assert isinstance(p.tokenizer, HttpTokenizer)
        # unreachable worker -> fail-open to the hash fallback
        from llm_d_inference_scheduler_amd.scheduling.types import (
            LLMRequest, SchedulingContext)
        req = LLMRequest(request_id="t", model="m", prompt="abc def")
        ctx = SchedulingContext(request=req)
        p.produce(ctx, [])
        assert req.prompt_tokens and p.tokenizer.errors == 1


class TestConversationsRoute:
    def test_conversations_path(self, client):
        r = client.post("/v1/conversations", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "hi there friend"}],
            "max_tokens": 3})
        assert r.status_code == 200


class TestServeCLI:
    """The `python -m ...server` entry (cmd/epp/main.go analog)."""

    def test_build_node_from_flags(self):
        from llm_d_inference_scheduler_amd.server.__main__ import (
            build_node, parse_args)
        args = parse_args(["--model", "tiny-llama", "--device", "cpu",
                           "--flow-control", "--kv-dtype", "bf16"])
        node = build_node(args)
        assert node.is_router and node.flow is not None
        node.shutdown()

    def test_config_text_flag(self):
        from llm_d_inference_scheduler_amd.server.__main__ import (
            build_node, parse_args)
        yaml_text = """
plugins:
  - type: queue-scorer
  - type: max-score-picker
schedulingProfiles:
  - name: default
    plugins:
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: max-score-picker}
"""
        args = parse_args(["--model", "tiny-llama", "--device", "cpu",
                           "--config-text", yaml_text])
        node = build_node(args)
        assert "default" in node.loaded.scheduler_config.profiles
        node.shutdown()
