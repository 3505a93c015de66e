"""L2 parsers: openai (all routes), passthrough skip, vertexai, mux, usage."""
import json

from llm_d_inference_scheduler_amd.handlers.parsers import (
    OpenAIParser, ParserMux, PassthroughParser, VertexAIParser)


class TestOpenAIParser:
    def test_completions(self):
        p = OpenAIParser()
        r = p.parse_request(json.dumps({
            "model": "llama-3-8b", "prompt": "hello", "max_tokens": 32,
            "temperature": 0.5, "stream": True}).encode(),
            {"x-request-id": "abc"}, "/v1/completions")
        assert r.request is not None and not r.skip
        req = r.request
        assert req.request_id == "abc" and req.model == "llama-3-8b"
        assert req.prompt == "hello" and req.max_tokens == 32
        assert req.temperature == 0.5 and req.streaming

    def test_chat_completions_with_image(self):
        p = OpenAIParser()
        body = {"model": "llava", "messages": [
            {"role": "user", "content": [
                {"type": "text", "text": "what is this"},
                {"type": "image_url", "image_url": {"url": "http://x/img.png"}},
            ]}]}
        r = p.parse_request(json.dumps(body).encode(), {}, "/v1/chat/completions")
        assert len(r.request.mm_items) == 1
        assert r.request.mm_items[0].kind == "image_url"
        assert r.request.mm_items[0].url == "http://x/img.png"

    def test_headers_extracted(self):
        p = OpenAIParser()
        r = p.parse_request(json.dumps({"model": "m", "prompt": "x"}).encode(),
                            {"x-gateway-inference-fairness-id": "tenant-a",
                             "x-gateway-inference-objective": "critical"}, "")
        assert r.request.fairness_id == "tenant-a"
        assert r.request.objective_name == "critical"

    def test_invalid_json(self):
        assert OpenAIParser().parse_request(b"{oops", {}, "").error

    def test_missing_model(self):
        assert OpenAIParser().parse_request(b'{"prompt": "x"}', {}, "").error

    def test_usage_json(self):
        u = OpenAIParser().parse_response_usage(json.dumps({
            "usage": {"prompt_tokens": 10, "completion_tokens": 5,
                      "prompt_tokens_details": {"cached_tokens": 4}}}).encode(),
            streaming=False)
        assert (u.prompt_tokens, u.completion_tokens, u.cached_tokens) == (10, 5, 4)

    def test_usage_sse(self):
        sse = (b'data: {"choices": []}\n\n'
               b'data: {"usage": {"prompt_tokens": 7, "completion_tokens": 3}}\n\n'
               b'data: [DONE]\n')
        u = OpenAIParser().parse_response_usage(sse, streaming=True)
        assert (u.prompt_tokens, u.completion_tokens) == (7, 3)


class TestOtherParsers:
    def test_passthrough_skips(self):
        r = PassthroughParser().parse_request(b"\x00\x01raw", {}, "")
        assert r.skip and r.request is None

    def test_vertexai(self):
        body = {"model": "gemini-x", "contents": [
            {"parts": [{"text": "hello"}, {"text": "world"}]}],
            "generationConfig": {"maxOutputTokens": 64, "temperature": 0.2}}
        r = VertexAIParser().parse_request(json.dumps(body).encode(), {}, "")
        assert r.request.model == "gemini-x"
        assert r.request.prompt == "hello\nworld"
        assert r.request.max_tokens == 64

    def test_mux_dispatch(self):
        mux = ParserMux()
        mux.register("application/grpc", PassthroughParser())
        r1 = mux.parse_request(json.dumps({"model": "m", "prompt": "x"}).encode(),
                               {"content-type": "application/json"}, "")
        assert r1.request is not None
        r2 = mux.parse_request(b"anything",
                               {"content-type": "application/grpc"}, "")
        assert r2.skip


class TestVllmGrpcParser:
    @staticmethod
    def _encode_varint(v):
        out = b""
        while True:
            b7 = v & 0x7F
            v >>= 7
            if v:
                out += bytes([b7 | 0x80])
            else:
                return out + bytes([b7])

    def _encode_generate(self, model, tokens, max_tokens, temp, stream,
                         rid=""):
        import struct
        ev = self._encode_varint
        msg = b""
        msg += bytes([0x0A]) + ev(len(model)) + model.encode()
        packed = b"".join(ev(t) for t in tokens)
        msg += bytes([0x1A]) + ev(len(packed)) + packed
        msg += bytes([0x20]) + ev(max_tokens)
        msg += bytes([0x2D]) + struct.pack("<f", temp)
        msg += bytes([0x30]) + ev(1 if stream else 0)
        if rid:
            msg += bytes([0x3A]) + ev(len(rid)) + rid.encode()
        return bytes([0]) + len(msg).to_bytes(4, "big") + msg

    def test_generate_roundtrip(self):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VllmGrpcParser
        p = VllmGrpcParser()
        body = self._encode_generate("llama-3-8b", [5, 300, 70000], 33,
                                     0.5, True, rid="abc")
        res = p.parse_request(body, {}, "/vllm.VllmEngine/Generate")
        assert res.error is None and not res.skip
        r = res.request
        assert r.model == "llama-3-8b"
        assert r.prompt_tokens == [5, 300, 70000]
        assert r.max_tokens == 33
        assert abs(r.temperature - 0.5) < 1e-6
        assert r.streaming and r.request_id == "abc"

    def test_embed_and_skip_methods(self):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VllmGrpcParser
        p = VllmGrpcParser()
        body = self._encode_generate("m", [1, 2], 1, 0.0, False)
        res = p.parse_request(body, {}, "/vllm.VllmEngine/Embed")
        assert res.request.is_embedding
        for m in ("HealthCheck", "Abort", "GetModelInfo", "GetServerInfo"):
            assert p.parse_request(b"", {}, f"/vllm.VllmEngine/{m}").skip

    def test_response_usage(self):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VllmGrpcParser
        ev = self._encode_varint
        msg = bytes([0x08]) + ev(100) + bytes([0x10]) + ev(32) + \
            bytes([0x18]) + ev(48)
        body = bytes([0]) + len(msg).to_bytes(4, "big") + msg
        u = VllmGrpcParser().parse_response_usage(body, False)
        assert (u.prompt_tokens, u.completion_tokens, u.cached_tokens) == \
            (100, 32, 48)

    def test_malformed(self):
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            VllmGrpcParser
        res = VllmGrpcParser().parse_request(
            b"\x00\x00\x00\x00\x03\xff\xff\xff", {},
            "/vllm.VllmEngine/Generate")
        assert res.error is not None


class TestModelRewriteHeader:
    def test_header_forces_target_model(self):
        from llm_d_inference_scheduler_amd.handlers.parsers import (
            MODEL_REWRITE_HEADER, OpenAIParser)
        body = json.dumps({"model": "m", "prompt": "x"}).encode()
        r = OpenAIParser("p").parse_request(
            body, {MODEL_REWRITE_HEADER: "m-forced"}, "/v1/completions")
        assert r.request.target_model == "m-forced"
        assert r.request.model == "m"

    def test_header_wins_over_crd_rewrite(self):
        import torch
        from llm_d_inference_scheduler_amd.api.modelrewrite import (
            InferenceModelRewrite, RewriteRule, RewriteTarget)
        from llm_d_inference_scheduler_amd.handlers.parsers import \
            MODEL_REWRITE_HEADER
        from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
        from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
        from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
        node = NodeRunner(NodeConfig(model=TINY_LLAMA, device="cpu",
                                     dtype=torch.float32, kv_blocks=64))
        node.datastore.put_model_rewrite(InferenceModelRewrite(
            name="rw", rules=[RewriteRule(
                model="tiny-llama",
                targets=[RewriteTarget("crd-target", weight=1)])]))
        req = LLMRequest(request_id="h1", model="tiny-llama", prompt="x",
                         headers={MODEL_REWRITE_HEADER: "hdr-target"},
                         target_model="hdr-target")
        d = node.director.handle_request(req)
        assert d.request.target_model == "hdr-target"
        node.director.handle_response_complete(d, None)
        # without the header, the CRD rule applies
        req2 = LLMRequest(request_id="h2", model="tiny-llama", prompt="x")
        d2 = node.director.handle_request(req2)
        assert d2.request.target_model == "crd-target"
        node.shutdown()
