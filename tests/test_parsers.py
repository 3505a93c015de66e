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
