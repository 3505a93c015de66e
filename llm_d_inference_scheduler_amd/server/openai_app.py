"""OpenAI-compatible HTTP front door (FastAPI).

Parity surface: the request paths the reference's parsers handle
(/v1/completions, /v1/chat/completions, /v1/embeddings, /v1/responses —
parsers/openai/openai.go) plus /metrics (Prometheus, vLLM-compatible
series) and /healthz (health.go:52 readiness semantics: pool synced).
Errors surface the canonical `x-request-dropped-reason` header
(pkg/common/error). SSE streaming re-emits per-token chunks like the
sidecar's chunked decode path (decode.go SSE re-emission).
"""
import asyncio
import json
import time
import uuid
from typing import Optional

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, StreamingResponse

from ..handlers.parsers import ParserMux, Usage
from ..metrics import prom
from ..models.tokenizer import HashTokenizer
from ..scheduling.types import LLMRequest
from .service import NodeService


def _vllm_compat(service: NodeService) -> bytes:
    """vLLM-compatible engine gauges appended to /metrics (the families
    the reference's datalayer extractor scrapes, options.go:121-125) —
    aggregated over this node's endpoints so a peer node's
    HttpMetricsSource (datalayer/extractor.py) can treat the whole node
    as one worker."""
    eps = list(service.node.datastore.endpoints())
    waiting = sum(ep.metrics.waiting_queue_size for ep in eps)
    running = sum(ep.metrics.running_requests_size for ep in eps)
    kv = max((ep.metrics.kv_cache_usage for ep in eps), default=0.0)
    blocks = [(ep.metrics.cache_block_size, ep.metrics.cache_num_blocks)
              for ep in eps if ep.metrics.cache_num_blocks]
    lines = [
        "# TYPE vllm:num_requests_waiting gauge",
        f"vllm:num_requests_waiting {waiting}",
        "# TYPE vllm:num_requests_running gauge",
        f"vllm:num_requests_running {running}",
        "# TYPE vllm:kv_cache_usage_perc gauge",
        f"vllm:kv_cache_usage_perc {kv}",
    ]
    if blocks:
        bs, nb = blocks[0]
        lines += ["# TYPE vllm:cache_config_info gauge",
                  f'vllm:cache_config_info{{block_size="{bs}",'
                  f'num_gpu_blocks="{nb}"}} 1']
    return ("\n".join(lines) + "\n").encode()

DROPPED_REASON_HEADER = "x-request-dropped-reason"


def build_app(service: NodeService,
              tokenizer: Optional[HashTokenizer] = None) -> FastAPI:
    app = FastAPI(title="llm-d-inference-scheduler-amd")
    mux = ParserMux()
    from ..handlers.parsers import VertexAIParser, VllmGrpcParser
    mux.register("application/grpc", VllmGrpcParser())
    mux.register("application/grpc+proto", VllmGrpcParser())
    mux.register("application/vnd.vertex-ai+json", VertexAIParser())
    tok = tokenizer or HashTokenizer()
    app.state.service = service

    def _error(status: int, reason: str, detail: str = "") -> JSONResponse:
        return JSONResponse(
            status_code=status,
            headers={DROPPED_REASON_HEADER: reason},
            content={"error": {"message": detail or reason,
                               "type": reason, "code": status}})

    async def _handle(request: Request, path: str):
        body = await request.body()
        headers = {k.lower(): v for k, v in request.headers.items()}
        result = mux.parse_request(body, headers, path)
        if result.error:
            return _error(400, "parse_error", result.error)
        req = result.request
        if result.skip or req is None:
            return _error(400, "unparseable", "passthrough not routable "
                          "without an upstream")
        prom.L(prom.request_sizes, req.model).observe(len(body))
        if req.prompt_tokens is None and not req.prompt and req.messages:
            pass  # token-producer will tokenize messages
        raw_stop = (result.request.raw_body or {}).get("stop") \
            if result.request.raw_body else None
        if raw_stop and req.stop_token_ids is None:
            stops = [raw_stop] if isinstance(raw_stop, str) else raw_stop
            ids = []
            for sword in stops[:4]:
                t = tok(str(sword))
                if t:
                    ids.append(t[0])   # single-token stop approximation
            if ids:
                req.stop_token_ids = ids
        t0 = time.time()
        handle = service.submit(req)
        if req.streaming and not req.is_embedding:
            return StreamingResponse(_sse_stream(req, handle, t0),
                                     media_type="text/event-stream")
        completion = await asyncio.to_thread(handle.wait, 120.0)
        if completion is None:
            service.cancel(req.request_id)     # reclaim engine resources
            return _error(504, "timeout")
        if completion.error:
            status = 429 if completion.error in (
                "saturated", "queue_capacity", "queue_timeout",
                "evicted") else 503
            return _error(status, completion.error)
        if req.is_embedding:
            return JSONResponse(_embedding_response(req, completion))
        return JSONResponse(_completion_response(req, completion, path))

    def _completion_response(req, completion, path):
        text = tok.decode(completion.tokens)
        usage = completion.usage.to_openai()
        created = int(time.time())
        if "chat" in path:
            return {"id": f"chatcmpl-{req.request_id}", "object":
                    "chat.completion", "created": created,
                    "model": req.model,
                    "choices": [{"index": 0, "message":
                                 {"role": "assistant", "content": text},
                                 "finish_reason": completion.finish_reason}],
                    "usage": usage}
        return {"id": f"cmpl-{req.request_id}", "object": "text_completion",
                "created": created, "model": req.model,
                "choices": [{"index": 0, "text": text,
                             "finish_reason": completion.finish_reason}],
                "usage": usage}

    def _embedding_response(req, completion):
        return {"object": "list", "model": req.model,
                "data": [{"object": "embedding", "index": 0,
                          "embedding": completion.tokens or []}],
                "usage": completion.usage.to_openai()}

    async def _sse_stream(req, handle, t0):
        idx = 0
        try:
            while True:
                item = await asyncio.to_thread(handle.token_queue.get)
                if item is None:
                    break
                chunk = {"id": f"cmpl-{req.request_id}", "object":
                         "text_completion.chunk", "model": req.model,
                         "choices": [{"index": 0,
                                      "text": tok.decode([item]),
                                      "finish_reason": None}]}
                idx += 1
                yield f"data: {json.dumps(chunk)}\n\n"
        except asyncio.CancelledError:
            # client disconnected mid-stream: stream-death cleanup
            # (server.go:246-253) — abort across the node
            service.cancel(req.request_id)
            raise
        completion = handle.completion
        if completion is not None and not completion.error:
            final = {"id": f"cmpl-{req.request_id}",
                     "object": "text_completion.chunk",
                     "model": req.model,
                     "choices": [{"index": 0, "text": "",
                                  "finish_reason":
                                      completion.finish_reason}],
                     "usage": completion.usage.to_openai()}
            yield f"data: {json.dumps(final)}\n\n"
        yield "data: [DONE]\n\n"

    @app.post("/v1/completions")
    async def completions(request: Request):
        return await _handle(request, "/v1/completions")

    @app.post("/vllm.VllmEngine/{method}")
    async def vllm_grpc(request: Request, method: str):
        """vLLM gRPC wire-format front door (parsers/vllmgrpc): bodies are
        gRPC-framed protobuf; dispatch runs through the same mux/parser
        plugins keyed by the :path the reference's ext-proc sees."""
        return await _handle(request, f"/vllm.VllmEngine/{method}")

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        return await _handle(request, "/v1/chat/completions")

    @app.post("/v1/responses")
    async def responses(request: Request):
        return await _handle(request, "/v1/responses")

    @app.post("/v1/conversations")
    async def conversations(request: Request):
        return await _handle(request, "/v1/conversations")

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        return await _handle(request, "/v1/embeddings")

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [
            {"id": service.node.cfg.model.name, "object": "model",
             "owned_by": "llm-d-inference-scheduler-amd"}]}

    @app.post("/tokenize")
    async def tokenize(request: Request):
        """vLLM-compatible tokenize route: lets this node serve as the
        remote-tokenizer worker for a peer's token producer (the surface
        the reference consumes via dataproducer/tokenizer/vllm_http.go)."""
        body = await request.json()
        toks = tok(str(body.get("prompt", "")))
        return {"tokens": toks, "count": len(toks),
                "max_model_len": service.node.engine.max_model_len}

    @app.post("/detokenize")
    async def detokenize(request: Request):
        body = await request.json()
        return {"prompt": tok.decode([int(t)
                                      for t in body.get("tokens", [])])}

    @app.get("/metrics")
    async def metrics():
        return Response(content=prom.render() + _vllm_compat(service),
                        media_type="text/plain; version=0.0.4")

    @app.post("/internal/v1/enqueue")
    async def internal_enqueue(request: Request):
        """Node-to-node execution API (node/remote.py): a peer router
        forwards an already-tokenized scheduled request here; this node
        runs it through its own engine pool and returns the completion.
        The pair (this route + the vLLM-compatible /metrics families) is
        what makes a front door usable as a remote endpoint."""
        try:
            body = json.loads(await request.body())
        except (ValueError, TypeError):
            return _error(400, "parse_error", "invalid JSON")
        if not isinstance(body, dict) or not body.get("request_id"):
            return _error(400, "parse_error", "request_id required")
        req = LLMRequest(
            request_id=f"fwd-{body['request_id']}",
            model=body.get("model", service.node.cfg.model.name),
            prompt=body.get("prompt", ""),
            prompt_tokens=body.get("prompt_tokens"),
            max_tokens=int(body.get("max_tokens", 16)),
            temperature=float(body.get("temperature") or 0.0),
            stop_token_ids=body.get("stop_token_ids"),
            streaming=bool(body.get("stream", False)),
            priority=int(body.get("priority", 0)))
        handle = service.submit(req)
        if req.streaming:
            # cross-node SSE token relay: the forwarding router re-emits
            # these chunks to ITS client as they arrive
            async def _relay():
                while True:
                    item = await asyncio.to_thread(handle.token_queue.get)
                    if item is None:
                        break
                    toks = item if isinstance(item, list) else [item]
                    yield ("data: " + json.dumps({"tokens": toks}) + "\n\n"
                           ).encode()
                comp = handle.completion
                u = comp.usage if comp else Usage()
                final = {"done": True,
                         "tokens": comp.tokens if comp else [],
                         "finish_reason": (comp.finish_reason
                                           if comp else "length"),
                         "error": comp.error if comp else "lost",
                         "usage": {"prompt_tokens": u.prompt_tokens,
                                   "completion_tokens": u.completion_tokens,
                                   "cached_tokens": u.cached_tokens,
                                   "ttft_ms": u.ttft_ms,
                                   "e2e_ms": u.e2e_ms}}
                yield ("data: " + json.dumps(final) + "\n\n").encode()
                yield b"data: [DONE]\n\n"
            return StreamingResponse(_relay(),
                                     media_type="text/event-stream")
        completion = await asyncio.to_thread(handle.wait, 300.0)
        if completion is None:
            service.cancel(req.request_id)
            return JSONResponse(status_code=504,
                                content={"error": "timeout"})
        u = completion.usage
        return {"tokens": completion.tokens,
                "finish_reason": completion.finish_reason,
                "error": completion.error,
                "usage": {"prompt_tokens": u.prompt_tokens,
                          "completion_tokens": u.completion_tokens,
                          "cached_tokens": u.cached_tokens,
                          "ttft_ms": u.ttft_ms, "e2e_ms": u.e2e_ms}}

    @app.post("/internal/v1/cancel")
    async def internal_cancel(request: Request):
        """Peer-side abort of a forwarded request (node/remote.py
        RemoteForwarder.cancel): unwinds the engine-resident sequence."""
        try:
            body = json.loads(await request.body())
            rid = body["request_id"]
        except (ValueError, TypeError, KeyError):
            return _error(400, "parse_error", "request_id required")
        service.cancel(f"fwd-{rid}")
        return {"canceled": rid}

    @app.get("/healthz")
    async def healthz():
        ready = service.node.datastore.pool_ready()
        return JSONResponse(status_code=200 if ready else 503,
                            content={"ready": ready})

    @app.get("/readyz")
    async def readyz():
        # readiness = pool synced + engine constructed (health.go:52)
        ready = service.node.datastore.pool_ready() and \
            service.node.engine is not None
        return JSONResponse(status_code=200 if ready else 503,
                            content={"ready": ready})

    @app.get("/debug/pprof/profile")
    async def pprof_profile(seconds: float = 1.0):
        """pprof analog (reference --enable-pprof, runner.go:318-324):
        cProfile the serving loop for N seconds, return pstats text."""
        import cProfile
        import io
        import pstats
        import time as _t
        prof = cProfile.Profile()
        prof.enable()
        end = _t.time() + min(seconds, 10.0)
        while _t.time() < end:
            await __import__("asyncio").sleep(0.05)
        prof.disable()
        buf = io.StringIO()
        pstats.Stats(prof, stream=buf).sort_stats("cumulative").print_stats(40)
        return Response(content=buf.getvalue(), media_type="text/plain")

    return app
