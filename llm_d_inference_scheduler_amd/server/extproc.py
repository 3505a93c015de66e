"""Envoy ext-proc FULL_DUPLEX_STREAMED endpoint-picker server (L1).

Wire-compatible second front door beside the OpenAI HTTP app: speaks
envoy.service.ext_proc.v3.ExternalProcessor/Process over gRPC with the
reference's stream semantics (/root/reference/pkg/epp/handlers/server.go):

* per-stream state machine RequestHeaders -> RequestBody(chunks, buffered
  until end_of_stream) -> route -> ResponseHeaders -> ResponseBody(chunks)
  -> Trailers (server.go:168-445);
* ordered sends: the request_headers response (destination-endpoint header
  mutation + `envoy.lb` dynamic metadata) is ALWAYS emitted before any
  request_body response; body mutations are chunked at 62 KB under Envoy's
  64 KB message cap (server.go:489-598, common/envoy/chunking.go:24-29);
* parse-skip and bodyless-GET fall back to a random ready endpoint
  (server.go:335-342, request.go:37-66);
* admission denial / mid-stream eviction surface as an ImmediateResponse
  with the HTTP status and `x-request-dropped-reason` header
  (server.go:262-284,493-517);
* model-name rewrite-back in response bodies, JSON and SSE
  (server.go:471 rewriteModelName);
* stream death forces response-complete hooks so accounting never leaks
  (server.go:246-253).
"""
import json
import random
import threading
import time
import uuid
from concurrent import futures
from typing import Dict, Iterator, Optional

import grpc

from ..handlers.parsers import ParserMux, Usage
from ..metrics import prom
from ..requestcontrol.admission import AdmissionDenied
from ..utils.logging import get_logger
from . import extproc_pb as pb

log = get_logger("server.extproc")

SERVICE = "envoy.service.ext_proc.v3.ExternalProcessor"
METHOD = "Process"
# Envoy rejects ext-proc messages over 64 KB; mirror the reference's 62 KB
# per-chunk budget (chunking.go:24-29)
BODY_BYTE_LIMIT = 62 * 1024

DEST_ENDPOINT_HEADER = "x-gateway-destination-endpoint"
DEST_METADATA_NAMESPACE = "envoy.lb"
DROPPED_REASON_HEADER = "x-request-dropped-reason"
FAIRNESS_HEADER = "x-gateway-inference-fairness-id"
OBJECTIVE_HEADER = "x-gateway-inference-objective"
REWRITE_HEADER = "x-gateway-model-name-rewrite"


def _chunk_body(body: bytes):
    if not body:
        yield b""
        return
    for i in range(0, len(body), BODY_BYTE_LIMIT):
        yield body[i:i + BODY_BYTE_LIMIT]


def _rewrite_model_name(body: bytes, target: str, client: str) -> bytes:
    """Rewrite the served model name back to the client-facing one
    (server.go:471): JSON bodies and SSE `data:` lines."""
    if not target or target == client:
        return body
    try:
        text = body.decode()
    except UnicodeDecodeError:
        return body
    if text.lstrip().startswith("{"):
        try:
            obj = json.loads(text)
            if obj.get("model") == target:
                obj["model"] = client
                return json.dumps(obj).encode()
        except ValueError:
            return body
        return body
    if "data:" in text:
        out_lines = []
        for line in text.split("\n"):
            if line.startswith("data:") and line[5:].strip() not in (
                    "", "[DONE]"):
                try:
                    obj = json.loads(line[5:])
                    if obj.get("model") == target:
                        obj["model"] = client
                        line = "data: " + json.dumps(obj)
                except ValueError:
                    pass
            out_lines.append(line)
        return "\n".join(out_lines).encode()
    return body


class _StreamState:
    def __init__(self):
        self.headers: Dict[str, str] = {}
        self.body = bytearray()
        self.decision = None
        self.request = None
        self.response_body = bytearray()
        self.streaming = False
        self.completed = False
        self.evicted: Optional[str] = None   # reason, set by evict()
        self.request_id = ""


class ExtProcServer:
    """gRPC server hosting Process(); `node` supplies director/datastore."""

    def __init__(self, node, max_workers: int = 16):
        self.node = node
        self.mux = ParserMux()
        self._grpc: Optional[grpc.Server] = None
        self.port: Optional[int] = None
        self._streams: Dict[str, _StreamState] = {}
        self._lock = threading.Lock()
        self._max_workers = max_workers

    # ---- lifecycle ----
    def start(self, port: int = 0) -> int:
        handler = grpc.method_handlers_generic_handler(SERVICE, {
            METHOD: grpc.stream_stream_rpc_method_handler(
                self._process,
                request_deserializer=pb.ProcessingRequest.FromString,
                response_serializer=lambda m: m.SerializeToString()),
        })
        self._grpc = grpc.server(
            futures.ThreadPoolExecutor(max_workers=self._max_workers))
        self._grpc.add_generic_rpc_handlers((handler,))
        self.port = self._grpc.add_insecure_port(f"127.0.0.1:{port}")
        self._grpc.start()
        # register with the node's in-flight evictor path so a saturation-
        # driven eviction reaches open ext-proc streams as a 429
        setattr(self.node, "extproc", self)
        log.info("ext-proc server listening", port=self.port)
        return self.port

    def stop(self) -> None:
        if self._grpc is not None:
            self._grpc.stop(grace=0.5)
            self._grpc = None

    # ---- mid-stream eviction hook (flowcontrol/eviction -> 429) ----
    def evict(self, request_id: str, reason: str = "evicted") -> bool:
        with self._lock:
            st = self._streams.get(request_id)
            if st is None or st.completed:
                return False
            st.evicted = reason
            return True

    # ---- helpers ----
    def _random_endpoint(self):
        eps = [ep for ep in self.node.datastore.endpoints()]
        return random.choice(eps) if eps else None

    def _headers_response(self, address: str, extra=None):
        resp = pb.ProcessingResponse()
        cr = resp.request_headers.response
        cr.status = pb.CONTINUE
        if address:
            pb.set_header(cr.header_mutation, DEST_ENDPOINT_HEADER, address)
            ns = resp.dynamic_metadata.fields[
                DEST_METADATA_NAMESPACE].struct_value
            ns.fields[DEST_ENDPOINT_HEADER].string_value = address
        for k, v in (extra or {}).items():
            pb.set_header(cr.header_mutation, k, v)
        return resp

    def _body_responses(self, body: Optional[bytes]):
        """Mutated-body chunks (<=62 KB each), or a plain CONTINUE pass-
        through when no mutation is needed."""
        if body is None:
            resp = pb.ProcessingResponse()
            resp.request_body.response.status = pb.CONTINUE
            yield resp
            return
        for chunk in _chunk_body(body):
            resp = pb.ProcessingResponse()
            cr = resp.request_body.response
            cr.status = pb.CONTINUE_AND_REPLACE
            cr.body_mutation.body = chunk
            yield resp

    def _immediate(self, status: int, reason: str, detail: str = ""):
        resp = pb.ProcessingResponse()
        imm = resp.immediate_response
        imm.status.code = status
        pb.set_header(imm.headers, DROPPED_REASON_HEADER, reason)
        imm.details = detail or reason
        body = json.dumps({"error": {"message": detail or reason,
                                     "type": reason, "code": status}})
        imm.body = body.encode()
        return resp

    # ---- the stream state machine ----
    def _process(self, request_iterator: Iterator, context
                 ) -> Iterator:
        st = _StreamState()
        try:
            for msg in request_iterator:
                if st.evicted is not None:
                    # flow-control eviction mid-stream (server.go:262-284)
                    yield self._immediate(429, st.evicted)
                    self._finish(st, error=st.evicted)
                    return
                which = msg.WhichOneof("request")
                if which == "request_headers":
                    hdrs = msg.request_headers
                    st.headers = pb.headers_to_dict(hdrs.headers)
                    st.request_id = st.headers.get(
                        "x-request-id", f"extproc-{uuid.uuid4().hex[:12]}")
                    if hdrs.end_of_stream:
                        # bodyless request (GET): random ready endpoint
                        # (request.go:55-66)
                        ep = self._random_endpoint()
                        yield self._headers_response(
                            ep.metadata.address if ep else "")
                elif which == "request_body":
                    st.body.extend(msg.request_body.body)
                    if msg.request_body.end_of_stream:
                        for resp in self._route(st):
                            yield resp
                        if st.completed:
                            # ImmediateResponse ends the exchange
                            return
                elif which == "response_headers":
                    if st.decision is not None:
                        self.node.director.handle_response_headers(
                            st.decision, {})
                    resp = pb.ProcessingResponse()
                    resp.response_headers.response.status = pb.CONTINUE
                    yield resp
                elif which == "response_body":
                    for resp in self._response_body(
                            st, msg.response_body.body,
                            msg.response_body.end_of_stream):
                        yield resp
                elif which == "request_trailers":
                    resp = pb.ProcessingResponse()
                    resp.request_trailers.SetInParent()
                    yield resp
                elif which == "response_trailers":
                    resp = pb.ProcessingResponse()
                    resp.response_trailers.SetInParent()
                    yield resp
        finally:
            # stream death / normal end: forced completion (server.go:246)
            self._finish(st, error="stream_closed"
                         if not st.completed else "")
            with self._lock:
                self._streams.pop(st.request_id, None)

    def _route(self, st: _StreamState):
        """Parse + route the buffered request body; emit the ordered
        header-then-body response sequence."""
        body = bytes(st.body)
        path = st.headers.get(":path", "/v1/chat/completions")
        result = self.mux.parse_request(body, st.headers, path)
        if result.error or result.skip or result.request is None:
            # fallbackToRandomEndpoint (server.go:335-342)
            ep = self._random_endpoint()
            prom.request_error_total.labels("unknown", "parse_skip").inc()
            yield self._headers_response(ep.metadata.address if ep else "")
            yield from self._body_responses(None)
            return
        req = result.request
        prom.L(prom.request_sizes, req.model).observe(len(body))
        req.request_id = st.request_id
        req.headers.update(st.headers)
        if FAIRNESS_HEADER in st.headers:
            req.fairness_id = st.headers[FAIRNESS_HEADER]
        if OBJECTIVE_HEADER in st.headers:
            req.objective_name = st.headers[OBJECTIVE_HEADER]
        st.request = req
        st.streaming = bool(req.streaming)
        with self._lock:
            self._streams[st.request_id] = st
        try:
            decision = self.node.director.handle_request(req)
        except AdmissionDenied as e:
            yield self._immediate(e.status, e.reason, str(e))
            st.completed = True
            return
        st.decision = decision
        # repackage the (possibly rewritten) body (director.go:289)
        mutated = None
        if req.raw_body is not None and req.target_model != req.model:
            out = dict(req.raw_body)
            out["model"] = req.target_model
            mutated = json.dumps(out).encode()
        yield self._headers_response(decision.target_header)
        yield from self._body_responses(mutated)

    def _response_body(self, st: _StreamState, chunk: bytes,
                       end_of_stream: bool):
        st.response_body.extend(chunk)
        req = st.request
        rewritten = chunk
        if req is not None:
            rewritten = _rewrite_model_name(chunk, req.target_model,
                                            req.model)
        resp = pb.ProcessingResponse()
        cr = resp.response_body.response
        if rewritten != chunk:
            cr.status = pb.CONTINUE_AND_REPLACE
            cr.body_mutation.body = rewritten
        else:
            cr.status = pb.CONTINUE
        yield resp
        if end_of_stream:
            usage = Usage()
            if st.decision is not None and req is not None:
                parsed = self.mux.parse_response_usage(
                    bytes(st.response_body), st.headers)
                if parsed is not None:
                    usage = parsed
            self._finish(st, usage=usage)

    def _finish(self, st: _StreamState, usage: Optional[Usage] = None,
                error: str = "") -> None:
        if st.completed:
            return
        st.completed = True
        if st.decision is not None:
            self.node.director.handle_response_complete(
                st.decision, usage or Usage())
            if error:
                prom.request_error_total.labels(
                    st.request.model if st.request else "unknown",
                    error).inc()
