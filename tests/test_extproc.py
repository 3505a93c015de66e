"""Loopback ext-proc gRPC tests (the hermetic harness analog of the
reference's test/integration/epp/grpc_test.go): a real grpc client drives
the FULL_DUPLEX_STREAMED ProcessingRequest/ProcessingResponse exchange
against the node's ext-proc front door."""
import json
import queue
import threading
import time

import grpc
import pytest
import torch

from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
from llm_d_inference_scheduler_amd.server import extproc_pb as pb
from llm_d_inference_scheduler_amd.server.extproc import (
    BODY_BYTE_LIMIT, DEST_ENDPOINT_HEADER, DEST_METADATA_NAMESPACE,
    DROPPED_REASON_HEADER, METHOD, SERVICE, ExtProcServer)


@pytest.fixture()
def node():
    cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                     device="cpu", dtype=torch.float32, kv_blocks=256)
    n = NodeRunner(cfg)
    yield n
    n.shutdown()


@pytest.fixture()
def server(node):
    s = ExtProcServer(node)
    s.start(0)
    yield s
    s.stop()


class Stream:
    """Interactive bidirectional stream: push requests, pull responses."""

    def __init__(self, port):
        self.channel = grpc.insecure_channel(f"127.0.0.1:{port}")
        self._q = queue.Queue()
        callable_ = self.channel.stream_stream(
            f"/{SERVICE}/{METHOD}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=pb.ProcessingResponse.FromString)
        self._resp_iter = callable_(iter(self._q.get, None))

    def send(self, msg):
        self._q.put(msg)

    def recv(self, timeout=10.0):
        box = {}

        def _take():
            try:
                box["v"] = next(self._resp_iter)
            except StopIteration:
                box["v"] = None
        t = threading.Thread(target=_take, daemon=True)
        t.start()
        t.join(timeout)
        assert "v" in box, "timed out waiting for ext-proc response"
        return box["v"]

    def close(self):
        self._q.put(None)
        self.channel.close()


def req_headers(headers, end_of_stream=False):
    m = pb.ProcessingRequest()
    for k, v in headers.items():
        m.request_headers.headers.headers.add(key=k, raw_value=v.encode())
    m.request_headers.end_of_stream = end_of_stream
    return m


def req_body(data: bytes, end_of_stream=False):
    m = pb.ProcessingRequest()
    m.request_body.body = data
    m.request_body.end_of_stream = end_of_stream
    return m


def resp_headers(end_of_stream=False):
    m = pb.ProcessingRequest()
    m.response_headers.SetInParent()
    return m


def resp_body(data: bytes, end_of_stream=False):
    m = pb.ProcessingRequest()
    m.response_body.body = data
    m.response_body.end_of_stream = end_of_stream
    return m


def completion_body(model="tiny-llama", prompt="hello world", **kw):
    return json.dumps({"model": model, "prompt": prompt,
                       "max_tokens": 4, **kw}).encode()


class TestExtProcHappyPath:
    def test_headers_then_chunked_body_routes(self, server):
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json",
                             "x-request-id": "req-1"}))
        body = completion_body()
        st.send(req_body(body[:10]))
        st.send(req_body(body[10:], end_of_stream=True))
        # ordered sends: header response FIRST (server.go:489-598)
        r1 = st.recv()
        assert r1.WhichOneof("response") == "request_headers"
        muts = {o.header.key: o.header.raw_value.decode()
                for o in r1.request_headers.response.header_mutation
                .set_headers}
        assert muts[DEST_ENDPOINT_HEADER] == "rank:0"
        md = r1.dynamic_metadata.fields[DEST_METADATA_NAMESPACE].struct_value
        assert md.fields[DEST_ENDPOINT_HEADER].string_value == "rank:0"
        # then exactly the body response (no mutation -> CONTINUE)
        r2 = st.recv()
        assert r2.WhichOneof("response") == "request_body"
        assert r2.request_body.response.status == pb.CONTINUE
        # response direction
        st.send(resp_headers())
        r3 = st.recv()
        assert r3.WhichOneof("response") == "response_headers"
        usage_body = json.dumps({
            "model": "tiny-llama", "choices": [],
            "usage": {"prompt_tokens": 2, "completion_tokens": 4}}).encode()
        st.send(resp_body(usage_body, end_of_stream=True))
        r4 = st.recv()
        assert r4.WhichOneof("response") == "response_body"
        assert r4.response_body.response.status == pb.CONTINUE
        st.close()

    def test_bodyless_get_falls_back_to_random_endpoint(self, server):
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/models", ":method": "GET"},
                            end_of_stream=True))
        r = st.recv()
        assert r.WhichOneof("response") == "request_headers"
        muts = {o.header.key: o.header.raw_value.decode()
                for o in r.request_headers.response.header_mutation
                .set_headers}
        assert muts[DEST_ENDPOINT_HEADER] == "rank:0"
        st.close()

    def test_parse_skip_falls_back(self, server):
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json"}))
        st.send(req_body(b"this is not json", end_of_stream=True))
        r1 = st.recv()
        assert r1.WhichOneof("response") == "request_headers"
        r2 = st.recv()
        assert r2.request_body.response.status == pb.CONTINUE
        st.close()


class TestExtProcRewrite:
    def test_body_mutation_chunked_and_rewritten_back(self, node, server):
        from llm_d_inference_scheduler_amd.api.modelrewrite import (
            InferenceModelRewrite, RewriteRule, RewriteTarget)
        node.datastore.put_model_rewrite(InferenceModelRewrite(
            name="rw", rules=[RewriteRule(
                model="tiny-llama",
                targets=[RewriteTarget("tiny-llama-instruct", weight=1)])]))
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json",
                             "x-request-id": "req-rw"}))
        # body large enough that the mutated body needs >1 chunk
        big_prompt = "x" * (BODY_BYTE_LIMIT + 4096)
        st.send(req_body(completion_body(prompt=big_prompt),
                         end_of_stream=True))
        r1 = st.recv()
        assert r1.WhichOneof("response") == "request_headers"
        chunks = []
        while True:
            r = st.recv()
            assert r.WhichOneof("response") == "request_body"
            cr = r.request_body.response
            assert cr.status == pb.CONTINUE_AND_REPLACE
            assert len(cr.body_mutation.body) <= BODY_BYTE_LIMIT
            chunks.append(cr.body_mutation.body)
            if len(b"".join(chunks)) >= len(big_prompt):
                break
        mutated = json.loads(b"".join(chunks))
        assert mutated["model"] == "tiny-llama-instruct"  # rewritten
        assert len(chunks) >= 2                           # 62KB chunking
        # response body: served-model name rewritten BACK (server.go:471)
        st.send(resp_headers())
        st.recv()
        served = json.dumps({"model": "tiny-llama-instruct",
                             "choices": [],
                             "usage": {"prompt_tokens": 1,
                                       "completion_tokens": 1}}).encode()
        st.send(resp_body(served, end_of_stream=True))
        r = st.recv()
        cr = r.response_body.response
        assert cr.status == pb.CONTINUE_AND_REPLACE
        assert json.loads(cr.body_mutation.body)["model"] == "tiny-llama"
        st.close()


class TestExtProcErrors:
    def test_no_endpoints_immediate_response(self, node, server):
        for ep in list(node.datastore.endpoints()):
            node.datastore.remove_endpoint(ep.name)
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json"}))
        st.send(req_body(completion_body(), end_of_stream=True))
        r = st.recv()
        assert r.WhichOneof("response") == "immediate_response"
        imm = r.immediate_response
        assert imm.status.code == 503
        muts = {o.header.key: o.header.raw_value.decode()
                for o in imm.headers.set_headers}
        assert muts[DROPPED_REASON_HEADER] == "no_endpoints"
        st.close()

    def test_midstream_eviction_429(self, server):
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json",
                             "x-request-id": "req-evict"}))
        st.send(req_body(completion_body(), end_of_stream=True))
        st.recv()   # headers response
        st.recv()   # body response
        # flow-control eviction fires while the response is pending
        deadline = time.time() + 5
        while not server.evict("req-evict", "evicted") and \
                time.time() < deadline:
            time.sleep(0.01)
        st.send(resp_headers())
        r = st.recv()
        assert r.WhichOneof("response") == "immediate_response"
        assert r.immediate_response.status.code == 429
        muts = {o.header.key: o.header.raw_value.decode()
                for o in r.immediate_response.headers.set_headers}
        assert muts[DROPPED_REASON_HEADER] == "evicted"
        st.close()


class TestExtProcTrailers:
    def test_trailers_get_trailer_responses(self, server):
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/completions",
                             "content-type": "application/json"}))
        st.send(req_body(completion_body(), end_of_stream=True))
        st.recv()   # headers response
        st.recv()   # body response
        m = pb.ProcessingRequest()
        m.request_trailers.SetInParent()
        st.send(m)
        r = st.recv()
        assert r.WhichOneof("response") == "request_trailers"
        st.send(resp_headers())
        st.recv()
        m = pb.ProcessingRequest()
        m.response_trailers.SetInParent()
        st.send(m)
        r = st.recv()
        assert r.WhichOneof("response") == "response_trailers"
        st.close()


class TestExtProcSSEResponses:
    def test_sse_chunks_rewritten_per_chunk(self, node, server):
        """Streaming upstream responses: each SSE chunk's served model
        name is rewritten back independently (server.go rewriteModelName
        applies per streamed chunk)."""
        from llm_d_inference_scheduler_amd.api.modelrewrite import (
            InferenceModelRewrite, RewriteRule, RewriteTarget)
        node.datastore.put_model_rewrite(InferenceModelRewrite(
            name="rw", rules=[RewriteRule(
                model="tiny-llama",
                targets=[RewriteTarget("tiny-llama-x", weight=1)])]))
        st = Stream(server.port)
        st.send(req_headers({":path": "/v1/chat/completions",
                             "content-type": "application/json",
                             "x-request-id": "req-sse"}))
        st.send(req_body(json.dumps({
            "model": "tiny-llama", "stream": True,
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 4}).encode(), end_of_stream=True))
        st.recv()               # headers response
        st.recv()               # mutated body (rewrite)
        st.send(resp_headers())
        st.recv()
        chunk1 = ('data: ' + json.dumps(
            {"model": "tiny-llama-x",
             "choices": [{"delta": {"content": "a"}}]}) + "\n\n").encode()
        st.send(resp_body(chunk1))
        r = st.recv()
        cr = r.response_body.response
        assert cr.status == pb.CONTINUE_AND_REPLACE
        out = cr.body_mutation.body.decode()
        assert '"model": "tiny-llama"' in out and "tiny-llama-x" not in out
        # final chunk with usage closes the stream accounting
        final = ("data: " + json.dumps(
            {"model": "tiny-llama-x",
             "usage": {"prompt_tokens": 1, "completion_tokens": 4}})
            + "\n\ndata: [DONE]\n\n").encode()
        st.send(resp_body(final, end_of_stream=True))
        r = st.recv()
        assert r.WhichOneof("response") == "response_body"
        st.close()


class TestExtProcConcurrentStreams:
    def test_parallel_streams_route_independently(self, server):
        """Several simultaneous Process streams (the gRPC thread pool +
        director scheduling lock): each gets its own routed response."""
        def drive(i, results):
            st = Stream(server.port)
            try:
                st.send(req_headers({":path": "/v1/completions",
                                     "content-type": "application/json",
                                     "x-request-id": f"par-{i}"}))
                st.send(req_body(completion_body(
                    prompt=f"prompt {i} " * 8), end_of_stream=True))
                r1 = st.recv()
                ok = r1.WhichOneof("response") == "request_headers"
                r2 = st.recv()
                ok = ok and r2.WhichOneof("response") == "request_body"
                st.send(resp_headers())
                st.recv()
                st.send(resp_body(b'{"model":"tiny-llama","usage":'
                                  b'{"prompt_tokens":1,'
                                  b'"completion_tokens":1}}',
                                  end_of_stream=True))
                r4 = st.recv()
                ok = ok and r4.WhichOneof("response") == "response_body"
                results[i] = ok
            finally:
                st.close()
        results = {}
        threads = [threading.Thread(target=drive, args=(i, results))
                   for i in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(20.0)
        assert len(results) == 8 and all(results.values()), results
