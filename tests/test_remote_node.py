"""Multi-node routing: node A schedules onto a remote endpoint that is
node B's front door (node/remote.py + /internal/v1/enqueue +
HttpMetricsSource). The reference analog is cross-pool routing
(InferencePoolImport); here two full nodes run in-process on CPU."""
import json
import time

import pytest
import torch
from fastapi.testclient import TestClient

from llm_d_inference_scheduler_amd.datalayer.extractor import \
    HttpMetricsSource
from llm_d_inference_scheduler_amd.models.configs import TINY_LLAMA
from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
from llm_d_inference_scheduler_amd.node.remote import (REMOTE_URL_LABEL,
                                                       remote_endpoint)
from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
from llm_d_inference_scheduler_amd.server import NodeService, build_app


@pytest.fixture(scope="module")
def node_b_client():
    cfg = NodeConfig(model=TINY_LLAMA, device="cpu", dtype=torch.float32,
                     kv_blocks=256)
    service = NodeService(NodeRunner(cfg))
    service.start()
    with TestClient(build_app(service)) as c:
        yield c
    service.stop()


class TestInternalEnqueue:
    def test_enqueue_returns_tokens(self, node_b_client):
        r = node_b_client.post("/internal/v1/enqueue", json={
            "request_id": "x1", "model": "tiny-llama",
            "prompt_tokens": list(range(40)), "max_tokens": 4})
        assert r.status_code == 200
        body = r.json()
        assert len(body["tokens"]) == 4 and body["error"] == ""
        assert body["usage"]["prompt_tokens"] == 40

    def test_enqueue_rejects_garbage(self, node_b_client):
        assert node_b_client.post("/internal/v1/enqueue",
                                  content=b"{{{").status_code == 400
        assert node_b_client.post("/internal/v1/enqueue",
                                  json={}).status_code == 400


class TestCrossNodeRouting:
    def test_request_routed_to_remote_node(self, node_b_client):
        """Node A's scheduler picks the remote endpoint (subset hint);
        the request executes on node B and the completion surfaces at
        node A with response hooks run."""
        def transport(url, payload):
            r = node_b_client.post("/internal/v1/enqueue", json=payload)
            assert r.status_code == 200
            return r.json()

        node_a = NodeRunner(NodeConfig(model=TINY_LLAMA, device="cpu",
                                       dtype=torch.float32, kv_blocks=64,
                                       remote_transport=transport))
        ep = remote_endpoint("peer-b", 1, "http://node-b:8000")
        assert ep.metadata.labels[REMOTE_URL_LABEL]
        node_a.datastore.add_endpoint(ep)

        node_a.submit(LLMRequest(
            request_id="r-remote", model=TINY_LLAMA.name, prompt="",
            prompt_tokens=list(range(32)), max_tokens=3,
            subset_hint=["peer-b"]))
        comps = []
        for _ in range(200):
            node_a.step()
            comps = node_a.drain_completions()
            if comps:
                break
            time.sleep(0.02)       # remote round trip is asynchronous
        assert comps and comps[0].request_id == "r-remote"
        assert comps[0].error == "" and len(comps[0].tokens) == 3
        assert comps[0].usage.prompt_tokens == 32
        assert "r-remote" not in node_a._decisions      # bookkeeping done
        node_a.shutdown()

    def test_remote_failure_surfaces_error(self):
        def boom(url, payload):
            raise ConnectionError("peer down")
        node_a = NodeRunner(NodeConfig(model=TINY_LLAMA, device="cpu",
                                       dtype=torch.float32, kv_blocks=64,
                                       remote_transport=boom))
        node_a.datastore.add_endpoint(
            remote_endpoint("peer-b", 1, "http://node-b:8000"))
        node_a.submit(LLMRequest(
            request_id="r-fail", model=TINY_LLAMA.name, prompt="",
            prompt_tokens=list(range(16)), max_tokens=2,
            subset_hint=["peer-b"]))
        comps = []
        for _ in range(200):
            node_a.step()
            comps = node_a.drain_completions()
            if comps:
                break
            time.sleep(0.01)
        assert comps and comps[0].error.startswith("remote_error")
        node_a.shutdown()

    def test_remote_metrics_scrape_loop(self, node_b_client):
        """Node B's /metrics feeds node A's datastore through the
        HttpMetricsSource — the other half of the remote-endpoint
        contract."""
        src = HttpMetricsSource(
            fetcher=lambda url: node_b_client.get("/metrics").content.decode())
        ep = remote_endpoint("peer-b", 1, "http://node-b:8000")
        m = src.collect(ep)
        assert m is not None and m.cache_num_blocks > 0


class TestRemoteCancel:
    def test_cancel_reaches_peer(self, node_b_client):
        """cancel() on node A fires the internal cancel API on node B
        (fwd- prefixed id), unwinding the peer's engine state."""
        calls = []

        def transport(url, payload):
            calls.append((url, payload))
            if url.endswith("/enqueue"):
                # hold the request "in flight" on the peer: return slowly
                time.sleep(0.2)
                return {"tokens": [], "finish_reason": "length",
                        "error": "canceled", "usage": {}}
            return node_b_client.post("/internal/v1/cancel",
                                      json=payload).json()

        node_a = NodeRunner(NodeConfig(model=TINY_LLAMA, device="cpu",
                                       dtype=torch.float32, kv_blocks=64,
                                       remote_transport=transport))
        node_a.datastore.add_endpoint(
            remote_endpoint("peer-b", 1, "http://node-b:8000"))
        node_a.submit(LLMRequest(
            request_id="r-c", model=TINY_LLAMA.name, prompt="",
            prompt_tokens=list(range(16)), max_tokens=64,
            subset_hint=["peer-b"]))
        for _ in range(100):
            node_a.step()
            if any(u.endswith("/enqueue") for u, _ in calls):
                break
            time.sleep(0.01)
        node_a.cancel("r-c")
        for _ in range(100):
            node_a.step()
            if any(u.endswith("/cancel") for u, _ in calls):
                break
            time.sleep(0.01)
        cancel_calls = [(u, p) for u, p in calls if u.endswith("/cancel")]
        assert cancel_calls and cancel_calls[0][1] == {"request_id": "r-c"}
        # peer-side API accepts the id (no fwd- request exists -> no-op)
        r = node_b_client.post("/internal/v1/cancel",
                               json={"request_id": "r-c"})
        assert r.status_code == 200 and r.json()["canceled"] == "r-c"
        node_a.shutdown()


class TestCrossNodeSSERelay:
    """Cross-node SSE token relay (round-2: streaming across nodes; the
    non-streaming v1 buffered whole completions)."""

    def test_streaming_enqueue_emits_sse(self):
        """The peer side: /internal/v1/enqueue with stream=true returns
        SSE token chunks then a final done record."""
        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=128)
        service = NodeService(NodeRunner(cfg))
        service.start()
        with TestClient(build_app(service)) as c:
            with c.stream("POST", "/internal/v1/enqueue", json={
                    "request_id": "s1", "model": "tiny-llama",
                    "prompt_tokens": list(range(40)), "max_tokens": 4,
                    "stream": True}) as r:
                assert r.status_code == 200
                events = []
                for line in r.iter_lines():
                    line = line.strip()
                    if line.startswith("data:") and \
                            line[5:].strip() != "[DONE]":
                        events.append(json.loads(line[5:]))
        service.stop()
        final = events[-1]
        assert final["done"] and final["error"] == ""
        assert final["usage"]["completion_tokens"] == 4
        streamed = [t for e in events[:-1] for t in e.get("tokens", [])]
        assert len(streamed) == 4
        assert streamed == final["tokens"]

    def test_forwarder_relays_stream_to_router(self):
        """The forwarding side: stream events surface as router token
        events while the peer generates, then close as a Completion."""
        from llm_d_inference_scheduler_amd.node.remote import (
            RemoteForwarder, remote_endpoint)

        def fake_stream(url, payload):
            assert payload["stream"] is True
            yield {"tokens": [11, 12]}
            yield {"tokens": [13]}
            yield {"done": True, "tokens": [11, 12, 13],
                   "finish_reason": "length", "error": "",
                   "usage": {"prompt_tokens": 5, "completion_tokens": 3}}

        cfg = NodeConfig(model=TINY_LLAMA, world_size=1, topology="mono",
                         device="cpu", dtype=torch.float32, kv_blocks=64)
        node = NodeRunner(cfg)
        node.remote = RemoteForwarder(stream_transport=fake_stream)
        node.datastore.add_endpoint(remote_endpoint(
            "peer0", 9, "http://peer:8000"))
        from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
        req = LLMRequest(request_id="sr1", model="tiny-llama",
                         prompt_tokens=list(range(30)), max_tokens=3,
                         streaming=True, subset_hint=["peer0"])
        node.submit(req)
        done = []
        for _ in range(100):
            node.step()
            done.extend(node.drain_completions())
            if done:
                break
        assert done and not done[0].error
        assert done[0].tokens == [11, 12, 13]
        toks = [t for _, chunk in node.drain_token_events() for t in chunk]
        assert toks == [11, 12, 13]
        node.shutdown()


class TestPoolImport:
    """InferencePoolImport analog: declaring a remote pool materializes
    peer front doors as schedulable endpoints; deletion removes them."""

    def test_apply_route_and_delete(self):
        from llm_d_inference_scheduler_amd.api.poolimport import (
            ImportedEndpoint, InferencePoolImport, PoolImportManager)

        def transport(url, payload):
            return {"tokens": [5, 6], "finish_reason": "length",
                    "error": "", "usage": {"prompt_tokens": 4,
                                           "completion_tokens": 2}}
        node = NodeRunner(NodeConfig(model=TINY_LLAMA, device="cpu",
                                     dtype=torch.float32, kv_blocks=64,
                                     remote_transport=transport))
        mgr = PoolImportManager(node.datastore)
        imp = InferencePoolImport("peerpool", endpoints=[
            ImportedEndpoint("http://peer-a:8000"),
            ImportedEndpoint("http://peer-b:8000")])
        names = mgr.apply(imp)
        assert len(names) == 2
        eps = {ep.name for ep in node.datastore.endpoints()}
        assert set(names) <= eps
        # imported endpoints are schedulable: steer a request there
        node.submit(LLMRequest(
            request_id="imp1", model=TINY_LLAMA.name, prompt="",
            prompt_tokens=list(range(12)), max_tokens=2,
            subset_hint=names))
        done = []
        for _ in range(200):
            node.step()
            done.extend(node.drain_completions())
            if done:
                break
            time.sleep(0.005)
        assert done and done[0].tokens == [5, 6]
        mgr.delete("peerpool")
        eps = {ep.name for ep in node.datastore.endpoints()}
        assert not (set(names) & eps)
        node.shutdown()

    def test_reapply_replaces(self):
        from llm_d_inference_scheduler_amd.api.poolimport import (
            ImportedEndpoint, InferencePoolImport, PoolImportManager)
        from llm_d_inference_scheduler_amd.datalayer.datastore import \
            Datastore
        ds = Datastore()
        mgr = PoolImportManager(ds)
        mgr.apply(InferencePoolImport("p", [ImportedEndpoint("http://a")]))
        mgr.apply(InferencePoolImport("p", [ImportedEndpoint("http://b")]))
        eps = list(ds.endpoints())
        assert len(eps) == 1
        assert eps[0].metadata.labels["llm-d.ai/remote-url"] == "http://b"
