"""Predicted-latency producer: in-process BayesianRidge training path
(reference dataproducer/predictedlatency + external predictor sidecar)."""
from types import SimpleNamespace

from llm_d_inference_scheduler_amd.datalayer.attributes import \
    LATENCY_PREDICTION_INFO
from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
from llm_d_inference_scheduler_amd.plugins.producers import \
    PredictedLatencyProducer
from llm_d_inference_scheduler_amd.scheduling.types import (LLMRequest,
                                                            SchedulingContext)


def make_ctx(rid, n_tokens=128):
    req = LLMRequest(request_id=rid, model="m", prompt="",
                     prompt_tokens=list(range(n_tokens)), max_tokens=8)
    return SchedulingContext(request=req)


def usage(ttft, tpot, prompt_tokens=128):
    return SimpleNamespace(ttft_ms=ttft, tpot_ms=tpot,
                           prompt_tokens=prompt_tokens, cached_tokens=0)


def test_cold_start_produces_headroom():
    p = PredictedLatencyProducer(ttftSLOms=1000.0)
    ep = make_endpoint("gpu0", 0)
    ep.update_metrics(Metrics(waiting_queue_size=2, running_requests_size=4))
    ctx = make_ctx("r0")
    p.produce(ctx, [ep])
    info = ctx.attributes[LATENCY_PREDICTION_INFO]
    assert info.predicted_ttft_ms["gpu0"] > 0
    assert info.ttft_headroom_ms["gpu0"] < 1000.0


def test_trains_and_uses_model():
    p = PredictedLatencyProducer(retrainEvery=64)
    ep = make_endpoint("gpu0", 0)
    # ground truth: ttft = 10*queue + 0.5*non_cached; tpot = 5 + 2*running
    for i in range(200):
        q, run, toks = i % 7, i % 5, 64 + (i % 3) * 64
        ep.update_metrics(Metrics(waiting_queue_size=q,
                                  running_requests_size=run))
        ctx = make_ctx(f"r{i}", n_tokens=toks)
        p.produce(ctx, [ep])
        p.pre_request(ctx, None, ep)
        p.response_complete(ctx, ep, usage(10.0 * q + 0.5 * toks,
                                           5.0 + 2.0 * run,
                                           prompt_tokens=toks))
    assert p._ttft_model is not None
    # prediction at a held-out state tracks the linear ground truth
    ep.update_metrics(Metrics(waiting_queue_size=3, running_requests_size=2))
    ctx = make_ctx("probe", n_tokens=128)
    p.produce(ctx, [ep])
    info = ctx.attributes[LATENCY_PREDICTION_INFO]
    want_ttft = 10.0 * 3 + 0.5 * 128
    assert abs(info.predicted_ttft_ms["gpu0"] - want_ttft) < 15.0
    want_tpot = 5.0 + 2.0 * 2
    assert abs(info.predicted_tpot_ms["gpu0"] - want_tpot) < 3.0


def test_lost_requests_do_not_leak():
    p = PredictedLatencyProducer(maxSamples=8)
    ep = make_endpoint("gpu0", 0)
    for i in range(100):
        ctx = make_ctx(f"lost{i}")
        p.pre_request(ctx, None, ep)
    assert len(p._pending_feats) <= 8 * p.max_samples
