"""Datalayer runtime collectors (reference runtime.go/collector.go behaviors
covered by runtime_polling_test.go): source registry, threaded 50ms ticker
lifecycle, datastore-driven track/untrack, and collect fail-open."""
import time

from llm_d_inference_scheduler_amd.datalayer.datastore import (Datastore,
                                                               make_endpoint)
from llm_d_inference_scheduler_amd.datalayer.endpoint import Metrics
from llm_d_inference_scheduler_amd.datalayer.extractor import \
    HttpMetricsSource
from llm_d_inference_scheduler_amd.datalayer.runtime import (CallableSource,
                                                             DataLayerRuntime)


def test_callable_source_stepped_collection():
    src = CallableSource()
    src.register("gpu0", lambda: Metrics(waiting_queue_size=5))
    rt = DataLayerRuntime()
    rt.add_source(src)
    ep = make_endpoint("gpu0", 0)
    rt.track(ep)
    rt.collect_all_now()
    assert ep.metrics.waiting_queue_size == 5


def test_threaded_ticker_and_stop():
    calls = []
    src = CallableSource()
    src.register("gpu0", lambda: calls.append(1) or Metrics(
        running_requests_size=len(calls)))
    rt = DataLayerRuntime(interval_s=0.005)
    rt.add_source(src)
    ep = make_endpoint("gpu0", 0)
    rt.track(ep)
    rt.start()
    time.sleep(0.06)
    rt.stop()
    n = len(calls)
    assert n >= 3                      # ticked repeatedly
    assert ep.metrics.running_requests_size >= 1
    time.sleep(0.02)
    assert len(calls) == n             # stopped means stopped


def test_bind_datastore_tracks_and_untracks():
    ds = Datastore()
    rt = DataLayerRuntime()
    src = CallableSource()
    src.register("a", lambda: Metrics(waiting_queue_size=1))
    rt.add_source(src)
    ds.add_endpoint(make_endpoint("a", 0))
    rt.bind_datastore(ds)
    assert "a" in rt._collectors
    ds.add_endpoint(make_endpoint("b", 1))
    assert "b" in rt._collectors      # event-driven track
    ds.remove_endpoint("a")
    assert "a" not in rt._collectors


def test_collect_exception_fail_open():
    class Boom(CallableSource):
        def collect(self, endpoint):
            raise RuntimeError("source down")
    rt = DataLayerRuntime()
    rt.add_source(Boom())
    ep = make_endpoint("gpu0", 0)
    ep.update_metrics(Metrics(waiting_queue_size=9, update_time=time.time()))
    rt.track(ep)
    rt.collect_all_now()               # must not raise
    assert ep.metrics.waiting_queue_size == 9   # stale snapshot retained


def test_http_source_through_collector():
    text = ("vllm:num_requests_waiting 4\n"
            "vllm:num_requests_running 2\n"
            "vllm:kv_cache_usage_perc 0.25\n")
    rt = DataLayerRuntime()
    rt.add_source(HttpMetricsSource(fetcher=lambda url: text))
    ep = make_endpoint("remote0", 0)
    rt.track(ep)
    rt.collect_all_now()
    m = ep.metrics
    assert (m.waiting_queue_size, m.running_requests_size,
            m.kv_cache_usage) == (4, 2, 0.25)
