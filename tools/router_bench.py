"""BASELINE config-1 microbench: EPP routing throughput on CPU — routed
req/s + p50 EPP latency through the full Director pipeline (parse-less
path) over mock endpoints with fake metrics, ext-proc-loopback style.

  python tools/router_bench.py [--endpoints 8] [--seconds 2]
"""
import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

from llm_d_inference_scheduler_amd.config import load_config  # noqa: E402
from llm_d_inference_scheduler_amd.datalayer.datastore import (  # noqa: E402
    Datastore, make_endpoint)
from llm_d_inference_scheduler_amd.datalayer.endpoint import \
    Metrics  # noqa: E402
from llm_d_inference_scheduler_amd.flowcontrol.saturation import \
    UtilizationSaturationDetector  # noqa: E402
from llm_d_inference_scheduler_amd.requestcontrol import (  # noqa: E402
    Director, EndpointCandidates, LegacyAdmissionController)
from llm_d_inference_scheduler_amd.scheduling.scheduler import \
    Scheduler  # noqa: E402
from llm_d_inference_scheduler_amd.scheduling.types import \
    LLMRequest  # noqa: E402

YAML = """
plugins:
  - type: decode-filter
  - type: queue-scorer
  - type: kv-cache-utilization-scorer
  - type: prefix-cache-scorer
  - type: inflight-load-producer
  - type: max-score-picker
  - type: single-profile-handler
schedulingProfiles:
  - name: default
    plugins:
      - {pluginRef: decode-filter}
      - {pluginRef: prefix-cache-scorer, weight: 3}
      - {pluginRef: queue-scorer, weight: 1}
      - {pluginRef: kv-cache-utilization-scorer, weight: 1}
      - {pluginRef: max-score-picker}
"""


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoints", type=int, default=8)
    ap.add_argument("--seconds", type=float, default=2.0)
    ap.add_argument("--prompt-tokens", type=int, default=1024)
    args = ap.parse_args()
    loaded = load_config(YAML)
    ds = Datastore()
    for i in range(args.endpoints):
        ep = make_endpoint(f"gpu{i}", i, role="decode")
        ep.update_metrics(Metrics(waiting_queue_size=i % 4,
                                  kv_cache_usage=0.1 * (i % 5),
                                  cache_block_size=16,
                                  cache_num_blocks=10000))
        ds.add_endpoint(ep)
    ds.set_pool_ready()
    director = Director(datastore=ds,
                        scheduler=Scheduler(loaded.scheduler_config),
                        admission=LegacyAdmissionController(
                            UtilizationSaturationDetector()),
                        candidates=EndpointCandidates(ds),
                        config=loaded.request_control)
    rng = random.Random(0)
    shared = [rng.randrange(256, 100000) for _ in range(512)]
    lat = []
    n = 0
    t_end = time.perf_counter() + args.seconds
    t_start = time.perf_counter()
    while time.perf_counter() < t_end:
        toks = shared + [rng.randrange(256, 100000) for _ in range(
            args.prompt_tokens - 512)]
        req = LLMRequest(request_id=f"r{n}", model="llama-3-8b", prompt="",
                         prompt_tokens=toks, max_tokens=128)
        t0 = time.perf_counter()
        decision = director.handle_request(req)
        lat.append((time.perf_counter() - t0) * 1e3)
        assert decision.target is not None
        n += 1
    dur = time.perf_counter() - t_start
    lat.sort()
    print({"routed_req_s": round(n / dur, 1),
           "p50_epp_latency_ms": round(lat[len(lat) // 2], 3),
           "p99_epp_latency_ms": round(lat[int(len(lat) * 0.99)], 3),
           "endpoints": args.endpoints, "n": n})


if __name__ == "__main__":
    main()
