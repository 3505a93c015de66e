#!/usr/bin/env python3
"""Router-core overhead profile at the 8-GPU node shape (round-1 verdict
item 6: quantify — or eliminate — the Python router tax).

Measures, on CPU (the router rank's work is CPU-side either way):
  1. director.handle_request throughput over 8 endpoints with the default
     production plugin stack (token producer, approx-prefix hash+match,
     inflight-load, scorers, max-score picker) — single-threaded, the
     node-loop routing path;
  2. the per-step mailbox exchange cost at world=8 (gloo
     all_gather_object of typical assign/metrics payloads);
  3. the derived per-step router overhead at a target offered load
     (default 2000 req/s, 20 ms steps -> 40 routes/step).

Writes profiles/router_tax.json and prints a summary. Run:
  python tools/router_tax_profile.py [--requests 20000] [--world 8]
"""
import argparse
import cProfile
import io
import json
import os
import pstats
import random
import statistics
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def build_rig(n_endpoints=8):
    import torch
    from llm_d_inference_scheduler_amd.models.configs import LLAMA_3_8B
    from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
    # world_size=8 mono topology: datastore carries 8 endpoints; only
    # rank 0 (router + its own engine) exists in-process. Tiny KV pool:
    # the engine itself is not under test.
    cfg = NodeConfig(model=LLAMA_3_8B, rank=0, world_size=n_endpoints,
                     topology="mono", device="cpu", dtype=torch.float32,
                     kv_blocks=64)
    return NodeRunner(cfg)


def gen_requests(n, rng, prompt_len=1024, shared=0.5, group=4, vocab=128000):
    from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
    reqs = []
    prefix = None
    for i in range(n):
        ns = int(prompt_len * shared)
        if i % group == 0 or prefix is None:
            prefix = [rng.randrange(256, vocab) for _ in range(ns)]
        toks = prefix + [rng.randrange(256, vocab)
                         for _ in range(prompt_len - ns)]
        reqs.append(LLMRequest(request_id=f"r{i}", model="llama-3-8b",
                               prompt_tokens=toks, max_tokens=1024,
                               prompt=""))
    return reqs


def bench_routing(node, reqs, profile=False):
    lat = []
    prof = cProfile.Profile() if profile else None
    if prof:
        prof.enable()
    t0 = time.perf_counter()
    for req in reqs:
        s = time.perf_counter()
        d = node.director.handle_request(req)
        lat.append(time.perf_counter() - s)
        node.director.handle_response_complete(d, None)
    wall = time.perf_counter() - t0
    if prof:
        prof.disable()
    top = ""
    if prof:
        buf = io.StringIO()
        pstats.Stats(prof, stream=buf).sort_stats("cumulative").print_stats(18)
        top = buf.getvalue()
    lat.sort()
    return {
        "requests": len(reqs),
        "wall_s": round(wall, 4),
        "req_per_s": round(len(reqs) / wall, 1),
        "p50_us": round(lat[len(lat) // 2] * 1e6, 1),
        "p99_us": round(lat[int(len(lat) * 0.99)] * 1e6, 1),
    }, top


def bench_mailbox(world):
    """gloo all_gather_object cost at world size, typical payload mix."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_mailbox_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    out = q.get(timeout=120)
    for p in procs:
        p.join(30)
    return out


def _mailbox_worker(rank, world, q):
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29617")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    payload = [{"type": "metrics", "src": rank,
                "m": {"waiting": 3, "running": 250, "kv": 0.71,
                      "bs": 16, "nb": 80000}}]
    if rank == 0:
        payload += [{"type": "assign", "req_id": f"q{i}", "dst": i % world,
                     "tokens": list(range(1024)), "max_tokens": 1024,
                     "temperature": 0.0, "cached": 512, "priority": 0,
                     "stop": None, "arrival": time.time()}
                    for i in range(2)]
    # warmup
    for _ in range(20):
        buf = [None] * world
        dist.all_gather_object(buf, payload)
    n = 200
    t0 = time.perf_counter()
    for _ in range(n):
        buf = [None] * world
        dist.all_gather_object(buf, payload)
    per_step_ms = (time.perf_counter() - t0) / n * 1e3
    if rank == 0:
        q.put({"world": world, "per_step_ms": round(per_step_ms, 3)})
    dist.destroy_process_group()


def bench_async_step_tax(reqs, per_step=40, world=8):
    """Step-thread cost of routing with route_async=True: submit + drain +
    assignment emission happen on the step thread; handle_request runs on
    the pool. This is what rank 0's lockstep actually pays per step."""
    import torch
    from llm_d_inference_scheduler_amd.models.configs import LLAMA_3_8B
    from llm_d_inference_scheduler_amd.node import NodeConfig, NodeRunner
    cfg = NodeConfig(model=LLAMA_3_8B, rank=0, world_size=world,
                     topology="mono", device="cpu", dtype=torch.float32,
                     kv_blocks=64, route_async=True)
    node = NodeRunner(cfg)
    it = iter(reqs)
    blocked = 0.0
    steps = 0
    routed = 0
    t_wall0 = time.perf_counter()
    exhausted = False
    while routed < len(reqs):
        if not exhausted:
            for _ in range(per_step):
                try:
                    node.submit(next(it))
                except StopIteration:
                    exhausted = True
                    break
        t0 = time.perf_counter()
        before = len(node._outbox)
        node._route_arrivals()
        blocked += time.perf_counter() - t0
        routed += len(node._outbox) - before
        node._outbox = []
        node._decisions.clear()     # keep bookkeeping bounded
        steps += 1
        time.sleep(0.001)           # step cadence stand-in (pool runs)
    wall = time.perf_counter() - t_wall0
    node.shutdown()
    return {
        "requests": len(reqs), "steps": steps,
        "step_blocked_ms": round(blocked / steps * 1e3, 3),
        "pool_req_per_s": round(len(reqs) / wall, 1),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=20000)
    ap.add_argument("--world", type=int, default=8)
    ap.add_argument("--target-req-s", type=float, default=2000.0)
    ap.add_argument("--step-ms", type=float, default=20.0)
    ap.add_argument("--profile", action="store_true")
    args = ap.parse_args()

    rng = random.Random(7)
    node = build_rig(args.world)
    reqs = gen_requests(args.requests, rng)
    # warmup (prefix index fills, code paths warm)
    bench_routing(node, reqs[:2000])
    routing, top = bench_routing(node, reqs[2000:], profile=args.profile)
    mailbox = bench_mailbox(args.world)
    routes_per_step = args.target_req_s * args.step_ms / 1e3
    async_tax = bench_async_step_tax(reqs[:8000],
                                     per_step=int(routes_per_step),
                                     world=args.world)

    route_ms = routes_per_step * routing["p50_us"] / 1e3
    # production config: routing on the async pool (step pays only
    # submit+drain+emit) and the mailbox exchange pipelined behind the
    # engine step (blocked time ~0 while exchange < step)
    mailbox_blocked = max(0.0, mailbox["per_step_ms"] - args.step_ms)
    total_ms = async_tax["step_blocked_ms"] + mailbox_blocked
    result = {
        "shape": {"endpoints": args.world, "target_req_s": args.target_req_s,
                  "step_ms": args.step_ms,
                  "routes_per_step": routes_per_step},
        "routing_inline": routing,
        "mailbox_raw": mailbox,
        "routing_async": async_tax,
        "inline_route_ms_per_step": round(route_ms, 3),
        "async_step_blocked_ms": async_tax["step_blocked_ms"],
        "mailbox_blocked_ms_pipelined": round(mailbox_blocked, 3),
        "router_overhead_pct_of_step": round(100 * total_ms / args.step_ms,
                                             2),
        "router_overhead_pct_of_step_inline": round(
            100 * (route_ms + mailbox["per_step_ms"]) / args.step_ms, 2),
    }
    print(json.dumps(result, indent=2))
    if top:
        print(top)
    out = os.path.join(os.path.dirname(__file__), "..", "profiles",
                       "router_tax.json")
    with open(out, "w") as f:
        json.dump(result, f, indent=2)
    node.shutdown()


if __name__ == "__main__":
    main()
