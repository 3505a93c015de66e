"""Flow-control synchronous pipeline benchmark.

Parity: pkg/epp/flowcontrol/benchmark/benchmark.go:17-60 — a no-sleep
closed pipeline over the controller/registry measuring dispatches/s,
rejects/s and zombies/s (finalized-after-eviction races) under a mock
saturation signal, exercised across priority bands and flows.

Run: python -m llm_d_inference_scheduler_amd.flowcontrol.benchmark
"""
import argparse
import json
import time
from dataclasses import dataclass

from ..scheduling.types import LLMRequest
from .controller import FlowController
from .registry import BandConfig, FlowRegistry
from .types import FlowControlRequest, QueueOutcome


@dataclass
class BenchResult:
    duration_s: float
    submitted: int
    dispatched: int
    rejected: int
    evicted: int
    zombies: int

    @property
    def dispatches_per_s(self) -> float:
        return self.dispatched / self.duration_s

    @property
    def rejects_per_s(self) -> float:
        return (self.rejected + self.evicted) / self.duration_s

    @property
    def zombies_per_s(self) -> float:
        return self.zombies / self.duration_s

    def to_json(self) -> str:
        return json.dumps({
            "metric": "flow-control dispatch rate",
            "d_per_s": round(self.dispatches_per_s, 1),
            "r_per_s": round(self.rejects_per_s, 1),
            "zombies_per_s": round(self.zombies_per_s, 1),
            "submitted": self.submitted,
            "duration_s": round(self.duration_s, 3)})


def run_bench(duration_s: float = 2.0, n_flows: int = 8,
              saturated_every: int = 5, max_items: int = 4096,
              bands=None) -> BenchResult:
    bands = bands or [BandConfig(1, ordering="edf"),
                      BandConfig(0, ordering="fcfs"),
                      BandConfig(-1, ordering="fcfs")]
    registry = FlowRegistry(bands=bands, global_max_items=max_items)
    cycle = {"sat": False}

    def saturated() -> bool:
        return cycle["sat"]

    dispatched = {"n": 0}

    def dispatch(item) -> bool:
        dispatched["n"] += 1
        return True

    fc = FlowController(registry, dispatch, saturated_fn=saturated)
    stats = {"submitted": 0, "rejected": 0, "evicted": 0, "zombies": 0}
    t0 = time.perf_counter()
    i = 0
    tick_i = 0
    prios = [b.priority for b in bands]
    pending = []
    while time.perf_counter() - t0 < duration_s:
        tick_i += 1
        cycle["sat"] = (tick_i % saturated_every == 0)
        # burst-submit then tick, synchronous: the reference's no-sleep loop
        for _ in range(64):
            item = FlowControlRequest(
                request=LLMRequest(request_id=f"b{i}", model="bench",
                                   prompt="x" * 512),
                flow_key=f"flow-{i % n_flows}",
                priority=prios[i % len(prios)], byte_size=512,
                ttl_s=5.0)
            fc.submit(item)
            pending.append(item)
            stats["submitted"] += 1
            i += 1
        fc.tick()
        still = []
        for item in pending:
            if not item.finalized:
                still.append(item)
                continue
            out = item.outcome
            if out == QueueOutcome.REJECTED_CAPACITY:
                stats["rejected"] += 1
            elif out in (QueueOutcome.EVICTED_TTL,
                         QueueOutcome.EVICTED_DISPLACED,
                         QueueOutcome.EVICTED_SHUTDOWN):
                stats["evicted"] += 1
            elif out != QueueOutcome.DISPATCHED:
                stats["zombies"] += 1
        pending = still
    dur = time.perf_counter() - t0
    fc.tick()
    return BenchResult(duration_s=dur, submitted=stats["submitted"],
                       dispatched=dispatched["n"],
                       rejected=stats["rejected"],
                       evicted=stats["evicted"], zombies=stats["zombies"])


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=float, default=2.0)
    ap.add_argument("--flows", type=int, default=8)
    args = ap.parse_args()
    res = run_bench(duration_s=args.duration, n_flows=args.flows)
    print(res.to_json())
