"""Latency-throughput curve over a request-rate sweep (the reference's
regression-harness methodology, single-workload-regression.yaml:30-45;
BASELINE.md "harness shape to reuse"). Run on a GPU box:

  python tools/rate_sweep.py --rates 10 20 30 40 50 --steps 400 -- --max-tokens 128

Unknown args after `--` pass through to bench.py (e.g. --mode fc,
--max-tokens). The reference's harness runs 300 s per point at out=1024;
a GPU-minute-bounded sweep instead uses out=128 so requests complete
inside the window and the latency-throughput knee is visible.
"""
import argparse
import json
import os
import subprocess
import sys

ROOT = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rates", type=float, nargs="+",
                    default=[4, 8, 12, 16, 20])
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--warmup", type=int, default=60)
    ap.add_argument("--extra", nargs="*", default=[])
    args, passthrough = ap.parse_known_args()
    args.extra = list(args.extra) + [a for a in passthrough if a != "--"]
    outdir = os.environ.get("SWEEP_OUT", os.path.join(ROOT, "gpurun_out"))
    os.makedirs(outdir, exist_ok=True)
    rows = []
    for r in args.rates:
        cmd = [sys.executable, os.path.join(ROOT, "bench.py"),
               "--steps", str(args.steps), "--warmup", str(args.warmup),
               "--arrival-rate", str(r)] + args.extra
        out = subprocess.run(cmd, capture_output=True, text=True)
        tag = f"sweep_r{r:g}" + "".join(args.extra).replace("--", "_")
        with open(os.path.join(outdir, tag + ".json"), "w") as f:
            f.write(out.stdout)
        if out.stderr:
            with open(os.path.join(outdir, tag + ".err"), "w") as f:
                f.write(out.stderr[-20000:])
        if not out.stdout.strip():
            print(f"rate {r}: bench produced no output "
                  f"(stderr tail: {out.stderr[-400:]})", flush=True)
            continue
        line = out.stdout.strip().splitlines()[-1]
        d = json.loads(line)
        rows.append((r, d))
        c = d["config"]
        print(f"rate {r:6.1f} req/s -> goodput {d['value']:8.1f} tok/s  "
              f"routed {c['routed_req_s']:6.2f}/s  "
              f"p50_ttft {c['p50_ttft_ms']}  p99_ttft {c['p99_ttft_ms']}",
              flush=True)
    print(json.dumps([{ "rate": r, "goodput": d["value"],
                        "p50_ttft_ms": d["config"]["p50_ttft_ms"],
                        "p99_ttft_ms": d["config"]["p99_ttft_ms"],
                        "routed_req_s": d["config"]["routed_req_s"]}
                      for r, d in rows]))


if __name__ == "__main__":
    main()
