"""Driver-contract guard: `bench.py` must emit exactly one JSON line with
the agreed fields (the round driver parses this; see BASELINE.json). Runs
the tiny model on CPU so the contract is checked without a GPU."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--device", "cpu",
         "--steps", "4", "--warmup", "2", "--prompt-len", "32",
         "--max-tokens", "8", "--concurrency", "4", "--kv-gb", "0.01"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines()
                  if l.startswith('{"metric"')]
    assert len(json_lines) == 1, out.stdout[-2000:]
    r = json.loads(json_lines[0])
    missing = [k for k in REQUIRED if k not in r]
    assert not missing, missing
    assert r["n_gpus"] == 1 and r["steps"] == 4 and r["warmup"] == 2
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["value"] > 0 and r["ms_per_step"] > 0
    cfg = r["config"]
    for k in ("model", "global_batch", "seq_len", "parallelism"):
        assert k in cfg, k
    assert cfg["seq_len"] == 40  # prompt 32 + out 8


def test_bench_multirank_contract():
    """The driver's exact N>1 invocation: torch.distributed.run, one rank
    per device, rank 0 prints ONE JSON line with n_gpus=N (scale-run
    insurance; gloo on CPU here, RCCL on the GPU box)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29631", "bench.py", "--gpus", "2",
         "--model", "tiny", "--device", "cpu", "--steps", "4",
         "--warmup", "2", "--prompt-len", "32", "--max-tokens", "8",
         "--concurrency", "4", "--kv-gb", "0.01"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines()
                  if l.startswith('{"metric"')]
    assert len(json_lines) == 1, out.stdout[-1500:]
    r = json.loads(json_lines[0])
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["value"] > 0
