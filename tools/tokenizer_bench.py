"""Tokenizer throughput microbench (reference `make bench-tokenizer`,
test/profiling/tokenizerbench/). CPU-only; measures the in-process
HashTokenizer the router uses for prefix hashing of chat bodies.

    python tools/tokenizer_bench.py [--seconds 2]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

from llm_d_inference_scheduler_amd.models.tokenizer import HashTokenizer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=2.0)
    args = ap.parse_args()
    tok = HashTokenizer()
    # mixed prose + punctuation, ~1.3 KB — the chat-body shape the router sees
    text = ("The quick brown fox, jumping over 13 lazy dogs near the xGMI "
            "fabric; paged KV-caches (block=16) hash deterministically! ") * 16
    n_tokens = len(tok(text))
    docs = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        tok(text)
        docs += 1
    dt = time.perf_counter() - t0
    print(f"{docs / dt:,.0f} docs/s | {docs * n_tokens / dt / 1e6:.2f} M tokens/s "
          f"| {docs * len(text) / dt / 1e6:.1f} MB/s ({n_tokens} tok/doc)")


if __name__ == "__main__":
    main()
