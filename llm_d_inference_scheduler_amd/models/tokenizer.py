"""Deterministic in-process tokenizer.

The reference delegates tokenization to vLLM's /render endpoint or a UDS
sidecar (dataproducer/tokenizer/{vllm_http,uds}.go) because the Go router
has no tokenizer. This node runs router and engines in one process space,
so they share one tokenizer. With no network egress there are no vocab
files; the HashTokenizer is a stable whitespace+punctuation splitter whose
ids are xxhash-derived — deterministic across router and engine, which is
all prefix-cache hashing and synthetic benchmarking need. A real
`transformers` tokenizer can be dropped in via the same callable interface
when vocab files are present.
"""
import re
from typing import List

from .. import _router_core as rc

_SPLIT = re.compile(r"\w+|[^\w\s]")


class HashTokenizer:
    def __init__(self, vocab_size: int = 128256):
        self.vocab_size = vocab_size

    def __call__(self, text: str) -> List[int]:
        return [rc.xxh64(w.encode("utf-8"), 0) % (self.vocab_size - 256) + 256
                for w in _SPLIT.findall(text)]

    def decode(self, token_ids: List[int]) -> str:
        # non-invertible by design; synthetic decode emits token markers
        return " ".join(f"<{t}>" for t in token_ids)
