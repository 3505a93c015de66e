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


class HttpTokenizer:
    """Remote tokenizer client — the reference's vLLM-HTTP / UDS-sidecar
    tokenizer surface (dataproducer/tokenizer/{vllm_http,uds}.go:41-80).

    POSTs `{model, prompt}` to `<base_url>/tokenize` (vLLM-compatible; this
    node's own front door serves the same route, so any peer node or real
    vLLM worker can act as the tokenizer). `uds_path` routes the same HTTP
    over a unix socket (the reference's UDS sidecar transport). Fail-open:
    any transport error falls back to the in-process hash tokenizer, like
    the reference's parse-skip fallbacks — routing must not die because a
    tokenizer worker is down."""

    def __init__(self, base_url: str = "", model: str = "",
                 uds_path: str = "", timeout_s: float = 0.4,
                 fallback=None, client=None, vocab_size: int = 128256):
        self.base_url = base_url.rstrip("/")
        self.model = model
        self.timeout_s = timeout_s
        self.fallback = fallback or HashTokenizer(vocab_size)
        self.errors = 0
        if client is not None:
            self._client = client       # injected (tests: TestClient)
        else:
            import httpx
            if uds_path:
                self._client = httpx.Client(
                    transport=httpx.HTTPTransport(uds=uds_path),
                    base_url=self.base_url or "http://tokenizer",
                    timeout=timeout_s)
                self.base_url = ""
            else:
                self._client = httpx.Client(timeout=timeout_s)

    def __call__(self, text: str) -> List[int]:
        try:
            r = self._client.post(self.base_url + "/tokenize",
                                  json={"model": self.model, "prompt": text})
            if r.status_code == 200:
                return list(r.json()["tokens"])
        except Exception:
            pass
        self.errors += 1
        return self.fallback(text)

    def decode(self, token_ids: List[int]) -> str:
        try:
            r = self._client.post(self.base_url + "/detokenize",
                                  json={"model": self.model,
                                        "tokens": list(token_ids)})
            if r.status_code == 200:
                return r.json()["prompt"]
        except Exception:
            pass
        return self.fallback.decode(token_ids)
