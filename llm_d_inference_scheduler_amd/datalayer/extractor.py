"""Prometheus-text metrics extractor for external vLLM-compatible workers.

Reference parity: `pkg/epp/backend/metrics/metrics.go` (promToPodMetrics,
getLatestLoraMetric:242) and the configurable metric specs of
`pkg/epp/server/options.go:82-125`; exercised there by `extractor_test.go`.

The in-node fast path reads engine snapshots directly (runtime.CallableSource)
— this module is the slow path that lets a REMOTE worker (a real vLLM server,
or another node's front door, which exposes the same families at /metrics)
join the endpoint pool: scrape its Prometheus text, map the vLLM metric
families onto the `Metrics` snapshot the scheduler consumes.

Default family names (options.go:121-125):
    vllm:num_requests_waiting   gauge  -> waiting_queue_size
    vllm:num_requests_running   gauge  -> running_requests_size
    vllm:kv_cache_usage_perc    gauge  -> kv_cache_usage
    vllm:lora_requests_info     gauge  -> active/waiting_models, max (latest
                                          timestamp-valued series wins)
    vllm:cache_config_info      gauge  -> cache block size / count (labels)
"""
import math
import re
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from .endpoint import Endpoint, Metrics
from .runtime import DataSource
from ..utils.logging import get_logger

log = get_logger("datalayer.extractor")

_SAMPLE_RE = re.compile(
    r'^([a-zA-Z_:][a-zA-Z0-9_:]*)\s*(?:\{(.*)\})?\s+(\S+)(?:\s+(-?\d+))?\s*$')
_LABEL_RE = re.compile(r'([a-zA-Z_][a-zA-Z0-9_]*)="((?:[^"\\]|\\.)*)"')

Sample = Tuple[Dict[str, str], float, Optional[int]]  # labels, value, ts(ms)


def _unescape(v: str) -> str:
    return v.replace(r"\\", "\\").replace(r"\"", '"').replace(r"\n", "\n")


def _parse_value(s: str) -> float:
    low = s.lower()
    if low in ("+inf", "inf"):
        return math.inf
    if low == "-inf":
        return -math.inf
    if low == "nan":
        return math.nan
    return float(s)


def parse_prom_text(text: str) -> Dict[str, List[Sample]]:
    """Parse Prometheus exposition text into family -> samples. Malformed
    lines are skipped, never raised (a half-written scrape must not poison
    the collector; the reference's parser is similarly lenient)."""
    out: Dict[str, List[Sample]] = {}
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        m = _SAMPLE_RE.match(line)
        if not m:
            continue
        name, label_body, value_s, ts_s = m.groups()
        try:
            value = _parse_value(value_s)
        except ValueError:
            continue
        labels = {k: _unescape(v)
                  for k, v in _LABEL_RE.findall(label_body or "")}
        ts = int(ts_s) if ts_s is not None else None
        out.setdefault(name, []).append((labels, value, ts))
    return out


@dataclass
class MetricSpec:
    """`family` or `family{label=value,...}` selector (options.go metric
    spec strings are configurable per deployment)."""
    family: str
    matchers: Dict[str, str] = field(default_factory=dict)

    @classmethod
    def parse(cls, spec: str) -> "MetricSpec":
        spec = spec.strip()
        if "{" in spec and spec.endswith("}"):
            fam, body = spec[:-1].split("{", 1)
            matchers = {}
            for part in body.split(","):
                if "=" in part:
                    k, v = part.split("=", 1)
                    matchers[k.strip()] = v.strip().strip('"')
            return cls(fam.strip(), matchers)
        return cls(spec)

    def select(self, families: Dict[str, List[Sample]]) -> List[Sample]:
        samples = families.get(self.family, [])
        if not self.matchers:
            return samples
        return [s for s in samples
                if all(s[0].get(k) == v for k, v in self.matchers.items())]


@dataclass
class ExtractorSpecs:
    waiting: str = "vllm:num_requests_waiting"
    running: str = "vllm:num_requests_running"
    kv_usage: str = "vllm:kv_cache_usage_perc"
    lora_info: str = "vllm:lora_requests_info"
    cache_info: str = "vllm:cache_config_info"


def _latest(samples: List[Sample]) -> Optional[Sample]:
    """Latest series wins: vLLM emits lora_requests_info with the
    *timestamp as the value* (metrics.go:242-270) — prefer the largest
    value, falling back to exposition timestamps."""
    if not samples:
        return None
    return max(samples, key=lambda s: (s[1] if not math.isnan(s[1]) else
                                       -math.inf, s[2] or 0))


def _adapters(csv: str) -> Dict[str, int]:
    return {a: 0 for a in (x.strip() for x in csv.split(",")) if a}


def extract_metrics(text: str,
                    specs: ExtractorSpecs = ExtractorSpecs()) -> Metrics:
    fams = parse_prom_text(text)
    m = Metrics(update_time=time.time())

    def gauge(spec: str) -> Optional[float]:
        got = MetricSpec.parse(spec).select(fams)
        if not got:
            return None
        v = got[-1][1]
        return None if math.isnan(v) else v

    v = gauge(specs.waiting)
    if v is not None:
        m.waiting_queue_size = int(v)
    v = gauge(specs.running)
    if v is not None:
        m.running_requests_size = int(v)
    v = gauge(specs.kv_usage)
    if v is not None:
        m.kv_cache_usage = float(v)

    lora = _latest(MetricSpec.parse(specs.lora_info).select(fams))
    if lora is not None:
        labels = lora[0]
        m.active_models = _adapters(labels.get("running_lora_adapters", ""))
        m.waiting_models = _adapters(labels.get("waiting_lora_adapters", ""))
        try:
            m.max_active_models = int(float(labels.get("max_lora", "0")))
        except ValueError:
            pass

    cache = MetricSpec.parse(specs.cache_info).select(fams)
    if cache:
        labels = cache[-1][0]
        for key, attr in (("block_size", "cache_block_size"),
                          ("num_gpu_blocks", "cache_num_blocks")):
            try:
                setattr(m, attr, int(float(labels[key])))
            except (KeyError, ValueError):
                pass
    return m


class HttpMetricsSource(DataSource):
    """Scrape-based source for remote endpoints (the reference's
    PodMetricsClient shape). `fetcher(url) -> text` is injectable for
    tests; the default uses httpx with the reference's 1 s collection
    timeout (collector.go). URL: the endpoint's `metrics_url` label, else
    http://<address>/metrics."""

    def __init__(self, fetcher: Optional[Callable[[str], str]] = None,
                 specs: ExtractorSpecs = ExtractorSpecs(),
                 timeout_s: float = 1.0):
        self.specs = specs
        self.timeout_s = timeout_s
        self._fetcher = fetcher
        self._client = None

    def _fetch(self, url: str) -> str:
        if self._fetcher is not None:
            return self._fetcher(url)
        if self._client is None:
            import httpx
            self._client = httpx.Client(timeout=self.timeout_s)
        r = self._client.get(url)
        r.raise_for_status()
        return r.text

    def url_for(self, endpoint: Endpoint) -> str:
        url = endpoint.metadata.labels.get("metrics_url")
        if url:
            return url
        return f"http://{endpoint.metadata.address}/metrics"

    def collect(self, endpoint: Endpoint) -> Optional[Metrics]:
        from ..metrics import prom
        try:
            text = self._fetch(self.url_for(endpoint))
        except Exception as e:
            log.v(4).info("scrape failed", endpoint=endpoint.name,
                          err=str(e))
            prom.datalayer_poll_errors.labels("HttpMetricsSource").inc()
            return None   # stale metrics -> saturation detector handles it
        try:
            return extract_metrics(text, self.specs)
        except Exception as e:   # the parser is lenient; this is a backstop
            log.v(4).info("extract failed", endpoint=endpoint.name,
                          err=str(e))
            prom.datalayer_extract_errors.labels(
                "HttpMetricsSource", "vllm").inc()
            return None
