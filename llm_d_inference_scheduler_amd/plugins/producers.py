"""DataProducer plugins
(parity: pkg/epp/framework/plugins/requestcontrol/dataproducer/*).

* token-producer        — tokenizer (reference: vLLM /render HTTP or UDS
                          sidecar; here in-process, the engine and router
                          share one deterministic tokenizer)
* approx-prefix-cache-producer — chained block hashing + longest-prefix
                          match against the native index (C++ PrefixIndex;
                          batched gfx950 kernel path via ops.prefix for the
                          admission queue)
* inflight-load-producer — request/token in-flight counters
* predicted-latency-producer — in-process TTFT/TPOT predictor (reference
                          delegates to an external python sidecar; here an
                          online per-endpoint regressor)
"""
import threading
import time
from typing import Dict, List, Optional

import numpy as np

from .. import _router_core as rc
from ..datalayer.attributes import (IN_FLIGHT_LOAD, LATENCY_PREDICTION_INFO,
                                    InFlightLoad, LatencyPredictionInfo,
                                    PrefixCacheMatchInfo,
                                    PREFIX_CACHE_MATCH_INFO)
from ..datalayer.endpoint import Endpoint
from ..metrics import prom
from ..scheduling.types import SchedulingContext
from ..utils.logging import get_logger
from .interface import DataProducer
from .registry import register_plugin

log = get_logger("plugins.producers")


@register_plugin("token-producer", default_producer_for="tokenized_prompt")
class TokenProducer(DataProducer):
    """Publishes TokenizedPrompt on the request (tokenizer/tokenizer.go:41-80).
    `tokenizer` param: callable(text) -> List[int]; defaults to the shared
    deterministic hash tokenizer (models.tokenizer)."""

    produces = "tokenized_prompt"

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.tokenizer = params.get("tokenizer")
        # remote-tokenizer surface (tokenizer/{vllm_http,uds}.go): mode
        #   inprocess (default) — shared hash tokenizer
        #   http  — POST <url>/tokenize on a vLLM-compatible worker
        #   uds   — same HTTP over a unix socket (the UDS sidecar analog)
        self.mode = params.get("mode", "inprocess")
        if self.tokenizer is None:
            if self.mode in ("http", "uds"):
                from ..models.tokenizer import HttpTokenizer
                self.tokenizer = HttpTokenizer(
                    base_url=params.get("url", ""),
                    uds_path=params.get("udsPath", ""),
                    model=params.get("model", ""),
                    timeout_s=float(params.get("timeoutMs", 400)) / 1e3)
            else:
                from ..models.tokenizer import HashTokenizer
                self.tokenizer = HashTokenizer()

    def produce(self, ctx: SchedulingContext, endpoints) -> None:
        req = ctx.request
        if req.prompt_tokens is None:
            req.prompt_tokens = self.tokenizer(req.flat_text())
        ctx.attributes["tokenized_prompt"] = req.prompt_tokens


@register_plugin("approx-prefix-cache-producer",
                 aliases=["prefix-cache-producer"],
                 default_producer_for=PREFIX_CACHE_MATCH_INFO)
class ApproxPrefixCacheProducer(DataProducer):
    """Chained-hash prefix matching (dataproducer/approximateprefix).

    Defaults re-derived for MI355X (types.go:92-113 did H100-80GB math:
    31,250 LRU entries/server): Llama-3-8B bf16 KV is 128 KB/token; with
    ~240 GB of the 288 GB HBM3E left for KV that is ~1.9M cached tokens =
    ~120K blocks of 16 per GPU — lruCapacityPerServer defaults to 131072
    and auto-tunes from scraped cache_num_blocks (plugin.go:233-245).
    """

    requires = ["token-producer"]
    produces = PREFIX_CACHE_MATCH_INFO

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.block_size = int(params.get("blockSizeTokens", 16))
        self.max_blocks = int(params.get("maxPrefixBlocksToMatch", 256))
        self.lru_capacity = int(params.get("lruCapacityPerServer", 131072))
        self.salt = str(params.get("salt", ""))
        self.index = rc.PrefixIndex(self.lru_capacity)
        self._seeds: Dict[str, int] = {}
        self._auto_tuned = False
        self._lock = threading.Lock()
        self._gpu = None  # ops.prefix.GpuPrefixIndex, attached by the node runner

    def attach_gpu_index(self, gpu_index) -> None:
        self._gpu = gpu_index

    def _seed(self, model: str) -> int:
        s = self._seeds.get(model)
        if s is None:
            s = rc.model_seed(model, self.salt)
            self._seeds[model] = s
        return s

    def _auto_tune(self, endpoints: List[Endpoint]) -> None:
        # adopt the worker-reported cache geometry once (plugin.go:233-245)
        for ep in endpoints:
            m = ep.metrics
            if m.cache_num_blocks > 0:
                if m.cache_block_size > 0:
                    self.block_size = m.cache_block_size
                self.lru_capacity = m.cache_num_blocks
                self.index.set_capacity(self.lru_capacity)
                self._auto_tuned = True
                return

    def hash_request(self, ctx: SchedulingContext) -> np.ndarray:
        req = ctx.request
        cached = ctx.state.get("prefix_hashes")
        if cached is not None:
            return cached
        tokens = np.asarray(req.prompt_tokens or [], dtype=np.int32)
        hashes = rc.hash_tokens(tokens, self.block_size, self.max_blocks,
                                self._seed(req.target_model))
        ctx.state["prefix_hashes"] = hashes
        return hashes

    def produce(self, ctx: SchedulingContext,
                endpoints: List[Endpoint]) -> None:
        if PREFIX_CACHE_MATCH_INFO in ctx.attributes:
            return  # precomputed by the batched gfx950 admission path
        if not self._auto_tuned:
            self._auto_tune(endpoints)
        hashes = self.hash_request(ctx)
        counts = self.index.match_longest(hashes, max(
            (ep.index for ep in endpoints), default=-1) + 1)
        info = PrefixCacheMatchInfo(total_blocks=int(len(hashes)),
                                    block_size_tokens=self.block_size)
        for ep in endpoints:
            info.match_blocks[ep.name] = int(counts[ep.index]) \
                if ep.index < len(counts) else 0
        ctx.attributes[PREFIX_CACHE_MATCH_INFO] = info
        if info.total_blocks:
            best = max(info.match_blocks.values(), default=0)
            prom.prefix_match_ratio.observe(best / info.total_blocks)
        prom.prefix_index_size.set(self.index.size())

    def pre_request(self, ctx: SchedulingContext, result, target) -> None:
        """Record the hashes for every picked endpoint — decode AND
        prefill/encode (plugin.go:164-200 adds for both)."""
        hashes = ctx.state.get("prefix_hashes")
        if hashes is None or not len(hashes):
            return
        for ep in result.all_endpoints():
            self.index.add(ep.index, hashes)
            if self._gpu is not None:
                self._gpu.add(ep.index, hashes)

    def remove_endpoint(self, ep: Endpoint) -> None:
        self.index.remove_endpoint(ep.index)


@register_plugin("inflight-load-producer",
                 default_producer_for=IN_FLIGHT_LOAD)
class InflightLoadProducer(DataProducer):
    """In-flight request/token counters via produce/pre-request/response
    hooks (dataproducer/inflightload)."""

    produces = IN_FLIGHT_LOAD

    def produce(self, ctx, endpoints) -> None:
        for ep in endpoints:
            if ep.get_attribute(IN_FLIGHT_LOAD) is None:
                ep.put_attribute(IN_FLIGHT_LOAD, InFlightLoad())

    def pre_request(self, ctx: SchedulingContext, result, target) -> None:
        if target is None:
            return
        load = target.get_attribute(IN_FLIGHT_LOAD)
        if load is None:
            load = InFlightLoad()
            target.put_attribute(IN_FLIGHT_LOAD, load)
        est = len(ctx.request.prompt_tokens or []) + ctx.request.max_tokens
        ctx.state["inflight_est_tokens"] = est
        load.add(1, est)

    def response_complete(self, ctx: SchedulingContext, target, usage) -> None:
        if target is None:
            return
        load = target.get_attribute(IN_FLIGHT_LOAD)
        if load is not None:
            load.add(-1, -ctx.state.get("inflight_est_tokens", 0))


class _OnlineStat:
    """EMA of a per-unit cost with variance (one per endpoint/metric)."""

    def __init__(self, init: float, alpha: float = 0.1):
        self.value = init
        self.alpha = alpha
        self.samples = 0

    def update(self, x: float) -> None:
        self.value = (1 - self.alpha) * self.value + self.alpha * x
        self.samples += 1


@register_plugin("predicted-latency-producer",
                 default_producer_for=LATENCY_PREDICTION_INFO)
class PredictedLatencyProducer(DataProducer):
    """TTFT/TPOT prediction + SLO headroom (dataproducer/predictedlatency).

    The reference trains XGBoost/LightGBM/BayesianRidge in an external
    python predictor sidecar over HTTP (predictedlatency/training.go,
    latencypredictorclient/). Here the predictor is in-process, same
    two-stage design:
      * cold start: per-endpoint online stats
          TTFT ~= queue_depth * step_ms + non_cached_tokens * per_token_ms
          TPOT ~= base_tpot * (1 + running / batch_scale)
      * trained: a BayesianRidge model per metric over routing-time
        features [queue, running, kv_usage, non_cached_tokens, prompt_len],
        retrained in-process every `retrainEvery` completed samples
        (features captured at PreRequest, labels at response-complete —
        the reference's training-sample collection hooks).
    """

    requires = ["token-producer", "approx-prefix-cache-producer"]
    produces = LATENCY_PREDICTION_INFO

    N_FEATURES = 5

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.default_ttft_slo_ms = float(params.get("ttftSLOms", 2000.0))
        self.default_tpot_slo_ms = float(params.get("tpotSLOms", 100.0))
        self.retrain_every = int(params.get("retrainEvery", 256))
        self.max_samples = int(params.get("maxSamples", 4096))
        self._per_token = {}
        self._step_ms = {}
        self._tpot = {}
        self._lock = threading.Lock()
        # trained-model state
        from collections import deque
        self._pending_feats: Dict[str, tuple] = {}   # req_id -> (ep, feats)
        self._ttft_samples = deque(maxlen=self.max_samples)
        self._tpot_samples = deque(maxlen=self.max_samples)
        self._since_train = 0
        self._ttft_model = None
        self._tpot_model = None

    @staticmethod
    def _features(ep: Endpoint, non_cached: int, n_tokens: int):
        m = ep.metrics
        return [float(m.waiting_queue_size),
                float(m.running_requests_size),
                float(m.kv_cache_usage),
                float(non_cached), float(n_tokens)]

    def _maybe_train(self) -> None:
        if self._since_train < self.retrain_every or \
                len(self._ttft_samples) < 64:
            return
        self._since_train = 0
        try:
            from sklearn.linear_model import BayesianRidge
        except ImportError:           # predictor degrades to online stats
            return
        import numpy as _np
        xs, ys = zip(*self._ttft_samples)
        m = BayesianRidge()
        m.fit(_np.asarray(xs), _np.asarray(ys))
        self._ttft_model = m
        if len(self._tpot_samples) >= 64:
            xs, ys = zip(*self._tpot_samples)
            m2 = BayesianRidge()
            m2.fit(_np.asarray(xs), _np.asarray(ys))
            self._tpot_model = m2

    def pre_request(self, ctx: SchedulingContext, result, target) -> None:
        """Capture routing-time features for training-label pairing."""
        if target is None:
            return
        req = ctx.request
        n_tokens = len(req.prompt_tokens or []) or \
            max(1, req.prompt_len_chars // 4)
        prefix = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        cached = 0
        if prefix is not None:
            cached = prefix.match_blocks.get(target.name, 0) * \
                prefix.block_size_tokens
        feats = self._features(target, max(0, n_tokens - cached), n_tokens)
        with self._lock:
            self._pending_feats[req.request_id] = (target.name, feats)
            if len(self._pending_feats) > 8 * self.max_samples:
                self._pending_feats.clear()    # leak guard on lost requests

    def _stats(self, name: str):
        with self._lock:
            if name not in self._per_token:
                self._per_token[name] = _OnlineStat(0.05)   # ms per prompt token
                self._step_ms[name] = _OnlineStat(20.0)     # ms per queued req
                self._tpot[name] = _OnlineStat(15.0)        # ms per output token
            return (self._per_token[name], self._step_ms[name],
                    self._tpot[name])

    def produce(self, ctx: SchedulingContext,
                endpoints: List[Endpoint]) -> None:
        req = ctx.request
        n_tokens = len(req.prompt_tokens or []) or max(1, req.prompt_len_chars // 4)
        prefix = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        info = LatencyPredictionInfo(
            ttft_slo_ms=req.ttft_slo_ms or self.default_ttft_slo_ms,
            tpot_slo_ms=req.tpot_slo_ms or self.default_tpot_slo_ms)
        ttft_model, tpot_model = self._ttft_model, self._tpot_model
        feats_all = []
        for ep in endpoints:
            cached = 0
            if prefix is not None:
                cached = prefix.match_blocks.get(ep.name, 0) * \
                    prefix.block_size_tokens
            feats_all.append(self._features(ep, max(0, n_tokens - cached),
                                            n_tokens))
        if ttft_model is not None:
            # bulk prediction across candidates (one predictor call per
            # request, as the reference's batched HTTP client does)
            import numpy as _np
            x = _np.asarray(feats_all)
            pred_ttfts = ttft_model.predict(x)
            pred_tpots = (tpot_model.predict(x) if tpot_model is not None
                          else [None] * len(endpoints))
        else:
            pred_ttfts = pred_tpots = [None] * len(endpoints)
        for ep, feats, p_ttft, p_tpot in zip(endpoints, feats_all,
                                             pred_ttfts, pred_tpots):
            per_token, step_ms, tpot = self._stats(ep.name)
            m = ep.metrics
            if p_ttft is None:
                p_ttft = (m.waiting_queue_size * step_ms.value
                          + feats[3] * per_token.value)
            if p_tpot is None:
                p_tpot = tpot.value * (1.0 + m.running_requests_size / 64.0)
            p_ttft = max(0.0, float(p_ttft))
            p_tpot = max(0.0, float(p_tpot))
            info.predicted_ttft_ms[ep.name] = p_ttft
            info.predicted_tpot_ms[ep.name] = p_tpot
            info.ttft_headroom_ms[ep.name] = info.ttft_slo_ms - p_ttft
            info.tpot_headroom_ms[ep.name] = info.tpot_slo_ms - p_tpot
        ctx.attributes[LATENCY_PREDICTION_INFO] = info

    def response_complete(self, ctx: SchedulingContext, target, usage) -> None:
        if target is None or usage is None:
            return
        per_token, step_ms, tpot = self._stats(target.name)
        ttft_ms = getattr(usage, "ttft_ms", None)
        n_prompt = getattr(usage, "prompt_tokens", 0) or 1
        cached = getattr(usage, "cached_tokens", 0)
        if ttft_ms is not None and ttft_ms > 0:
            non_cached = max(1, n_prompt - cached)
            per_token.update(ttft_ms / non_cached)
        tpot_ms = getattr(usage, "tpot_ms", None)
        if tpot_ms is not None and tpot_ms > 0:
            tpot.update(tpot_ms)
        # trained-model sample: pair routing-time features with the label
        with self._lock:
            rec = self._pending_feats.pop(ctx.request.request_id, None)
            if rec is not None:
                _, feats = rec
                if ttft_ms is not None and ttft_ms > 0:
                    self._ttft_samples.append((feats, float(ttft_ms)))
                    self._since_train += 1
                if tpot_ms is not None and tpot_ms > 0:
                    self._tpot_samples.append((feats, float(tpot_ms)))
            self._maybe_train()
