"""Director — request-control orchestration
(parity: pkg/epp/requestcontrol/director.go:182-261 HandleRequest,
:274 mutateModel, :317 selectWeightedModel, :347 prepareRequest,
:384/:407 response hooks).

Per request: model rewrite -> objective/priority lookup -> admission ->
candidate location (subset hint) -> DataProducer plugins (DAG-ordered,
400 ms budget, director.go:55) -> Admitter plugins -> Scheduler.schedule ->
prepare (target endpoint + stage headers) -> PreRequest plugins.
"""
import random
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..api.modelrewrite import select_weighted_target
from ..datalayer.datastore import Datastore
from ..datalayer.endpoint import Endpoint
from ..metrics import prom
from ..plugins.interface import (Admitter, DataProducer, PreRequest,
                                 ResponseComplete, ResponseReceived,
                                 ResponseStreaming)
from ..scheduling.scheduler import Scheduler
from ..scheduling.types import (LLMRequest, SchedulingContext,
                                SchedulingResult)
from ..telemetry import get_tracer
from ..utils.logging import get_logger
from .admission import AdmissionController, AdmissionDenied
from .candidates import SUBSET_HINT_HEADER, EndpointCandidates

log = get_logger("director")

DATA_PRODUCER_BUDGET_S = 0.4  # director.go:55
TARGET_ENDPOINT_HEADER = "x-gateway-destination-endpoint"


@dataclass
class RequestControlConfig:
    """Ordered plugin lists per extension point
    (requestcontrol/request_control_config.go)."""
    data_producers: List[DataProducer] = field(default_factory=list)
    admitters: List[Admitter] = field(default_factory=list)
    pre_request: List[PreRequest] = field(default_factory=list)
    response_received: List[ResponseReceived] = field(default_factory=list)
    response_streaming: List[ResponseStreaming] = field(default_factory=list)
    response_complete: List[ResponseComplete] = field(default_factory=list)

    def topo_sort_producers(self) -> None:
        """DAG-order producers by their `requires` lists
        (datalayer/data_graph.go topological sort)."""
        by_type: Dict[str, DataProducer] = {p.type_name: p
                                            for p in self.data_producers}
        seen: Dict[str, int] = {}
        order: List[DataProducer] = []

        def visit(p: DataProducer):
            state = seen.get(p.type_name, 0)
            if state == 1:
                raise ValueError(f"producer cycle at {p.type_name}")
            if state == 2:
                return
            seen[p.type_name] = 1
            for dep in getattr(p, "requires", []):
                if dep in by_type:
                    visit(by_type[dep])
            seen[p.type_name] = 2
            order.append(p)

        for p in self.data_producers:
            visit(p)
        self.data_producers = order


@dataclass
class RoutingDecision:
    request: LLMRequest
    target: Optional[Endpoint]
    result: SchedulingResult
    ctx: SchedulingContext
    epp_latency_ms: float = 0.0

    @property
    def target_header(self) -> str:
        """x-gateway-destination-endpoint value — multi-endpoint picks are
        joined with "," (director.go prepareRequest)."""
        eps = self.result.primary.picks if self.result.primary else []
        return ",".join(ep.metadata.address for ep in eps)


class Director:
    def __init__(self, datastore: Datastore, scheduler: Scheduler,
                 admission: AdmissionController,
                 candidates: EndpointCandidates,
                 config: RequestControlConfig,
                 seed: int = 0xD1CE):
        self.datastore = datastore
        self.scheduler = scheduler
        self.admission = admission
        self.candidates = candidates
        self.config = config
        self.config.topo_sort_producers()
        self._rng = random.Random(seed)
        # serializes produce+schedule+prepare when requests are routed from
        # several threads (flow-control mode); admission itself (which may
        # block in the queue) stays outside the lock.
        self._sched_lock = threading.Lock()

    # ---- request path ----
    def handle_request(self, req: LLMRequest,
                       precomputed: Optional[Dict] = None) -> RoutingDecision:
        """`precomputed`: request-scoped attributes produced ahead of this
        call (the batched gfx950 prefix hash/match over a whole admission
        batch) — producers skip keys already present."""
        t0 = time.monotonic()
        tracer = get_tracer()
        with tracer.span("director.handle_request",
                         request_id=req.request_id, model=req.model) as span:
            # remote context from the W3C traceparent header (the otel
            # propagation the reference wires from Envoy, handlers/request.go)
            from ..telemetry.tracing import parse_traceparent
            tp = parse_traceparent(req.headers.get("traceparent"))
            if tp is not None:
                span.trace_id, parent_span = tp
                span.set_attribute("remote_parent_span_id", parent_span)
            ctx = SchedulingContext(request=req)
            if precomputed:
                ctx.attributes.update(
                    {k: v for k, v in precomputed.items()
                     if not k.startswith("_state:")})
                ctx.state.update(
                    {k[7:]: v for k, v in precomputed.items()
                     if k.startswith("_state:")})
            self._mutate_model(req)
            self._resolve_objective(req)

            subset = req.subset_hint
            if subset is None and SUBSET_HINT_HEADER in req.headers:
                subset = [s.strip() for s in
                          req.headers[SUBSET_HINT_HEADER].split(",")]
            endpoints = self.candidates.locate(subset)
            if not endpoints:
                raise AdmissionDenied("no_endpoints",
                                      "no candidate endpoints", status=503)

            # admission may block (flow control) or shed
            self.admission.admit(ctx, endpoints)

            with self._sched_lock:
                self._run_producers(ctx, endpoints)
                for admitter in self.config.admitters:
                    ok, reason = admitter.admit(ctx, endpoints)
                    if not ok:
                        raise AdmissionDenied("admitter_denied", reason)

                t_sched = time.monotonic()
                result = self.scheduler.schedule(ctx, endpoints)
                prom.scheduler_e2e.observe(time.monotonic() - t_sched)

                target = result.primary.target if result.primary else None
                if target is None:
                    raise AdmissionDenied("no_target",
                                          "scheduler produced no target",
                                          status=503)
                self._prepare_request(ctx, result, target)
            latency_ms = (time.monotonic() - t0) * 1e3
            prom.L(prom.request_total, req.model, req.target_model).inc()
            prom.L(prom.running_requests, req.target_model).inc()
            return RoutingDecision(request=req, target=target, result=result,
                                   ctx=ctx, epp_latency_ms=latency_ms)

    def _mutate_model(self, req: LLMRequest) -> None:
        # a header-forced target (x-gateway-model-name-rewrite) wins over
        # InferenceModelRewrite rules (director.go mutateModelIfNeeded
        # skips when TargetModelName was set from the header)
        if req.target_model != req.model:
            return
        for rw in self.datastore.model_rewrites():
            rule = rw.match(req.model)
            if rule is None:
                continue
            target = select_weighted_target(rule, self._rng)
            if target:
                req.target_model = target
                prom.rewrite_decision_total.labels(req.model, target).inc()
            return

    def _resolve_objective(self, req: LLMRequest) -> None:
        name = req.objective_name or req.headers.get(
            "x-gateway-inference-objective", "")
        if not name:
            return
        obj = self.datastore.get_objective(name)
        if obj is None:
            return
        req.priority = obj.priority
        if req.ttft_slo_ms is None:
            req.ttft_slo_ms = obj.ttft_slo_ms
        if req.tpot_slo_ms is None:
            req.tpot_slo_ms = obj.tpot_slo_ms

    def _run_producers(self, ctx: SchedulingContext,
                       endpoints: List[Endpoint]) -> None:
        deadline = time.monotonic() + DATA_PRODUCER_BUDGET_S
        for producer in self.config.data_producers:
            if time.monotonic() > deadline:
                log.v(3).info("data producer budget exhausted",
                              skipped=producer.name)
                break
            with prom.L(prom.plugin_latency, producer.name).time():
                try:
                    producer.produce(ctx, endpoints)
                except Exception as e:
                    log.error("data producer failed", plugin=producer.name,
                              err=str(e))

    def _prepare_request(self, ctx: SchedulingContext,
                         result: SchedulingResult, target: Endpoint) -> None:
        ctx.request.headers[TARGET_ENDPOINT_HEADER] = \
            ",".join(ep.metadata.address
                     for ep in (result.primary.picks if result.primary else []))
        handler = self.scheduler.config.profile_handler
        if hasattr(handler, "pre_request"):
            handler.pre_request(ctx, result, target)
        for plugin in self.config.pre_request:
            with prom.L(prom.plugin_latency, plugin.name).time():
                plugin.pre_request(ctx, result, target)

    # ---- response path (director.go:384 HandleResponseHeader,
    #      :407-464 per-chunk async queue -> plugins, sync on final) ----
    def handle_response_headers(self, decision: RoutingDecision,
                                headers: Dict[str, str]) -> None:
        for plugin in self.config.response_received:
            plugin.response_received(decision.ctx, decision.target, headers)

    def handle_response_chunk(self, decision: RoutingDecision, chunk) -> None:
        for plugin in self.config.response_streaming:
            plugin.response_streaming(decision.ctx, decision.target, chunk)

    def handle_response_complete(self, decision: RoutingDecision,
                                 usage) -> None:
        req = decision.request
        for plugin in self.config.response_complete:
            plugin.response_complete(decision.ctx, decision.target, usage)
        prom.L(prom.running_requests, req.target_model).dec()
        if usage is not None:
            model = req.model
            if getattr(usage, "prompt_tokens", 0):
                prom.input_tokens.labels(model).observe(usage.prompt_tokens)
            if getattr(usage, "completion_tokens", 0):
                prom.output_tokens.labels(model).observe(usage.completion_tokens)
            if getattr(usage, "cached_tokens", 0):
                prom.cached_tokens.labels(model).observe(usage.cached_tokens)
            ttft_ms = getattr(usage, "ttft_ms", None)
            if ttft_ms:
                prom.ttft.labels(model).observe(ttft_ms / 1e3)
            tpot_ms = getattr(usage, "tpot_ms", None)
            if tpot_ms:
                prom.tpot.labels(model).observe(tpot_ms / 1e3)
            e2e_ms = getattr(usage, "e2e_ms", None)
            if e2e_ms:
                prom.request_duration.labels(model).observe(e2e_ms / 1e3)
