"""Scheduling types (parity: pkg/epp/scheduling/types + requesthandling/types.go:65
unified InferenceRequestBody)."""
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..datalayer.endpoint import Endpoint


@dataclass
class MultiModalItem:
    """One multimodal content item (image_url / video_url / input_audio)."""
    kind: str
    url: str


@dataclass
class LLMRequest:
    """Unified request body + routing metadata.

    Parity: InferenceRequestBody (requesthandling/types.go:65) carrying
    /v1/completions | /v1/chat/completions | /v1/responses | /v1/embeddings
    payloads, plus the reference's per-request headers
    (pkg/epp/metadata/consts.go): fairness id, objective name, subset hint.
    """
    request_id: str
    model: str
    prompt: str = ""
    messages: List[Dict[str, Any]] = field(default_factory=list)
    prompt_tokens: Optional[List[int]] = None      # set by token-producer
    max_tokens: int = 16
    temperature: float = 0.0
    stop_token_ids: Optional[List[int]] = None
    streaming: bool = False
    is_embedding: bool = False
    mm_items: List[MultiModalItem] = field(default_factory=list)

    # routing metadata
    headers: Dict[str, str] = field(default_factory=dict)
    fairness_id: str = ""
    objective_name: str = ""
    priority: int = 0
    subset_hint: Optional[List[str]] = None        # endpoint-name allowlist
    target_model: str = ""                         # post-rewrite model
    session_id: str = ""
    arrival_ns: int = 0
    ttft_slo_ms: Optional[float] = None
    tpot_slo_ms: Optional[float] = None
    # raw body for repackaging (director.go:289)
    raw_body: Optional[Dict[str, Any]] = None

    def __post_init__(self):
        if not self.arrival_ns:
            self.arrival_ns = time.monotonic_ns()
        if not self.target_model:
            self.target_model = self.model

    @property
    def prompt_len_chars(self) -> int:
        if self.prompt:
            return len(self.prompt)
        return sum(len(str(m.get("content", ""))) for m in self.messages)

    def flat_text(self) -> str:
        if self.prompt:
            return self.prompt
        return "\n".join(str(m.get("content", "")) for m in self.messages)


@dataclass
class SchedulingContext:
    """Per-scheduling-cycle state handed to plugins (cycle state analog)."""
    request: LLMRequest
    state: Dict[str, Any] = field(default_factory=dict)   # plugin scratch
    attributes: Dict[str, Any] = field(default_factory=dict)  # request-scoped produced data


@dataclass
class ProfileRunResult:
    """Result of one SchedulerProfile run (scheduler_profile.go)."""
    profile_name: str
    picks: List[Endpoint] = field(default_factory=list)
    scores: Dict[str, float] = field(default_factory=dict)  # endpoint name -> score

    @property
    def target(self) -> Optional[Endpoint]:
        return self.picks[0] if self.picks else None


@dataclass
class SchedulingResult:
    """Aggregate over profiles (scheduler.go:54-102 / ProcessResults)."""
    profile_results: Dict[str, ProfileRunResult] = field(default_factory=dict)
    primary_profile: str = ""

    @property
    def primary(self) -> Optional[ProfileRunResult]:
        return self.profile_results.get(self.primary_profile)

    def all_endpoints(self) -> List[Endpoint]:
        out, seen = [], set()
        for r in self.profile_results.values():
            for ep in r.picks:
                if ep.name not in seen:
                    seen.add(ep.name)
                    out.append(ep)
        return out
