"""Plugin extension points
(parity: pkg/epp/framework/interface/scheduling/plugins.go:43-78 —
Filter/Scorer/Picker/ProfileHandler — and
interface/requestcontrol/plugins.go:36-82 — PreRequest/DataProducer/Admitter,
plus the response hooks driven by the director's async queue).
"""
from typing import Dict, List, Optional, Tuple

from ..datalayer.endpoint import Endpoint
from ..scheduling.types import LLMRequest, ProfileRunResult, SchedulingContext


class Plugin:
    type_name: str = "plugin"

    def __init__(self, name: str = "", **params):
        self.name = name or self.type_name
        self.params = params


class Filter(Plugin):
    def filter(self, ctx: SchedulingContext,
               endpoints: List[Endpoint]) -> List[Endpoint]:
        raise NotImplementedError


class Scorer(Plugin):
    # returns endpoint name -> score in [0,1] (clamped by the core)
    def score(self, ctx: SchedulingContext,
              endpoints: List[Endpoint]) -> Dict[str, float]:
        raise NotImplementedError

    # native fast path: return (ScorerKind, param) when the C++ ProfileRunner
    # implements this scorer's formula; None keeps the python path.
    def native_spec(self) -> Optional[Tuple[int, float]]:
        return None


class Picker(Plugin):
    def pick(self, ctx: SchedulingContext, scored: Dict[str, float],
             endpoints: List[Endpoint], max_endpoints: int) -> List[Endpoint]:
        raise NotImplementedError


class ProfileHandler(Plugin):
    """Multi-pass orchestration: decides which profiles run next given
    results so far, and folds profile results into the scheduling result
    (disagg_profile_handler.go:246-354)."""

    def pick_profiles(self, ctx: SchedulingContext,
                      profiles: Dict[str, "SchedulerProfile"],  # noqa: F821
                      results: Dict[str, ProfileRunResult]) -> List[str]:
        raise NotImplementedError

    def process_results(self, ctx: SchedulingContext,
                        results: Dict[str, ProfileRunResult]) -> str:
        """Returns the primary profile name."""
        raise NotImplementedError


class DataProducer(Plugin):
    """Produces request-scoped data before scheduling (400 ms budget,
    director.go:55). `requires` lists producer type-names this one consumes
    (DAG-ordered, datalayer/data_graph.go)."""
    requires: List[str] = []
    produces: str = ""

    def produce(self, ctx: SchedulingContext,
                endpoints: List[Endpoint]) -> None:
        raise NotImplementedError


class Admitter(Plugin):
    def admit(self, ctx: SchedulingContext,
              endpoints: List[Endpoint]) -> Tuple[bool, str]:
        """Returns (admitted, reason)."""
        raise NotImplementedError


class PreRequest(Plugin):
    def pre_request(self, ctx: SchedulingContext, result,
                    target: Endpoint) -> None:
        raise NotImplementedError


class ResponseReceived(Plugin):
    def response_received(self, ctx: SchedulingContext, target: Endpoint,
                          headers: Dict[str, str]) -> None:
        raise NotImplementedError


class ResponseStreaming(Plugin):
    def response_streaming(self, ctx: SchedulingContext, target: Endpoint,
                           chunk) -> None:
        raise NotImplementedError


class ResponseComplete(Plugin):
    def response_complete(self, ctx: SchedulingContext, target: Endpoint,
                          usage) -> None:
        raise NotImplementedError
