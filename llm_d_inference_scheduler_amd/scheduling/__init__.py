from .types import (  # noqa: F401
    LLMRequest, ProfileRunResult, SchedulingContext, SchedulingResult,
)
from .scheduler import Scheduler, SchedulerConfig, SchedulerProfile  # noqa: F401
