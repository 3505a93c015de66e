from .types import FlowControlRequest, QueueOutcome  # noqa: F401
from .controller import FlowController  # noqa: F401
from .registry import FlowRegistry, BandConfig  # noqa: F401
from .saturation import (  # noqa: F401
    ConcurrencySaturationDetector, SaturationDetector,
    UtilizationSaturationDetector,
)
