from .director import Director, RequestControlConfig, RoutingDecision  # noqa: F401
from .admission import (  # noqa: F401
    AdmissionController, AdmissionDenied, FlowControlAdmissionController,
    LegacyAdmissionController,
)
from .candidates import EndpointCandidates  # noqa: F401
