from .objectives import InferenceObjective  # noqa: F401
from .modelrewrite import InferenceModelRewrite, RewriteRule, RewriteTarget  # noqa: F401
from .endpointpicker_config import (  # noqa: F401
    EndpointPickerConfig, PluginSpec, ProfilePluginRef, SchedulingProfileSpec,
)
