from .registry import PluginRegistry, register_plugin, global_registry  # noqa: F401
from .interface import (  # noqa: F401
    Admitter, DataProducer, Filter, Picker, PreRequest, ProfileHandler,
    ResponseComplete, ResponseReceived, ResponseStreaming, Scorer,
)


def register_all_plugins() -> None:
    """Import every plugin module so their @register_plugin decorators run
    (parity: pkg/epp/framework/plugins/register.go:23-56 +
    cmd/epp/runner/runner.go:463-515 registerInTreePlugins)."""
    from . import filters, scorers, pickers, profile_handlers, producers, admitters  # noqa: F401
    from ..flowcontrol import evictor  # noqa: F401  (eviction policy plugins)
