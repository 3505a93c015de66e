from .tracing import Span, get_tracer, init_tracing  # noqa: F401
