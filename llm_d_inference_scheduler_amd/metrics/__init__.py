from . import prom  # noqa: F401
