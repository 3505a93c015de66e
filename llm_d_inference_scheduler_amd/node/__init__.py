from .runner import NodeConfig, NodeRunner  # noqa: F401
