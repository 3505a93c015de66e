from .kvcache import BlockManager, KVPool  # noqa: F401
from .worker import EngineRequest, EngineWorker, RequestOutput  # noqa: F401
