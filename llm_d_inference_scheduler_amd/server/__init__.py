from .service import NodeService  # noqa: F401
from .openai_app import build_app  # noqa: F401
