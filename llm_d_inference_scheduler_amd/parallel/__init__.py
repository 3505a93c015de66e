from .topology import NodeTopology, RankSpec  # noqa: F401
from .mailbox import Mailbox  # noqa: F401
from .transfer import KVTransferEngine  # noqa: F401
