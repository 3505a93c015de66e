"""Control-plane mailbox: per-step message exchange across ranks.

Replaces the reference's HTTP/JSON + ZMQ control channels
with one `all_gather_object` over a dedicated gloo process group per engine
step: every rank publishes its outbox (assignments from the router rank,
prefill-done / completion notices from workers) and receives everyone
else's. Message volume is tiny (queue metadata, no payloads — KV bytes ride
RCCL over xGMI, see transfer.py), so a CPU-side gloo collective keeps the
control plane off the compute streams entirely.
"""
from typing import Any, Dict, List, Optional

import torch.distributed as dist


class Mailbox:
    def __init__(self, group: Optional[object] = None, rank: int = 0,
                 world_size: int = 1):
        self.group = group
        self.rank = rank
        self.world_size = world_size

    def exchange(self, outbox: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        """Returns the concatenation of every rank's outbox, rank order."""
        if self.world_size == 1:
            return list(outbox)
        gathered: List[Any] = [None] * self.world_size
        dist.all_gather_object(gathered, outbox, group=self.group)
        merged: List[Dict[str, Any]] = []
        for msgs in gathered:
            merged.extend(msgs or [])
        return merged

    def for_me(self, msgs: List[Dict[str, Any]],
               key: str = "dst") -> List[Dict[str, Any]]:
        return [m for m in msgs if m.get(key) == self.rank]
