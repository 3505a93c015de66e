"""Node topology: GPU-role assignment per rank.

The reference's "parallelism inventory"  maps Kubernetes
pod pools onto this node's ranks: N identical decode workers (pool DP),
prefill/decode role split (P/D), encode roles (E/PD, E/P/D). Spec strings:

  "mono"  / "dp"            - every rank decode
  "pd:2p6d"                 - ranks 0-1 prefill, 2-7 decode
  "epd:1e2p5d"              - rank 0 encode, 1-2 prefill, 3-7 decode
  "pd-combined"             - every rank prefill-decode (role label both)
"""
import re
from dataclasses import dataclass, field
from typing import Dict, List

from ..datalayer.endpoint import Role


@dataclass
class RankSpec:
    rank: int
    role: Role
    labels: Dict[str, str] = field(default_factory=dict)

    @property
    def role_label(self) -> str:
        parts = []
        if self.role & Role.ENCODE:
            parts.append("encode")
        if self.role & Role.PREFILL:
            parts.append("prefill")
        if self.role & Role.DECODE:
            parts.append("decode")
        return "-".join(parts) or "decode"


@dataclass
class NodeTopology:
    world_size: int
    ranks: List[RankSpec]

    @staticmethod
    def parse(spec: str, world_size: int) -> "NodeTopology":
        spec = (spec or "mono").strip().lower()
        if spec in ("mono", "dp", "decode"):
            ranks = [RankSpec(r, Role.DECODE) for r in range(world_size)]
            return NodeTopology(world_size, ranks)
        if spec in ("pd-combined", "both"):
            ranks = [RankSpec(r, Role.PREFILL | Role.DECODE)
                     for r in range(world_size)]
            return NodeTopology(world_size, ranks)
        m = re.match(r"^(pd|epd):((?:\d+[epd])+)$", spec)
        if not m:
            raise ValueError(f"bad topology spec {spec!r}")
        counts = {"e": 0, "p": 0, "d": 0}
        for num, kind in re.findall(r"(\d+)([epd])", m.group(2)):
            counts[kind] += int(num)
        total = counts["e"] + counts["p"] + counts["d"]
        if total != world_size:
            raise ValueError(f"topology {spec!r} wants {total} ranks, "
                             f"world_size={world_size}")
        ranks = []
        r = 0
        for _ in range(counts["e"]):
            ranks.append(RankSpec(r, Role.ENCODE)); r += 1
        for _ in range(counts["p"]):
            ranks.append(RankSpec(r, Role.PREFILL)); r += 1
        for _ in range(counts["d"]):
            ranks.append(RankSpec(r, Role.DECODE)); r += 1
        return NodeTopology(world_size, ranks)

    def ranks_with(self, role: Role) -> List[int]:
        return [rs.rank for rs in self.ranks if rs.role & role]
