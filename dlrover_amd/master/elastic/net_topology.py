"""Network-topology-aware rank ordering.

Parity target: ref master/elastic_training/net_topology.py:22-82
(NodeTopologyMeta, DpTopologySorter: sort nodes so contiguous ranks share an
access switch). For multi-node MI355X pods the same applies to the RoCE/IB
fabric BETWEEN nodes (intra-node is all-to-all xGMI, ordering-free).
"""

from dataclasses import dataclass
from typing import Dict, List


@dataclass
class NodeTopologyMeta:
    node_rank: int = 0
    asw: str = ""  # access switch
    psw: str = ""  # pod/aggregation switch
    node_group: str = ""  # super-pod id


class DpTopologySorter:
    """Order nodes so that ring neighbors share switches: group by psw, then
    asw, then node rank (ref: DpTopologySorter :55)."""

    def sort(self, metas: Dict[int, NodeTopologyMeta]) -> List[int]:
        return [
            m.node_rank
            for m in sorted(
                metas.values(), key=lambda m: (m.psw, m.asw, m.node_rank)
            )
        ]

    def world_order(
        self, world: Dict[int, int], metas: Dict[int, NodeTopologyMeta]
    ) -> Dict[int, int]:
        """Reorder a comm world {rank: local_world} topologically."""
        known = {r: m for r, m in metas.items() if r in world}
        missing = [r for r in world if r not in known]
        ordered = self.sort(known) + sorted(missing)
        return {r: world[r] for r in ordered}
