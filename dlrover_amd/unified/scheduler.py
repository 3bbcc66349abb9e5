"""Placement scheduling for the unified architecture.

Parity target: ref dlrover/python/unified/schedule/scheduler.py:37-253 +
schedule/graph.py:269 — the reference packs workload actors into Ray
placement groups honoring collocation ("these roles share a node/GPU") and
per-node resource capacity. Ours schedules the SAME abstractions onto a
simulated (or described) node pool; the local process backend applies the
assignment via env (NODE_ID / device index), and a k8s backend can map
bundles to pods 1:1.
"""

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from dlrover_amd.common.log import logger


@dataclass
class NodeSpec:
    """One schedulable node of the pool."""

    name: str
    gpus: int = 8
    cpus: float = 64.0

    def clone(self) -> "NodeSpec":
        return NodeSpec(self.name, self.gpus, self.cpus)


@dataclass
class Bundle:
    """One placement bundle: vertices that MUST land on the same node
    (a collocation group slice, or a single vertex)."""

    vertices: List[object] = field(default_factory=list)
    gpus: int = 0
    cpus: float = 0.0
    node: Optional[str] = None


@dataclass
class Placement:
    bundles: List[Bundle] = field(default_factory=list)
    # vertex name -> (node name, device index)
    assignments: Dict[str, tuple] = field(default_factory=dict)


class PlacementError(RuntimeError):
    pass


class Scheduler:
    """Bin-pack execution-graph vertices into node-sized bundles.

    Collocation (DLJobBuilder.with_collocation) means: the i-th worker of
    every collocated role shares a bundle (the reference's per-group
    placement, scheduler.py:37). Non-collocated roles get one bundle per
    vertex. Bundles are then first-fit packed onto the node pool.
    """

    def __init__(self, nodes: List[NodeSpec]):
        if not nodes:
            raise PlacementError("empty node pool")
        self.nodes = [n.clone() for n in nodes]

    def schedule(self, graph) -> Placement:
        job = graph.job
        coll_groups: List[List[str]] = list(getattr(job, "collocations", []))
        in_group = {r for grp in coll_groups for r in grp}
        bundles: List[Bundle] = []

        for grp in coll_groups:
            roles = [job.roles[r] for r in grp if r in job.roles]
            if not roles:
                continue
            n = max(r.total for r in roles)
            for i in range(n):
                b = Bundle()
                for role in roles:
                    if i < role.total:
                        vs = [v for v in graph.by_role(role.name) if v.rank == i]
                        b.vertices.extend(vs)
                        b.gpus += int(role.resource.get("gpu", 0)) * len(vs)
                        b.cpus += float(role.resource.get("cpu", 1)) * len(vs)
                bundles.append(b)

        for role in job.roles.values():
            if role.name in in_group:
                continue
            for v in graph.by_role(role.name):
                bundles.append(
                    Bundle(
                        vertices=[v],
                        gpus=int(role.resource.get("gpu", 0)),
                        cpus=float(role.resource.get("cpu", 1)),
                    )
                )

        # first-fit decreasing by gpu then cpu (stable, deterministic)
        order = sorted(
            range(len(bundles)),
            key=lambda i: (-bundles[i].gpus, -bundles[i].cpus),
        )
        free = {n.name: n for n in self.nodes}
        placement = Placement(bundles=bundles)
        for i in order:
            b = bundles[i]
            chosen = None
            for n in self.nodes:  # stable order
                f = free[n.name]
                if f.gpus >= b.gpus and f.cpus >= b.cpus:
                    chosen = f
                    break
            if chosen is None:
                raise PlacementError(
                    f"bundle needs gpu={b.gpus} cpu={b.cpus}; no node fits "
                    f"(pool: {[(n.name, free[n.name].gpus) for n in self.nodes]})"
                )
            chosen.gpus -= b.gpus
            chosen.cpus -= b.cpus
            b.node = chosen.name
            gpu_cursor = chosen.gpus
            for v in b.vertices:
                placement.assignments[v.name] = (chosen.name, gpu_cursor)
                gpu_cursor += 1 if b.gpus else 0
        logger.info(
            "scheduled %s bundles onto %s nodes", len(bundles), len(self.nodes)
        )
        return placement
