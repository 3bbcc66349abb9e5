"""Node model: resources + lifecycle state of one training node.

Parity target: ref dlrover/python/common/node.py:44-460 (NodeResource,
NodeGroupResource, Node). Rebuilt fresh; MI355X-specific defaults (8 GPUs,
288 GB HBM per GPU) live in constants.GpuConstant.
"""

import copy
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from dlrover_amd.common.constants import (
    NodeExitReason,
    NodeStatus,
    NodeType,
)


@dataclass
class NodeResource:
    cpu: float = 0.0
    memory_mb: int = 0
    gpu_num: int = 0
    gpu_type: str = ""
    priority: str = ""

    @classmethod
    def from_dict(cls, d: Dict) -> "NodeResource":
        return cls(
            cpu=float(d.get("cpu", 0)),
            memory_mb=int(d.get("memory_mb", d.get("memory", 0))),
            gpu_num=int(d.get("gpu_num", d.get("gpu", 0))),
            gpu_type=d.get("gpu_type", ""),
            priority=d.get("priority", ""),
        )

    def to_dict(self) -> Dict:
        return {
            "cpu": self.cpu,
            "memory_mb": self.memory_mb,
            "gpu_num": self.gpu_num,
            "gpu_type": self.gpu_type,
        }


@dataclass
class NodeGroupResource:
    count: int = 0
    node_resource: NodeResource = field(default_factory=NodeResource)


class Node:
    """One node (pod / local process group) in the job."""

    def __init__(
        self,
        node_type: str,
        node_id: int,
        rank_index: Optional[int] = None,
        name: str = "",
        status: str = NodeStatus.INITIAL,
        config_resource: Optional[NodeResource] = None,
        max_relaunch_count: int = 3,
        service_addr: str = "",
    ):
        self.type = node_type
        self.id = node_id
        self.rank_index = rank_index if rank_index is not None else node_id
        self.name = name or f"{node_type}-{node_id}"
        self.status = status
        self.config_resource = config_resource or NodeResource()
        self.used_resource = NodeResource()
        self.service_addr = service_addr
        self.host_ip = ""

        self.create_time: Optional[float] = None
        self.start_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        self.heartbeat_time: float = 0.0

        self.relaunch_count = 0
        self.max_relaunch_count = max_relaunch_count
        self.relaunchable = True
        self.is_released = False
        self.exit_reason = ""
        self.eliminated = False  # judged fault/straggler by node check
        self.start_hang = False
        self.hang_time: float = 0.0
        self.reported_status = ""
        self.restart_training = False
        self.group: Optional[int] = None  # super-pod / node-group id
        self.paral_config = None

    # -- status transitions -------------------------------------------------

    def update_status(self, status: str):
        if status == self.status:
            return False
        self.status = status
        now = time.time()
        if status == NodeStatus.RUNNING and self.start_time is None:
            self.start_time = now
        if status in NodeStatus.terminal():
            self.finish_time = now
        return True

    def update_heartbeat(self, ts: Optional[float] = None):
        self.heartbeat_time = ts if ts is not None else time.time()

    # -- relaunch policy -----------------------------------------------------

    def inc_relaunch_count(self):
        self.relaunch_count += 1

    def exceeded_max_relaunch(self) -> bool:
        return self.relaunch_count >= self.max_relaunch_count

    def should_relaunch(self) -> bool:
        """Node-level relaunch decision (ref: dist_job_manager._should_relaunch
        :1083): relaunch on node errors, within budget, unless eliminated."""
        if self.eliminated or not self.relaunchable:
            return False
        if self.exceeded_max_relaunch():
            return False
        if self.exit_reason == NodeExitReason.FATAL_ERROR:
            return False
        return True

    def is_unrecoverable_failure(self) -> bool:
        return (
            self.exit_reason == NodeExitReason.FATAL_ERROR
            or self.exceeded_max_relaunch()
        )

    def new_incarnation(self, new_id: int) -> "Node":
        """Build the replacement Node after a relaunch decision."""
        n = Node(
            self.type,
            new_id,
            rank_index=self.rank_index,
            status=NodeStatus.INITIAL,
            config_resource=copy.deepcopy(self.config_resource),
            max_relaunch_count=self.max_relaunch_count,
        )
        n.relaunch_count = self.relaunch_count + 1
        n.group = self.group
        return n

    def is_alive(self) -> bool:
        return self.status in (NodeStatus.PENDING, NodeStatus.RUNNING, NodeStatus.INITIAL)

    def __repr__(self):
        return (
            f"Node({self.type}-{self.id} rank={self.rank_index} "
            f"status={self.status} relaunch={self.relaunch_count})"
        )


def new_worker(node_id: int, rank: Optional[int] = None, **kw) -> Node:
    return Node(NodeType.WORKER, node_id, rank_index=rank, **kw)
