"""Singleton job state: node registry + per-node diagnosis action queues
(ref: dlrover/python/master/node/job_context.py:44-411)."""

import threading
from typing import Dict, List, Optional

from dlrover_amd.common.constants import JobStage, NodeType
from dlrover_amd.common.node import Node
from dlrover_amd.diagnosis.actions import DiagnosisAction


class JobContext:
    _instance: Optional["JobContext"] = None
    _lock = threading.Lock()

    def __init__(self):
        self._mutex = threading.RLock()
        self._nodes: Dict[str, Dict[int, Node]] = {}
        self._actions: Dict[int, List[DiagnosisAction]] = {}
        self.job_stage = JobStage.INIT
        self.exit_reason = ""
        self.exit_code = 0

    @classmethod
    def singleton_instance(cls) -> "JobContext":
        if cls._instance is None:
            with cls._lock:
                if cls._instance is None:
                    cls._instance = cls()
        return cls._instance

    @classmethod
    def _reset_for_tests(cls):
        with cls._lock:
            cls._instance = None

    # -- nodes -------------------------------------------------------------------

    def update_node(self, node: Node):
        with self._mutex:
            self._nodes.setdefault(node.type, {})[node.id] = node

    def get_node(self, node_type: str, node_id: int) -> Optional[Node]:
        with self._mutex:
            return self._nodes.get(node_type, {}).get(node_id)

    def remove_node(self, node_type: str, node_id: int):
        with self._mutex:
            self._nodes.get(node_type, {}).pop(node_id, None)

    def job_nodes(self, node_type: str = NodeType.WORKER) -> Dict[int, Node]:
        with self._mutex:
            return dict(self._nodes.get(node_type, {}))

    def alive_nodes(self, node_type: str = NodeType.WORKER) -> List[Node]:
        with self._mutex:
            return [n for n in self._nodes.get(node_type, {}).values() if n.is_alive()]

    # -- diagnosis actions ----------------------------------------------------------

    def enqueue_action(self, action: DiagnosisAction):
        with self._mutex:
            if action.node_id == -1:
                # broadcast: fan out one copy per alive node so EVERY agent
                # receives it on its own heartbeat (a single -1 queue would be
                # consumed by whichever agent polls first)
                import copy as _copy

                targets = [n.id for t in self._nodes.values() for n in t.values()
                           if n.is_alive()]
                for nid in targets or [-1]:
                    a = _copy.copy(action)
                    a.node_id = nid
                    self._actions.setdefault(nid, []).append(a)
                return
            self._actions.setdefault(action.node_id, []).append(action)

    def next_action(self, node_id: int) -> Optional[DiagnosisAction]:
        """Pop the next un-expired action for a node (node_id=-1 = broadcast)."""
        with self._mutex:
            for key in (node_id, -1):
                queue = self._actions.get(key, [])
                while queue:
                    action = queue.pop(0)
                    if action.is_needed():
                        return action
            return None

    def request_stop(self, reason: str = "", code: int = 0):
        with self._mutex:
            self.job_stage = JobStage.STOPPING
            self.exit_reason = reason
            self.exit_code = code

    def is_stopping(self) -> bool:
        with self._mutex:
            return self.job_stage in (JobStage.STOPPING, JobStage.STOPPED)
