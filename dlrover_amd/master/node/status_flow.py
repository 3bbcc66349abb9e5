"""Allowed node status transitions (ref: master/node/status_flow.py:1-164)."""

from dlrover_amd.common.constants import NodeStatus

# from -> set of allowed targets
_FLOW = {
    NodeStatus.INITIAL: {
        NodeStatus.PENDING,
        NodeStatus.RUNNING,
        NodeStatus.FAILED,
        NodeStatus.DELETED,
        NodeStatus.SUCCEEDED,
    },
    NodeStatus.PENDING: {
        NodeStatus.RUNNING,
        NodeStatus.FAILED,
        NodeStatus.DELETED,
        NodeStatus.SUCCEEDED,
    },
    NodeStatus.RUNNING: {
        NodeStatus.SUCCEEDED,
        NodeStatus.FAILED,
        NodeStatus.DELETED,
        NodeStatus.BREAKDOWN,
    },
    NodeStatus.SUCCEEDED: {NodeStatus.DELETED},
    NodeStatus.FAILED: {NodeStatus.DELETED},
    NodeStatus.BREAKDOWN: {NodeStatus.DELETED, NodeStatus.FAILED},
    NodeStatus.DELETED: set(),
    NodeStatus.UNKNOWN: set(NodeStatus.terminal())
    | {NodeStatus.PENDING, NodeStatus.RUNNING},
}


def allowed_transition(src: str, dst: str) -> bool:
    if src == dst:
        return False
    return dst in _FLOW.get(src, set())
