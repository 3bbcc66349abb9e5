"""DiagnosisAction hierarchy (ref: dlrover/python/diagnosis/common/
diagnosis_action.py — NoAction/EventAction/NodeAction/JobAbortAction).

Actions are produced by diagnosticians on the master and delivered to agents
piggy-backed on heartbeat responses (ref: servicer._report_heartbeat :833).
They serialize as (class name, kwargs) so the wire format stays dataclass-free.
"""

import time
from dataclasses import dataclass, field
from typing import Dict, Optional


class DiagnosisActionType:
    NONE = "no_action"
    EVENT = "event"
    RESTART_WORKER = "restart_worker"
    RELAUNCH_WORKER = "relaunch_worker"
    JOB_ABORT = "job_abort"
    # ask every agent to dump its workers' hiptimer kernel-trace rings
    # (ref: py_xpu_timer dump_timeline fan-out across hosts)
    DUMP_TIMELINE = "dump_timeline"


@dataclass
class DiagnosisAction:
    action_type: str = DiagnosisActionType.NONE
    node_id: int = -1
    reason: str = ""
    timestamp: float = field(default_factory=time.time)
    expired_secs: float = 600.0

    def is_expired(self) -> bool:
        return time.time() > self.timestamp + self.expired_secs

    def is_needed(self) -> bool:
        return self.action_type != DiagnosisActionType.NONE and not self.is_expired()

    def to_wire(self):
        return type(self).__name__, {
            "action_type": self.action_type,
            "node_id": self.node_id,
            "reason": self.reason,
            "timestamp": self.timestamp,
        }


@dataclass
class NoAction(DiagnosisAction):
    action_type: str = DiagnosisActionType.NONE


@dataclass
class EventAction(DiagnosisAction):
    action_type: str = DiagnosisActionType.EVENT
    event_type: str = ""
    msg: str = ""


@dataclass
class NodeAction(DiagnosisAction):
    """Restart (same node) or relaunch (new node) the training workers."""

    action_type: str = DiagnosisActionType.RESTART_WORKER
    instance: str = ""


@dataclass
class JobAbortAction(DiagnosisAction):
    action_type: str = DiagnosisActionType.JOB_ABORT


_REGISTRY = {c.__name__: c for c in (DiagnosisAction, NoAction, EventAction, NodeAction, JobAbortAction)}


def action_from_wire(cls_name: str, kwargs: Dict) -> Optional[DiagnosisAction]:
    if not cls_name:
        return None
    cls = _REGISTRY.get(cls_name, DiagnosisAction)
    known = {k: v for k, v in kwargs.items() if k in cls.__dataclass_fields__}
    return cls(**known)
