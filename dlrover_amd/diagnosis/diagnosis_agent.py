"""Worker-side failure diagnosis.

Parity target: ref dlrover/python/elastic_agent/diagnosis/diagnosis_agent.py
:67-330 (diagnose_training_failure: parse worker error logs -> restart vs
relaunch vs abort) + diagnosis/datacollector/training_log_collector.py.

Classification ladder (ref: the reference's error-code tables in
common/constants.py + diagnose_training_failure):
  - software errors (Python tracebacks, assertion, shape errors) -> RESTART
    workers in place: same node is fine;
  - hardware/GPU errors (HIP/HSA failures, RCCL unhandled system error,
    ECC/uncorrectable, device lost) -> RELAUNCH on a new node;
  - user errors that restart can't fix (OOM with fixed batch, module import
    errors on every incarnation) -> ABORT after max_restarts.
"""

import re
from dataclasses import dataclass
from typing import List

from dlrover_amd.common.log import logger

RESTART_WORKER = "restart"
RELAUNCH_NODE = "relaunch"
ABORT_JOB = "abort"

# MI355X/ROCm hardware signatures (the reference keys on CUDA/XID analogs)
_HARDWARE_PATTERNS = [
    r"HSA_STATUS_ERROR",
    r"hipErrorUnknown",
    r"hipError(IllegalAddress|HardwareStackError|LaunchFailure)",
    r"memory access fault",
    r"uncorrectable .*error",
    r"GPU is lost",
    r"ras .*error",
    r"unhandled system error.*(nccl|rccl)",
    r"amdgpu.*(hang|reset|timeout)",
]

_FATAL_USER_PATTERNS = [
    r"ModuleNotFoundError",
    r"ImportError",
    r"SyntaxError",
]

# OOM is RELAUNCH, not abort: the master grows the replacement pod's memory
# request (job_manager OOM recovery); repeated OOMs exhaust the relaunch
# budget and only then abort
_OOM_PATTERNS = [
    r"torch\.OutOfMemoryError",
    r"(HIP|CUDA) out of memory",
    r"hipErrorOutOfMemory",
]


@dataclass
class TrainingLog:
    """Collected tail of a worker's log (ref: training_log_collector.py)."""

    node_rank: int = -1
    lines: List[str] = None

    def text(self) -> str:
        return "\n".join(self.lines or [])


class TrainingLogCollector:
    def __init__(self, log_file: str = "", n_lines: int = 200):
        self.log_file = log_file
        self.n_lines = n_lines

    def collect(self) -> TrainingLog:
        lines: List[str] = []
        if self.log_file:
            try:
                with open(self.log_file, errors="replace") as f:
                    lines = f.readlines()[-self.n_lines :]
            except OSError:
                pass
        return TrainingLog(lines=[l.rstrip("\n") for l in lines])


def classify_error(error_text: str) -> str:
    """-> RESTART_WORKER | RELAUNCH_NODE | ABORT_JOB."""
    for pat in _HARDWARE_PATTERNS:
        if re.search(pat, error_text, re.IGNORECASE):
            return RELAUNCH_NODE
    for pat in _OOM_PATTERNS:
        if re.search(pat, error_text):
            return RELAUNCH_NODE
    for pat in _FATAL_USER_PATTERNS:
        if re.search(pat, error_text):
            return ABORT_JOB
    return RESTART_WORKER


def _load_failover_extension():
    """User-pluggable failover strategy (ref: torch/dynamic_failover.py
    DynamicAgentFailoverExtension): DLROVER_FAILOVER_EXTENSION="pkg.mod:Cls"
    names a class whose get_user_failover_strategy(failure_text,
    restart_count) returns one of "restart"/"relaunch"/"abort" or None/"" to
    defer to the built-in ladder."""
    import importlib
    import os

    spec = os.getenv("DLROVER_FAILOVER_EXTENSION", "")
    if not spec or ":" not in spec:
        return None
    mod_name, _, cls_name = spec.partition(":")
    try:
        cls = getattr(importlib.import_module(mod_name), cls_name)
        return cls()
    except Exception:  # noqa: BLE001 — a broken extension must not break failover
        logger.exception("failover extension %s failed to load", spec)
        return None


class WorkerDiagnosisAgent:
    """Used by the elastic agent on worker failure (ref: diagnosis_agent
    .diagnose_training_failure :153)."""

    def __init__(self, client=None, log_file: str = ""):
        self.client = client
        self.collector = TrainingLogCollector(log_file)
        self.extension = _load_failover_extension()

    def diagnose_training_failure(
        self, failure_text: str, restart_count: int, max_restarts: int
    ) -> str:
        log = self.collector.collect()
        combined = f"{failure_text}\n{log.text()}"
        if self.extension is not None:
            try:
                user = self.extension.get_user_failover_strategy(
                    combined, restart_count
                )
                if user in (RESTART_WORKER, RELAUNCH_NODE, ABORT_JOB):
                    logger.info("failover extension verdict: %s", user)
                    return user
            except Exception:  # noqa: BLE001
                logger.exception("failover extension errored — using ladder")
        verdict = classify_error(combined)
        if verdict == RESTART_WORKER and restart_count >= max_restarts:
            verdict = RELAUNCH_NODE
        logger.info(
            "failure diagnosis: %s (restarts %s/%s)",
            verdict,
            restart_count,
            max_restarts,
        )
        if self.client is not None:
            try:
                self.client.report_diagnosis_data(
                    "failure", combined[-4096:], node_rank=-1
                )
            except Exception:  # noqa: BLE001
                pass
        return verdict
