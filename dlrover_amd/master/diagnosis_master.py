"""Master-side diagnosis: observe -> resolve -> action loop.

Parity target: ref dlrover/python/master/diagnosis/diagnosis_master.py +
diagnosis/diagnostician/training_hang.py:61-300. Two hang signals:
  (1) no global-step progress for > hang_downtime while the job reported
      at least one step (PerfMonitor);
  (2) every worker's profiler hang metric (our hiptimer equivalent of
      XPU_TIMER_COMMON_HANG) held 1 for > hang_downtime — workers report it
      via DiagnosisReportData.
Resolution: broadcast RESTART_WORKER once, JOB_ABORT if the hang repeats.
"""

import json
import threading
import time
from collections import deque
from typing import Deque, Dict

from dlrover_amd.common import comm
from dlrover_amd.common.global_context import Context
from dlrover_amd.common.log import logger
from dlrover_amd.diagnosis.actions import (
    DiagnosisActionType,
    JobAbortAction,
    NodeAction,
)
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.perf_monitor import PerfMonitor


class DiagnosisDataManager:
    """Ring buffer of reported diagnosis payloads (ref:
    diagnosis_data_manager.py)."""

    def __init__(self, maxlen: int = 1024):
        self._lock = threading.Lock()
        self._data: Deque[comm.DiagnosisReportData] = deque(maxlen=maxlen)

    def store(self, data: comm.DiagnosisReportData):
        with self._lock:
            self._data.append(data)

    def latest_by_node(self, data_cls: str) -> Dict[int, comm.DiagnosisReportData]:
        with self._lock:
            out: Dict[int, comm.DiagnosisReportData] = {}
            for d in self._data:
                if d.data_cls == data_cls:
                    out[d.node_id] = d
            return out


class DiagnosisMaster:
    HANG_METRIC = "hang"  # data_cls reported by the hiptimer agent collector

    def __init__(self, perf_monitor: PerfMonitor, job_context: JobContext = None):
        self.perf = perf_monitor
        self.ctx = job_context or JobContext.singleton_instance()
        self.data = DiagnosisDataManager()
        self._config = Context.singleton_instance()
        self._stop = threading.Event()
        self._thread = None
        self._hang_restarts = 0
        self._last_hang_action = 0.0

    def collect_data(self, data: comm.DiagnosisReportData):
        self.data.store(data)

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="diagnosis", daemon=True
        )
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=3)

    # -- observe/resolve -------------------------------------------------------

    def _loop(self):
        while not self._stop.wait(10.0):
            try:
                self._check_hang()
            except Exception:  # noqa: BLE001
                logger.exception("diagnosis loop error")

    def _is_metric_hang(self) -> bool:
        """All workers' hang metric == 1 continuously for hang_downtime
        (ref: training_hang.py:160-234)."""
        reports = self.data.latest_by_node(self.HANG_METRIC)
        if not reports:
            return False
        now = time.time()
        for rep in reports.values():
            try:
                payload = json.loads(rep.data_content)
            except (ValueError, TypeError):
                return False
            if not payload.get("hang"):
                return False
            if now - payload.get("since", now) < self._config.hang_downtime:
                return False
        return True

    def _is_step_hang(self) -> bool:
        if self.perf.completed_global_step <= 0:
            return False
        return self.perf.seconds_since_last_step() > self._config.hang_downtime

    def _check_hang(self):
        if not (self._is_step_hang() or self._is_metric_hang()):
            return
        now = time.time()
        # give the restart time to re-rendezvous AND produce a first step
        if now - self._last_hang_action < max(3 * self._config.hang_downtime, 60):
            return
        self._last_hang_action = now
        self._hang_restarts += 1
        if self._hang_restarts > 2:
            logger.error("training hang persists after restarts — aborting job")
            self.ctx.enqueue_action(JobAbortAction(node_id=-1, reason="hang"))
            self.ctx.request_stop("HangError", code=1)
            return
        logger.warning(
            "training hang detected (restart %s): broadcasting worker restart",
            self._hang_restarts,
        )
        self.ctx.enqueue_action(
            NodeAction(
                action_type=DiagnosisActionType.RESTART_WORKER,
                node_id=-1,
                reason="training hang",
            )
        )
        # restarting IS activity: don't re-diagnose the same stall
        self.perf.mark_activity()
