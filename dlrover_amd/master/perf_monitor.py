"""Throughput + resource monitor (ref: master/monitor/perf_monitor.py:45-210).

Collects global-step reports and per-node resource stats; computes running
speed (steps/s) over a sliding window — input for straggler/hang diagnosis
and the auto-scaler.
"""

import threading
import time
from collections import deque
from typing import Deque, Dict, Optional, Tuple

from dlrover_amd.common import comm


class PerfMonitor:
    model_info = None

    def __init__(self, window: int = 32):
        self._lock = threading.Lock()
        self._samples: Deque[Tuple[float, int]] = deque(maxlen=window)
        self._resources: Dict[int, comm.ResourceStats] = {}
        self._last_step = 0
        self._last_step_time = 0.0

    def report_global_step(self, step: int, timestamp: float):
        with self._lock:
            self._samples.append((timestamp, step))
            self._last_step = max(self._last_step, step)
            self._last_step_time = timestamp

    def report_model_info(self, info: comm.ModelInfo):
        """Trainer-reported model card (ref: stats/job_collector): used by
        the strategy generator's activation estimate and the dashboard."""
        self.model_info = info

    def report_resource(self, node_id: int, stats: comm.ResourceStats):
        with self._lock:
            self._resources[node_id] = stats

    def running_speed(self) -> float:
        """steps/s over the window (ref: perf_monitor.running_speed :132)."""
        with self._lock:
            if len(self._samples) < 2:
                return 0.0
            (t0, s0), (t1, s1) = self._samples[0], self._samples[-1]
            if t1 <= t0:
                return 0.0
            return (s1 - s0) / (t1 - t0)

    @property
    def completed_global_step(self) -> int:
        with self._lock:
            return self._last_step

    def mark_activity(self):
        """Reset the progress timer (e.g. when a recovery action was just
        issued — the job is legitimately not stepping while it restarts)."""
        with self._lock:
            self._last_step_time = time.time()

    def seconds_since_last_step(self) -> float:
        with self._lock:
            if self._last_step_time == 0.0:
                return 0.0
            return time.time() - self._last_step_time

    def node_resource(self, node_id: int) -> Optional[comm.ResourceStats]:
        with self._lock:
            return self._resources.get(node_id)
