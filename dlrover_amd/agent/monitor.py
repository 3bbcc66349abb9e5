"""Agent-side monitors.

Parity target: ref dlrover/python/elastic_agent/monitor/resource.py:219-330
(ResourceMonitor: psutil + GPU stats -> report_used_resource) and
training.py:40-140 (TorchTrainingMonitor: read the ElasticTrainer step file,
report global step). MI355X: GPU utilization/memory via torch.cuda (HIP) and
rocm-smi when available.
"""

import json
import os
import threading
import time
from typing import List, Optional

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.common import comm
from dlrover_amd.common.log import logger
from dlrover_amd.trainer.elastic.trainer import STEP_FILE_DIR


def collect_gpu_stats() -> List[dict]:
    """Per-GPU used/total memory + utilization. Torch first (always present
    on ROCm), rocm-smi as enrichment."""
    stats: List[dict] = []
    try:
        import torch

        if not torch.cuda.is_available():
            return stats
        for i in range(torch.cuda.device_count()):
            free, total = torch.cuda.mem_get_info(i)
            stats.append(
                {
                    "index": i,
                    "total_mb": total >> 20,
                    "used_mb": (total - free) >> 20,
                }
            )
    except Exception:  # noqa: BLE001
        pass
    return stats


class ResourceMonitor:
    def __init__(self, client: Optional[MasterClient] = None, interval: float = 15.0):
        self._client = client
        self.interval = interval
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="resource-monitor", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def snapshot(self) -> comm.ResourceStats:
        import psutil

        return comm.ResourceStats(
            node_id=int(os.getenv("NODE_ID", "0")),
            cpu_percent=psutil.cpu_percent(interval=None),
            memory_mb=psutil.virtual_memory().used >> 20,
            gpu_stats=collect_gpu_stats(),
        )

    def _loop(self):
        while not self._stop.wait(self.interval):
            try:
                if self._client is not None:
                    self._client.report_used_resource(self.snapshot())
            except Exception:  # noqa: BLE001
                logger.warning("resource report failed", exc_info=True)


class TorchTrainingMonitor:
    """Reports the trainer-written global step to the master (throughput +
    hang-detection input)."""

    def __init__(self, client: Optional[MasterClient] = None, interval: float = 15.0):
        self._client = client
        self.interval = interval
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._last_reported = -1

    def _step_file(self) -> str:
        return os.path.join(
            STEP_FILE_DIR, f"global_step_{os.getenv('ELASTIC_JOB_NAME', 'job')}.json"
        )

    def read_step(self) -> Optional[dict]:
        try:
            with open(self._step_file()) as f:
                return json.load(f)
        except (OSError, ValueError):
            return None

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="training-monitor", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _loop(self):
        while not self._stop.wait(self.interval):
            try:
                data = self.read_step()
                if data and data.get("step", -1) != self._last_reported:
                    self._last_reported = data["step"]
                    if self._client is not None:
                        self._client.report_global_step(
                            data["step"], data.get("ts", time.time())
                        )
            except Exception:  # noqa: BLE001
                logger.warning("step report failed", exc_info=True)
