"""Global configuration singleton (ref: dlrover/python/common/global_context.py:1-309)."""

import os
import socket
import threading

from dlrover_amd.common.constants import (
    CommServiceType,
    DEFAULT_MASTER_PORT,
    JobConstant,
)


class DefaultValues:
    SERVICE_TYPE = CommServiceType.TCP
    MASTER_PORT = DEFAULT_MASTER_PORT
    RELAUNCH_ERROR_MAX = 3
    SECONDS_TO_WAIT_PENDING = 900
    SECONDS_HEARTBEAT_TIMEOUT = JobConstant.NODE_HEARTBEAT_TIMEOUT
    HANG_DOWNTIME_SECS = JobConstant.HANG_DOWNTIME_SECS
    SECONDS_INTERVAL_COLLECT = 15
    TRAINING_LOG_LINES = 64


class Context:
    _instance = None
    _lock = threading.Lock()

    def __init__(self):
        self.master_port = int(os.getenv("DLROVER_MASTER_PORT", DefaultValues.MASTER_PORT))
        self.master_service_type = os.getenv(
            "DLROVER_MASTER_SERVICE_TYPE", DefaultValues.SERVICE_TYPE
        )
        self.job_name = os.getenv("ELASTIC_JOB_NAME", "dlrover-job")
        self.relaunch_error_max = DefaultValues.RELAUNCH_ERROR_MAX
        self.seconds_to_wait_pending = DefaultValues.SECONDS_TO_WAIT_PENDING
        self.heartbeat_timeout = float(
            os.getenv("DLROVER_HEARTBEAT_TIMEOUT", DefaultValues.SECONDS_HEARTBEAT_TIMEOUT)
        )
        self.hang_downtime = float(
            os.getenv("DLROVER_HANG_DOWNTIME", DefaultValues.HANG_DOWNTIME_SECS)
        )
        self.seconds_interval_collect = DefaultValues.SECONDS_INTERVAL_COLLECT
        self.auto_worker_enabled = os.getenv("DLROVER_AUTO_WORKER", "") == "1"
        self.auto_ps_enabled = False
        self.pre_check_operators = []
        self.is_tfv1_ps = False
        self.standalone = False

    @classmethod
    def singleton_instance(cls) -> "Context":
        if cls._instance is None:
            with cls._lock:
                if cls._instance is None:
                    cls._instance = cls()
        return cls._instance

    @classmethod
    def _reset_for_tests(cls):
        with cls._lock:
            cls._instance = None


def find_free_port(host: str = "127.0.0.1") -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind((host, 0))
        return s.getsockname()[1]


def find_free_port_in_range(start: int, end: int, host: str = "127.0.0.1") -> int:
    for port in range(start, end):
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
            try:
                s.bind((host, port))
                return port
            except OSError:
                continue
    raise RuntimeError(f"no free port in [{start}, {end})")
