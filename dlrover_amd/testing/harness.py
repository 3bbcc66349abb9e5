"""Sim-master + fake-agent harness (ref: dlrover/python/testing/
master_setup.py:15-45 + agent/test_agent.py).

MasterProcess: a real LocalJobMaster in a subprocess (production servicer,
rendezvous, diagnosis). FakeAgent: drives the same MasterClient a real
ElasticTrainingAgent would — join rendezvous, heartbeat, report events —
without spawning workers. Scenario scripts compose these to exercise
failover paths deterministically.
"""

import os
import subprocess
import sys
import tempfile
import threading
import time
import uuid
from typing import Callable, Dict, List, Optional

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.common.constants import NodeEventType, RendezvousName
from dlrover_amd.diagnosis.actions import action_from_wire


class MasterProcess:
    def __init__(self, env: Optional[dict] = None):
        self.env = dict(os.environ)
        self.env.update(env or {})
        self.proc: Optional[subprocess.Popen] = None
        self.addr = ""

    def start(self) -> "MasterProcess":
        port_file = os.path.join(
            tempfile.gettempdir(), f"simmaster_{uuid.uuid4().hex[:8]}.port"
        )
        self.proc = subprocess.Popen(
            [sys.executable, "-m", "dlrover_amd.master.main", "--platform",
             "local", "--port", "0", "--port_file", port_file],
            env=self.env,
        )
        deadline = time.time() + 60
        while time.time() < deadline:
            if os.path.exists(port_file):
                content = open(port_file).read().strip()
                if content:
                    self.addr = f"127.0.0.1:{content}"
                    return self
            if self.proc.poll() is not None:
                raise RuntimeError("sim master died during startup")
            time.sleep(0.2)
        raise TimeoutError("sim master did not start")

    def stop(self):
        if self.proc is not None and self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=15)
            except subprocess.TimeoutExpired:
                self.proc.kill()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()


class FakeAgent:
    """Protocol-level agent: real RPC, no worker processes."""

    def __init__(self, master_addr: str, node_rank: int, nproc: int = 8):
        self.node_rank = node_rank
        self.nproc = nproc
        self.client = MasterClient(master_addr, node_id=node_rank)
        self.actions: List = []
        self._hb_stop = threading.Event()
        self._hb_thread: Optional[threading.Thread] = None

    def join(self, rdzv: str = RendezvousName.TRAINING) -> int:
        return self.client.join_rendezvous(self.node_rank, self.nproc, rdzv_name=rdzv)

    def wait_world(self, rdzv: str = RendezvousName.TRAINING, timeout: float = 60
                   ) -> Dict[int, int]:
        deadline = time.time() + timeout
        while time.time() < deadline:
            _, _, world = self.client.get_comm_world(rdzv, self.node_rank)
            if world and self.node_rank in world:
                return world
            time.sleep(0.3)
        raise TimeoutError(f"node {self.node_rank} never saw a complete world")

    def start_heartbeats(self, interval: float = 2.0):
        def loop():
            while not self._hb_stop.wait(interval):
                try:
                    resp = self.client.report_heart_beat(self.node_rank)
                    action = action_from_wire(resp.action_cls, resp.action_kwargs)
                    if action is not None and action.is_needed():
                        self.actions.append(action)
                except Exception:  # noqa: BLE001
                    return
        self._hb_thread = threading.Thread(target=loop, daemon=True)
        self._hb_thread.start()
        return self

    def report_success(self):
        self.client.report_node_event(NodeEventType.SUCCEEDED_EXITED)

    def report_failure(self, reason: str = "boom"):
        self.client.report_node_event(NodeEventType.FAILED_EXITED, reason)

    def stop(self):
        self._hb_stop.set()
        self.client.close()


def run_scenario(scenario: Callable[[MasterProcess], None],
                 env: Optional[dict] = None):
    """Run a scenario function against a fresh sim master."""
    with MasterProcess(env) as master:
        scenario(master)
