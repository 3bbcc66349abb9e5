"""PrimeMaster: the unified-architecture controller.

Parity target: ref dlrover/python/unified/controller/master.py +
manager.py:88-797 (PrimeManager: prepare -> schedule execution graph ->
_main_loop; deal_with_actor_restarting: per-role failover; restart_job;
state save/self-recover) and schedule/graph.py (DLExecutionGraph).

Backend: local processes via multiprocessing (the reference uses Ray actors;
this image has no Ray). Vertices, role-level failover budgets and the
save/recover state file behave the same.
"""

import json
import multiprocessing as mp
import os
import threading
import time
from dataclasses import dataclass
from typing import List, Optional

from dlrover_amd.common.log import logger
from dlrover_amd.unified.api import DLJob
from dlrover_amd.unified.scheduler import NodeSpec, Placement, Scheduler


@dataclass
class Vertex:
    role: str
    rank: int
    world_size: int
    restarts: int = 0
    proc: Optional[object] = None

    @property
    def name(self) -> str:
        return f"{self.role}-{self.rank}"


class DLExecutionGraph:
    """Roles -> per-worker vertices (ref: schedule/graph.py:269)."""

    def __init__(self, job: DLJob):
        self.job = job
        self.vertices: List[Vertex] = []
        for role in job.roles.values():
            for r in range(role.total):
                self.vertices.append(Vertex(role.name, r, role.total))

    def by_role(self, role: str) -> List[Vertex]:
        return [v for v in self.vertices if v.role == role]


def _worker_entry(desc_env, role, rank, world, entry_func, entry_args):
    os.environ.update(desc_env)
    os.environ.update(
        {
            "ROLE": role,
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
        }
    )
    entry_func(*entry_args)


class PrimeMaster:
    """Controller driving the execution graph with per-role failover."""

    def __init__(self, job: DLJob, state_path: str = "",
                 nodes: Optional[List[NodeSpec]] = None):
        self.job = job
        self.graph = DLExecutionGraph(job)
        self.state_path = state_path or f"/tmp/dlrover_amd_prime_{job.name}.json"
        self._ctx = mp.get_context("spawn")
        self._stop = threading.Event()
        self._monitor: Optional[threading.Thread] = None
        self.status = "INIT"
        self.exit_code: Optional[int] = None
        self.nodes = nodes or self._default_pool()
        self.placement: Optional[Placement] = None
        # self-recovery (ref: manager state save/self-recover :591-644):
        # a master restarted mid-job resumes failover budgets from disk
        prev = self.load_state(self.state_path)
        if prev and prev.get("status") == "RUNNING":
            budgets = {v["name"]: v.get("restarts", 0)
                       for v in prev.get("vertices", [])}
            for v in self.graph.vertices:
                v.restarts = budgets.get(v.name, 0)
            logger.info("prime master self-recovered state: %s", budgets)

    def _default_pool(self) -> List[NodeSpec]:
        # DLROVER_PRIME_NODES="node0:8,node1:8" (gpus per node); defaults to
        # one node sized to hold the whole graph (local backend)
        spec = os.getenv("DLROVER_PRIME_NODES", "")
        if spec:
            out = []
            for part in spec.split(","):
                name, _, g = part.partition(":")
                out.append(NodeSpec(name, int(g or 8)))
            return out
        total_gpu = sum(
            int(r.resource.get("gpu", 0)) * r.total
            for r in self.job.roles.values()
        )
        total_cpu = sum(
            float(r.resource.get("cpu", 1)) * r.total
            for r in self.job.roles.values()
        )
        return [NodeSpec("local", max(total_gpu, 8), max(total_cpu, 64.0))]

    # -- lifecycle (ref: manager.prepare :141 / start :189) -------------------

    def prepare(self):
        self.status = "SCHEDULING"
        # placement: collocation-aware bundles bin-packed onto the pool
        # (ref: DLExecutionGraph.create + scheduler placement groups)
        self.placement = Scheduler(self.nodes).schedule(self.graph)
        self._save_state()
        return self

    def start(self):
        for v in self.graph.vertices:
            self._launch(v)
        self.status = "RUNNING"
        self._save_state()
        self._monitor = threading.Thread(
            target=self._main_loop, name="prime-main", daemon=True
        )
        self._monitor.start()
        return self

    def _launch(self, v: Vertex):
        desc = self.job.roles[v.role]
        if desc.entry_func is None:
            raise ValueError(f"role {v.role}: local backend needs entry_func")
        env = dict(desc.env)
        env["DLROVER_PRIME_JOB"] = self.job.name
        if self.placement is not None and v.name in self.placement.assignments:
            node, dev = self.placement.assignments[v.name]
            env["NODE_ID"] = node
            env["DEVICE_INDEX"] = str(dev)
        p = self._ctx.Process(
            target=_worker_entry,
            args=(env, v.role, v.rank, v.world_size, desc.entry_func,
                  desc.entry_args),
            name=v.name,
        )
        p.start()
        v.proc = p
        logger.info("launched %s (pid %s)", v.name, p.pid)

    # -- main loop (ref: manager._main_loop :203 + restart ladder :292-508) ----

    def _main_loop(self):
        while not self._stop.is_set():
            time.sleep(0.5)
            alive, failed, done = [], [], []
            for v in self.graph.vertices:
                if v.proc is None:
                    continue
                if v.proc.is_alive():
                    alive.append(v)
                elif v.proc.exitcode == 0:
                    done.append(v)
                else:
                    failed.append(v)
            if failed:
                for v in failed:
                    desc = self.job.roles[v.role]
                    if v.restarts < desc.max_restarts:
                        v.restarts += 1
                        logger.warning(
                            "%s failed (exit %s): restarting role %s (%s/%s)",
                            v.name, v.proc.exitcode, v.role, v.restarts,
                            desc.max_restarts,
                        )
                        # per-role failover: restart every vertex of the role
                        # (ref: deal_with_actor_restarting :292)
                        for peer in self.graph.by_role(v.role):
                            if peer.proc is not None and peer.proc.is_alive():
                                peer.proc.terminate()
                                peer.proc.join(timeout=10)
                            peer.restarts = max(peer.restarts, v.restarts)
                            self._launch(peer)
                        break
                    self.status = "FAILED"
                    self.exit_code = v.proc.exitcode
                    self._stop.set()
                    self._save_state()
                    return
            elif not alive:
                self.status = "SUCCEEDED"
                self.exit_code = 0
                self._stop.set()
                self._save_state()
                return

    def wait(self, timeout: Optional[float] = None) -> int:
        deadline = time.time() + timeout if timeout else None
        while not self._stop.is_set():
            if deadline and time.time() > deadline:
                raise TimeoutError("job did not finish in time")
            time.sleep(0.2)
        if self._monitor is not None:
            self._monitor.join(timeout=5)
        return self.exit_code if self.exit_code is not None else 1

    def stop(self):
        self._stop.set()
        for v in self.graph.vertices:
            if v.proc is not None and v.proc.is_alive():
                v.proc.terminate()
        self.status = "STOPPED"
        self._save_state()

    # -- state persistence (ref: controller/state_backend.py) ------------------

    def _save_state(self):
        try:
            with open(self.state_path, "w") as f:
                json.dump(
                    {
                        "job": self.job.name,
                        "status": self.status,
                        "vertices": [
                            {"name": v.name, "restarts": v.restarts}
                            for v in self.graph.vertices
                        ],
                    },
                    f,
                )
        except OSError:
            pass

    @classmethod
    def load_state(cls, state_path: str) -> Optional[dict]:
        try:
            with open(state_path) as f:
                return json.load(f)
        except (OSError, ValueError):
            return None
