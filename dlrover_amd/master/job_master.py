"""Job masters: the per-job brain the agents talk to.

Parity target: ref dlrover/python/master/dist_master.py:101-460 and
local_master.py:41-130. One composite owns the rendezvous managers
(TRAINING + NETWORK_CHECK), KV store, sync service, task manager, perf
monitor, diagnosis master and the RPC servicer; Local vs Distributed differ
only in the job manager (in-process agents vs platform scaler/watcher).
"""

import threading
import time
from typing import Dict, Optional

from dlrover_amd.common import comm
from dlrover_amd.common.constants import JobStage, RendezvousName
from dlrover_amd.common.global_context import Context
from dlrover_amd.common.log import logger
from dlrover_amd.master.diagnosis_master import DiagnosisMaster
from dlrover_amd.master.elastic.kv_store import KVStoreService
from dlrover_amd.master.elastic.rdzv_manager import (
    ElasticTrainingRendezvousManager,
    NetworkCheckRendezvousManager,
)
from dlrover_amd.master.elastic.sync_service import SyncService
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.node.job_manager import (
    DistributedJobManager,
    LocalJobManager,
)
from dlrover_amd.master.perf_monitor import PerfMonitor
from dlrover_amd.master.servicer import start_master_service
from dlrover_amd.master.shard.task_manager import TaskManager


class JobMaster:
    def __init__(
        self,
        port: Optional[int] = None,
        service_type: str = "",
        job_manager=None,
        elastic_run_configs: Optional[Dict[str, str]] = None,
    ):
        cfg = Context.singleton_instance()
        self.ctx = JobContext.singleton_instance()
        self.rdzv_managers = {
            RendezvousName.TRAINING: ElasticTrainingRendezvousManager(),
            RendezvousName.NETWORK_CHECK: NetworkCheckRendezvousManager(),
        }
        self.kv_store = KVStoreService()
        self.sync_service = SyncService()
        self.task_manager = TaskManager()
        self.perf_monitor = PerfMonitor()
        self.diagnosis_manager = DiagnosisMaster(self.perf_monitor, self.ctx)
        self.job_manager = job_manager or LocalJobManager(
            job_context=self.ctx, rdzv_managers=self.rdzv_managers
        )
        self.job_manager.rdzv_managers = self.rdzv_managers
        # dead worker -> its in-flight data shards re-queue (ref:
        # event_callback.py TaskRescheduleCallback)
        self.job_manager.add_node_failure_callback(
            lambda node, reason: self.task_manager.recover_tasks(node.id)
        )
        self._elastic_run_configs = elastic_run_configs or {}
        self._service_type = service_type or cfg.master_service_type
        # port=0 means "bind an ephemeral port" (standalone); None = default
        self._port = cfg.master_port if port is None else port
        self._server = None
        self._ckpt_sync_nodes: Dict[int, int] = {}
        self._ckpt_sync_lock = threading.Lock()

    # -- servicer hooks ------------------------------------------------------------

    def paral_config(self) -> comm.ParallelConfig:
        """Versioned hyperparam suggestion from runtime stats (ref:
        get_paral_config servicer:424 + SimpleStrategyGenerator): uses the
        trainer-reported model card for the activation-memory bound."""
        from dlrover_amd.master.auto_scale import SimpleStrategyGenerator

        info = getattr(self.perf_monitor, "model_info", None)
        model = None
        if info is not None:
            model = {
                "seq_len": info.seq_len or 4096,
                "n_layers": info.n_layers or 32,
                "n_heads": info.n_heads or 32,
                "hidden_size": info.hidden_size or 4096,
            }
        try:
            cfg = SimpleStrategyGenerator(self.perf_monitor).generate_parallel_config(
                0, current=self._last_paral_config, model=model
            )
            self._last_paral_config = cfg
            return cfg
        except Exception:  # noqa: BLE001 — suggestions must never fail a poll
            return self._last_paral_config or comm.ParallelConfig()

    _last_paral_config = None

    def elastic_run_config(self) -> Dict[str, str]:
        return dict(self._elastic_run_configs)

    _precheck_chain = None

    def pre_check_result(self, node_id: int) -> comm.PreCheckResponse:
        """Pluggable operator chain (ref: precheck_operator.py; operators
        named in Context.pre_check_operators / DLROVER_PRE_CHECK_OPS)."""
        import os

        if self._precheck_chain is None:
            from dlrover_amd.common.global_context import Context
            from dlrover_amd.master.precheck import PreCheckChain

            names = list(Context.singleton_instance().pre_check_operators)
            env = os.getenv("DLROVER_PRE_CHECK_OPS", "")
            names += [n.strip() for n in env.split(",") if n.strip()]
            self._precheck_chain = PreCheckChain(names)
        status, reason = self._precheck_chain.evaluate(self)
        return comm.PreCheckResponse(status=status, reason=reason)

    def ckpt_sync(self, node_id: int, step: int) -> bool:
        """Consensus that all nodes persisted `step` (ref:
        rdzv_manager.sync_ckpt_nodes :429)."""
        with self._ckpt_sync_lock:
            self._ckpt_sync_nodes[node_id] = step
            world = self.rdzv_managers[RendezvousName.TRAINING].current_world()
            if not world:
                return True
            return all(
                self._ckpt_sync_nodes.get(rank) == step for rank in world
            )

    # -- lifecycle --------------------------------------------------------------------

    def prepare(self):
        self._server = start_master_service(self, self._service_type, self._port)
        self._port = self._server.port
        self.job_manager.start()
        self.diagnosis_manager.start()
        self.ctx.job_stage = JobStage.RUNNING
        from dlrover_amd.common.events import master_events

        master_events().instant(
            "master_start",
            {"port": self._port, "service": self._service_type},
        )
        logger.info(
            "job master ready on port %s (%s)", self._port, self._service_type
        )
        return self

    @property
    def port(self) -> int:
        return self._port

    def run(self) -> int:
        """Block until the job completes or is aborted; returns exit code
        (ref: dist_master.run :293)."""
        try:
            while not self.ctx.is_stopping():
                time.sleep(1)
        except KeyboardInterrupt:
            self.ctx.request_stop("KeyboardInterrupt", code=1)
        logger.info(
            "job master exiting: reason=%s code=%s",
            self.ctx.exit_reason,
            self.ctx.exit_code,
        )
        return self.ctx.exit_code

    def stop(self):
        self.ctx.job_stage = JobStage.STOPPED
        from dlrover_amd.common.events import master_events

        master_events().instant(
            "master_exit",
            {"reason": self.ctx.exit_reason, "code": self.ctx.exit_code},
        )
        self.diagnosis_manager.stop()
        self.job_manager.stop()
        if self._server is not None:
            self._server.stop()


class LocalJobMaster(JobMaster):
    """Standalone (single node) master (ref: local_master.py)."""


class DistributedJobMaster(JobMaster):
    """Cluster master with platform scaler/watcher (ref: dist_master.py)."""

    def __init__(self, scaler=None, watcher=None, job_watcher=None,
                 scaleplan_watcher=None, **kw):
        ctx = JobContext.singleton_instance()
        jm = DistributedJobManager(
            scaler=scaler, watcher=watcher, job_watcher=job_watcher,
            scaleplan_watcher=scaleplan_watcher, job_context=ctx,
        )
        super().__init__(job_manager=jm, **kw)
