"""Resource optimization + auto-scaling.

Parity targets:
  - ref master/node/job_auto_scaler.py:71-375 (AllreduceTrainingAutoScaler:
    scale workers by alive/pending counts; OOM recovery)
  - ref master/resource/local_optimizer.py:66-400 (PSLocalOptimizer resource
    plans per job stage)
  - ref master/hyperparams/simple_strategy_generator.py (ParallelConfig
    tuning: dataloader batch size / workers from node stats)

The MI355X build keeps the same shape: observe PerfMonitor + node stats ->
produce a ResourcePlan / ParallelConfig; the scaler executes node-count
changes, the agents pick up ParallelConfig over RPC.
"""

import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

from dlrover_amd.common import comm
from dlrover_amd.common.constants import NodeStatus
from dlrover_amd.common.log import logger
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.perf_monitor import PerfMonitor


@dataclass
class ResourcePlan:
    """Desired worker group size/resources (ref: comm ResourcePlan)."""

    node_count: int = 0
    node_resource: Dict[str, float] = field(default_factory=dict)
    comment: str = ""


class QuotaChecker:
    """Free-capacity oracle bounding scale-up (ref: master/cluster/quota.py).
    The default assumes capacity is always available; platforms plug their
    own."""

    def get_free_node_num(self) -> int:
        import sys

        return sys.maxsize


class NoFreeQuotaChecker(QuotaChecker):
    def get_free_node_num(self) -> int:
        return 0


class LocalResourceOptimizer:
    """Single-job heuristics (ref: PSLocalOptimizer, reduced to the
    allreduce/worker case that matters for GPU training)."""

    def __init__(self, perf: PerfMonitor, ctx: Optional[JobContext] = None,
                 min_nodes: int = 1, max_nodes: int = 1, job_name: str = "",
                 quota: Optional[QuotaChecker] = None):
        import os

        self.perf = perf
        self.ctx = ctx or JobContext.singleton_instance()
        self.min_nodes = min_nodes
        self.max_nodes = max_nodes
        self.quota = quota or QuotaChecker()
        self.job_name = job_name or os.getenv("ELASTIC_JOB_NAME", "job")
        # optimizeMode=cluster: a Brain endpoint outranks local heuristics
        # (ref: brain_optimizer.py wrapper; falls back when unreachable)
        self._brain = None
        if os.getenv("DLROVER_BRAIN_ADDR", ""):
            from dlrover_amd.brain_client import BrainClient

            client = BrainClient()
            self._brain = client if client.available else None

    def _brain_plan(self) -> Optional[ResourcePlan]:
        if self._brain is None:
            return None
        alive = [n for n in self.ctx.job_nodes().values() if n.is_alive()]
        speed = self.perf.running_speed() if self.perf is not None else 0.0
        self._brain.report_metrics(self.job_name, {"steps_per_sec": speed})
        plan = self._brain.get_optimization_plan(
            self.job_name, "running",
            {"current_nodes": len(alive) or 1, "max_nodes": self.max_nodes},
        )
        if not plan or not plan.get("node_count"):
            return None
        count = max(self.min_nodes, min(int(plan["node_count"]), self.max_nodes))
        if count == len(alive):
            return None  # hold
        return ResourcePlan(node_count=count,
                            comment=plan.get("comment", "brain"))

    def generate_plan(self) -> Optional[ResourcePlan]:
        brain = self._brain_plan()
        if brain is not None:
            return brain
        nodes = self.ctx.job_nodes()
        alive = [n for n in nodes.values() if n.is_alive()]
        pending = [n for n in alive if n.status == NodeStatus.PENDING]
        running = [n for n in alive if n.status == NodeStatus.RUNNING]
        # ref: AllreduceTrainingAutoScaler :276 — if pods stay pending the
        # cluster can't satisfy us: shrink to what actually runs (respecting
        # min_nodes) so training proceeds instead of waiting forever
        if pending and len(running) >= self.min_nodes:
            return ResourcePlan(
                node_count=len(running),
                comment=f"{len(pending)} nodes pending: shrink to running set",
            )
        # scale-up bounded by the platform quota (ref: cluster/quota.py)
        free = self.quota.get_free_node_num()
        target = min(self.max_nodes, len(alive) + max(0, free))
        if not pending and len(alive) < target:
            return ResourcePlan(
                node_count=target,
                comment="capacity available: grow to max_nodes",
            )
        return None


class JobAutoScaler:
    """Periodic plan->execute loop (ref: job_auto_scaler.py:71)."""

    def __init__(self, optimizer: LocalResourceOptimizer, scaler,
                 interval: float = 60.0):
        self.optimizer = optimizer
        self.scaler = scaler
        self.interval = interval
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="auto-scaler", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def execute_once(self) -> Optional[ResourcePlan]:
        plan = self.optimizer.generate_plan()
        if plan is None or self.scaler is None:
            return plan
        nodes = list(self.optimizer.ctx.job_nodes().values())
        logger.info("auto-scale: %s -> %s (%s)",
                    len([n for n in nodes if n.is_alive()]),
                    plan.node_count, plan.comment)
        self.scaler.scale_to(plan.node_count, nodes)
        return plan

    def _loop(self):
        while not self._stop.wait(self.interval):
            try:
                self.execute_once()
            except Exception:  # noqa: BLE001
                logger.exception("auto-scale iteration failed")


class SimpleStrategyGenerator:
    """Dataloader/optimizer hyperparam tuning from node stats (ref:
    simple_strategy_generator.py — versioned concrete suggestions, NOT
    TP/PP tuning): grow the dataloader batch into free GPU memory bounded
    by an activation-memory estimate, and rescale the learning rate by the
    square-root batch rule. Trainers apply a config only when its version
    is newer than the one they run."""

    # headroom to keep free so a suggestion cannot OOM (ref keeps 2400 MB;
    # sized up for 288 GB HBM3E parts)
    MIN_FREE_MB = 8192

    def __init__(self, perf: PerfMonitor):
        self.perf = perf
        self._versions: dict = {}

    @staticmethod
    def _activation_mb(batch_size: int, model: dict) -> float:
        """Per-step activation memory of a Llama-style block stack
        (same estimator family as the reference: linear term per token +
        quadratic attention term)."""
        s = model.get("seq_len", 4096)
        layers = model.get("n_layers", 32)
        heads = model.get("n_heads", 32)
        embd = model.get("hidden_size", 4096)
        bytes_per = 2  # bf16 activations
        lin = 34 * batch_size * s * embd * bytes_per
        att = 5 * batch_size * s * s * heads * bytes_per / 1024
        return (lin + att) * layers / (1 << 20)

    def generate_parallel_config(
        self,
        node_id: int = 0,
        current: Optional[comm.ParallelConfig] = None,
        model: Optional[dict] = None,
    ) -> comm.ParallelConfig:
        cfg = current or comm.ParallelConfig()
        stats = self.perf.node_resource(node_id)
        if stats is None or not stats.gpu_stats:
            return cfg
        free_mb = min(
            (g.get("total_mb", 0) - g.get("used_mb", 0))
            for g in stats.gpu_stats
        )
        if free_mb <= self.MIN_FREE_MB:
            return cfg
        bs = cfg.dataloader.batch_size or 1
        act = self._activation_mb(bs, model or {"seq_len": 4096})
        if act <= 0:
            return cfg
        grown = int(bs + bs * (free_mb - self.MIN_FREE_MB) / act)
        if grown <= bs:
            return cfg
        new_bs = min(grown, bs * 4)  # bounded growth per suggestion
        lr = cfg.optimizer.learning_rate
        out = comm.ParallelConfig(
            dataloader=comm.DataLoaderConfig(
                dataloader_name=cfg.dataloader.dataloader_name,
                batch_size=new_bs,
                num_workers=max(cfg.dataloader.num_workers, 4),
                version=cfg.dataloader.version + 1,
            ),
            optimizer=comm.OptimizerConfig(
                optimizer_name=cfg.optimizer.optimizer_name,
                # square-root LR scaling with batch growth
                learning_rate=(lr * (new_bs / bs) ** 0.5) if lr else lr,
                version=cfg.optimizer.version + 1,
            ),
        )
        logger.info(
            "strategy: batch %s -> %s (free %.0f MB, act %.0f MB), lr x%.3f",
            bs, new_bs, free_mb, act, (new_bs / bs) ** 0.5,
        )
        return out
