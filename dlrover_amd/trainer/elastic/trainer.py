"""ElasticTrainer: keep the EFFECTIVE global batch size fixed while the
world size changes (ref: dlrover/trainer/torch/elastic/trainer.py:181-340).

Mechanism: gradient accumulation steps = ceil(max_workers / current_world);
optimizer.step() fires every `accum` micro-batches, with DDP gradient sync
suppressed (no_sync) on non-boundary micro-batches. Writes the global step to
a file the agent's training monitor reports to the master (throughput +
hang detection input).
"""

import contextlib
import json
import os
import time
from typing import Optional

import torch.distributed as dist

from dlrover_amd.common.log import logger

STEP_FILE_DIR = "/tmp/dlrover_amd_monitor"


def _world_size() -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return int(os.getenv("WORLD_SIZE", "1"))


class ElasticTrainer:
    def __init__(self, model, dataloader=None, max_workers: Optional[int] = None):
        self.model = model
        self.dataloader = dataloader
        self.max_workers = max_workers or int(
            os.getenv("DLROVER_MAX_WORKERS", str(_world_size()))
        )
        self.gradient_state = _GradState(self._accum_steps())
        self.global_step = 0
        self._step_file = os.path.join(
            STEP_FILE_DIR, f"global_step_{os.getenv('ELASTIC_JOB_NAME', 'job')}.json"
        )
        os.makedirs(STEP_FILE_DIR, exist_ok=True)
        # in-process GC/dataloader tracing feeding the hang/starvation
        # metrics (ref xpu_timer py_tracing; enabled with the profiler)
        self.py_tracer = None
        if os.getenv("DLROVER_PY_TRACE", "") == "1" or (
            os.getenv("DLROVER_HIPTIMER", "") == "1"
            and os.getenv("HIPTIMER_METRICS_DIR")
        ):
            from dlrover_amd.diagnosis.py_runtime_tracer import PyRuntimeTracer

            self.py_tracer = PyRuntimeTracer().start()
            if self.dataloader is not None:
                self.dataloader = self.py_tracer.wrap_loader(self.dataloader)
        self._report_model_info()

    def _report_model_info(self):
        """Best-effort model card to the master (ref: stats/job_collector);
        feeds the strategy generator and the dashboard."""
        if not os.getenv("DLROVER_MASTER_ADDR"):
            return
        try:
            from dlrover_amd.agent.master_client import MasterClient

            mod = self.model.module if hasattr(self.model, "module") else self.model
            cfg = getattr(mod, "cfg", None)
            MasterClient.singleton_instance().report_model_info(
                model_name=type(mod).__name__,
                params=sum(p.numel() for p in mod.parameters()),
                n_layers=getattr(cfg, "n_layers", 0),
                n_heads=getattr(cfg, "n_heads", 0),
                hidden_size=getattr(cfg, "hidden_size", 0),
                seq_len=getattr(cfg, "max_seq_len", 0),
                dtype=str(next(mod.parameters()).dtype).replace("torch.", ""),
            )
        except Exception:  # noqa: BLE001 — reporting must never break training
            logger.debug("model info report skipped", exc_info=True)

    def _accum_steps(self) -> int:
        world = max(_world_size(), 1)
        return max(1, (self.max_workers + world - 1) // world)

    def reset(self):
        """Call after a re-rendezvous changed the world size."""
        self.gradient_state = _GradState(self._accum_steps())
        logger.info(
            "elastic trainer: world=%s accumulation=%s",
            _world_size(),
            self.gradient_state.accum_steps,
        )

    @contextlib.contextmanager
    def step(self):
        """Context manager around one micro-batch: suppresses DDP allreduce
        except on accumulation boundaries (ref: trainer.py:271-277)."""
        gs = self.gradient_state
        gs.micro_step += 1
        sync = gs.micro_step % gs.accum_steps == 0
        ctx = contextlib.nullcontext()
        if not sync and hasattr(self.model, "no_sync"):
            ctx = self.model.no_sync()
        with ctx:
            yield sync
        if sync:
            self.global_step += 1
            self._report_step()

    @property
    def step_boundary(self) -> bool:
        gs = self.gradient_state
        return gs.micro_step % gs.accum_steps == 0

    def prepare(self, optimizer, lr_scheduler=None):
        """Reference-API convenience (ref: trainer.py:229 prepare): returns
        wrappers whose ``step()`` fires only on accumulation boundaries, so
        a loop written for the reference runs unchanged:

            optimizer, scheduler = elastic_trainer.prepare(opt, sched)
            with elastic_trainer.step():
                loss.backward(); optimizer.step(); optimizer.zero_grad()
        """
        opt = _BoundaryStepper(self, optimizer, zero_on_boundary_only=True)
        if lr_scheduler is None:
            return opt
        return opt, _BoundaryStepper(self, lr_scheduler)

    def _report_step(self):
        try:
            rank = dist.get_rank() if dist.is_initialized() else 0
            if rank == 0:
                with open(self._step_file, "w") as f:
                    json.dump({"step": self.global_step, "ts": time.time()}, f)
        except OSError:
            pass


class _GradState:
    def __init__(self, accum_steps: int):
        self.accum_steps = accum_steps
        self.micro_step = 0


class _BoundaryStepper:
    """Proxy for an optimizer/scheduler whose step() only fires on the
    trainer's accumulation boundary; everything else passes through."""

    def __init__(self, trainer: ElasticTrainer, inner, zero_on_boundary_only=False):
        self._trainer = trainer
        self._inner = inner
        self._zero_gate = zero_on_boundary_only

    def step(self, *a, **kw):
        if self._trainer.step_boundary:
            return self._inner.step(*a, **kw)
        return None

    def zero_grad(self, *a, **kw):
        # off-boundary zeroing would drop accumulated gradients
        if not self._zero_gate or self._trainer.step_boundary:
            return self._inner.zero_grad(*a, **kw)
        return None

    def __getattr__(self, name):
        return getattr(self._inner, name)
