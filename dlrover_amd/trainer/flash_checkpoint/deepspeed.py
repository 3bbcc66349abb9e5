"""DeepSpeed flash checkpointer (ref: flash_checkpoint/deepspeed.py:98 +
deepspeed_engine.py:31).

DeepSpeed is not part of the MI355X image (FSDP2 + our TP/PP cover the same
parallelism space natively — PARITY.md), but the integration point is kept:
a DeepSpeedEngine's state splits into the module (ZeRO-gathered or local) and
per-rank optimizer shards, which map onto our ShardedCheckpointEngine shm
layout unchanged. Import is lazy so this module only needs deepspeed when
actually used.
"""

from typing import Optional

from dlrover_amd.common.log import logger
from dlrover_amd.trainer.flash_checkpoint.checkpointer import (
    Checkpointer,
    StorageType,
)
from dlrover_amd.trainer.flash_checkpoint.engine import CheckpointEngine


class DeepSpeedCheckpointEngine(CheckpointEngine):
    """Each rank snapshots its module replica + its ZeRO optimizer shard."""

    def rank_saves(self) -> bool:
        return True

    def gather_state_dict(self, model_engine, _optimizer_unused=None):
        # model_engine: deepspeed.DeepSpeedEngine
        sd = {
            "module": model_engine.module.state_dict(),
            "optimizer": model_engine.optimizer.state_dict()
            if getattr(model_engine, "optimizer", None) is not None
            else {},
            "lr_scheduler": model_engine.lr_scheduler.state_dict()
            if getattr(model_engine, "lr_scheduler", None) is not None
            else {},
            "ds_config": getattr(model_engine, "config", {}) or {},
        }
        return sd

    def load_into(self, model_engine, _optimizer_unused, state_dict):
        model_engine.module.load_state_dict(state_dict["module"])
        if state_dict.get("optimizer") and getattr(model_engine, "optimizer", None):
            model_engine.optimizer.load_state_dict(state_dict["optimizer"])
        if state_dict.get("lr_scheduler") and getattr(
            model_engine, "lr_scheduler", None
        ):
            model_engine.lr_scheduler.load_state_dict(state_dict["lr_scheduler"])


class DeepSpeedCheckpointer(Checkpointer):
    def __init__(self, model_engine, checkpoint_dir: str, storage=None):
        try:
            import deepspeed  # noqa: F401
        except ImportError as e:
            raise ImportError(
                "DeepSpeedCheckpointer requires deepspeed (not installed in "
                "the MI355X image); use FsdpShardCheckpointer or "
                "MegatronCheckpointer instead"
            ) from e
        self.model_engine = model_engine
        self.checkpoint_dir = checkpoint_dir
        self.engine = DeepSpeedCheckpointEngine(checkpoint_dir, storage=storage)

    def save_checkpoint(self, step: int, state_dict: Optional[dict] = None,
                        path: str = "", storage_type: int = StorageType.DISK
                        ) -> float:
        sd = state_dict or self.engine.gather_state_dict(self.model_engine)
        sd["step"] = step
        if storage_type == StorageType.MEMORY:
            return self.engine.save_to_memory(step, sd, path=path)
        return self.engine.save_to_storage(step, sd, path=path)

    def load_checkpoint(self, resume_path: str = "") -> Optional[dict]:
        sd = self.engine.load(resume_path)
        if sd is None:
            return None
        self.engine.load_into(self.model_engine, None, sd)
        logger.info("restored DeepSpeed checkpoint step=%s", sd.get("step"))
        return sd

    def wait_latest_checkpoint(self, timeout: int = 600):
        self.engine.wait_saving()

    def close(self):
        self.engine.close()
