"""FlashCkptTrainer: HuggingFace Trainer with flash checkpointing.

Parity target: ref dlrover/trainer/torch/flash_checkpoint/hf_trainer.py:119 —
a transformers.Trainer subclass whose _save_checkpoint goes through the flash
engine (shm snapshot + async persist) instead of Trainer's blocking
save_model/save_state path.
"""

from typing import Optional

from dlrover_amd.common.log import logger

try:
    from transformers import Trainer  # type: ignore

    _HAS_TRANSFORMERS = True
except ImportError:  # pragma: no cover
    Trainer = object
    _HAS_TRANSFORMERS = False


class FlashCkptTrainer(Trainer):  # type: ignore[misc]
    """Drop-in for transformers.Trainer. Checkpoints go to
    args.output_dir/checkpoint-<step>/ through the flash engine; the newest
    committed step is discoverable via get_last_checkpoint()."""

    def __init__(self, *args, flash_checkpoint_dir: str = "", **kwargs):
        if not _HAS_TRANSFORMERS:
            raise ImportError("transformers is not installed")
        super().__init__(*args, **kwargs)
        from dlrover_amd.trainer.flash_checkpoint.ddp import DdpCheckpointer

        ckpt_dir = flash_checkpoint_dir or self.args.output_dir
        self._flash = DdpCheckpointer(ckpt_dir)

    def _save_checkpoint(self, model, trial=None, metrics=None):
        step = int(self.state.global_step)
        mod = model.module if hasattr(model, "module") else model
        sd = {
            "step": step,
            "model": mod.state_dict(),
            "optimizer": self.optimizer.state_dict() if self.optimizer else {},
            "lr_scheduler": (
                self.lr_scheduler.state_dict() if self.lr_scheduler else {}
            ),
            "trainer_state": self.state.__dict__.copy(),
        }
        blocking = self._flash.save_checkpoint(step, state_dict=sd)
        logger.info("flash checkpoint @%s blocked training %.3fs", step, blocking)

    def get_last_checkpoint(self) -> Optional[int]:
        step = self._flash.engine.latest_step()
        return step if step > 0 else None

    def load_flash_checkpoint(self, model=None, resume_path: str = ""):
        sd = self._flash.engine.load(resume_path)
        if sd is None:
            return None
        mod = model or self.model
        mod = mod.module if hasattr(mod, "module") else mod
        mod.load_state_dict(sd["model"])
        if self.optimizer is not None and sd.get("optimizer"):
            self.optimizer.load_state_dict(sd["optimizer"])
        if self.lr_scheduler is not None and sd.get("lr_scheduler"):
            self.lr_scheduler.load_state_dict(sd["lr_scheduler"])
        return sd
