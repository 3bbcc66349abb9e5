"""TP/PP (Megatron-style) flash checkpoint engine.

Parity target: ref dlrover/trainer/torch/flash_checkpoint/megatron.py:54 +
megatron_dist_ckpt.py:152 — shards indexed by (tp_rank, pp_rank), distributed-
optimizer aware, shm-first with async persist. Our build checkpoints OUR
TP/PP stack (dlrover_amd.parallel + models.llama_parallel); the on-disk
layout keeps Megatron's mp_rank naming so tooling reads it:
    <dir>/<step>/mp_rank_{tp:02d}_{pp:03d}.pt  + .done files + tracker.
Only dp_rank==0 saves (model+optimizer state is replicated across DP).
"""

import os
from typing import Optional

import torch.distributed as dist

from dlrover_amd.common.log import logger
from dlrover_amd.parallel.pgroups import ParallelGroups
from dlrover_amd.trainer.flash_checkpoint.checkpointer import (
    Checkpointer,
    StorageType,
)
from dlrover_amd.trainer.flash_checkpoint.engine import CheckpointEngine


class TpPpCheckpointEngine(CheckpointEngine):
    def __init__(self, checkpoint_dir: str, groups: ParallelGroups, **kw):
        self.groups = groups
        super().__init__(checkpoint_dir, **kw)

    def rank_saves(self) -> bool:
        return self.groups.dp_rank == 0

    def expected_shards(self) -> int:
        return self.groups.dims.tp * self.groups.dims.pp

    def _shard_file_name(self, global_rank: int = -1) -> str:
        return f"mp_rank_{self.groups.tp_rank:02d}_{self.groups.pp_rank:03d}.pt"

    def gather_state_dict(self, model, optimizer):
        sd = {"model": model.state_dict()}
        if optimizer is not None:
            sd["optimizer"] = optimizer.state_dict()
        sd["parallel"] = {
            "tp_rank": self.groups.tp_rank,
            "pp_rank": self.groups.pp_rank,
            "tp": self.groups.dims.tp,
            "pp": self.groups.dims.pp,
        }
        return sd

    def load_into(self, model, optimizer, state_dict):
        par = state_dict.get("parallel", {})
        if par and (
            par.get("tp_rank") != self.groups.tp_rank
            or par.get("pp_rank") != self.groups.pp_rank
        ):
            raise RuntimeError(
                f"checkpoint shard is for tp{par.get('tp_rank')}/pp"
                f"{par.get('pp_rank')}, this rank is tp{self.groups.tp_rank}/"
                f"pp{self.groups.pp_rank}"
            )
        model.load_state_dict(state_dict["model"])
        if optimizer is not None and state_dict.get("optimizer"):
            optimizer.load_state_dict(state_dict["optimizer"])

    def load_from_storage(self, path: str = "", device=None):
        """PARALLEL load (ref: megatron_dist_ckpt.py parallel load,
        242 s -> 156 s on NAS): every DP rank of a (tp, pp) pair needs the
        SAME shard file — instead of dp_world concurrent reads of slow
        shared storage, dp_rank 0 reads once and broadcasts over the DP
        group. Disable with DLROVER_MEGATRON_PARALLEL_LOAD=0."""
        if not path:
            from dlrover_amd.common.storage import read_tracker_step

            step = read_tracker_step(self.checkpoint_dir)
            if step < 0:
                return None
            path = os.path.join(self.checkpoint_dir, str(step))
        shard = os.path.join(path, self._shard_file_name())
        import torch

        g = self.groups
        use_bcast = (
            os.getenv("DLROVER_MEGATRON_PARALLEL_LOAD", "1") != "0"
            and g.dp_group is not None
            and dist.is_initialized()
        )
        if not use_bcast:
            if not os.path.exists(shard):
                return None
            return torch.load(shard, map_location=device or "cpu",
                              weights_only=False)
        obj = [None]
        if g.dp_rank == 0:
            if os.path.exists(shard):
                obj[0] = torch.load(shard, map_location="cpu",
                                    weights_only=False)
        # global rank of this (tp, pp) pair's dp_rank-0 member
        src = g.pp_rank * g.dims.tp + g.tp_rank
        dist.broadcast_object_list(obj, src=src, group=g.dp_group)
        return obj[0]


class MegatronCheckpointer(Checkpointer):
    """User-facing TP/PP checkpointer (ref: flash_checkpoint/megatron.py:54)."""

    def __init__(self, checkpoint_dir: str, groups: ParallelGroups,
                 model=None, optimizer=None, storage=None):
        self.checkpoint_dir = checkpoint_dir
        self.model = model
        self.optimizer = optimizer
        self.engine = TpPpCheckpointEngine(checkpoint_dir, groups, storage=storage)

    def save_checkpoint(
        self,
        step: int,
        state_dict: Optional[dict] = None,
        path: str = "",
        storage_type: int = StorageType.DISK,
    ) -> float:
        sd = state_dict
        if sd is None:
            sd = self.engine.gather_state_dict(self.model, self.optimizer)
        sd["step"] = step
        if storage_type == StorageType.MEMORY:
            return self.engine.save_to_memory(step, sd, path=path)
        return self.engine.save_to_storage(step, sd, path=path)

    def load_checkpoint(self, resume_path: str = "") -> Optional[dict]:
        sd = self.engine.load(resume_path)
        if sd is None:
            return None
        if self.model is not None:
            self.engine.load_into(self.model, self.optimizer, sd)
        if dist.is_available() and dist.is_initialized():
            dist.barrier()
        logger.info(
            "restored TP/PP checkpoint step=%s (%s)",
            sd.get("step"),
            self.engine._shard_file_name(),
        )
        return sd

    def wait_latest_checkpoint(self, timeout: int = 600):
        self.engine.wait_saving()

    def close(self):
        self.engine.close()
