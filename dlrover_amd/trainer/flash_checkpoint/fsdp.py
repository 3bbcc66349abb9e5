"""FSDP sharded checkpointer (ref: flash_checkpoint/fsdp.py:36
FsdpShardCheckpointer — ours stores each rank's local shards through the
ShardedCheckpointEngine instead of DCP's StorageWriter plumbing; the shm
layout and commit protocol are shared with every other engine).

Works with both FSDP1 (FullyShardedDataParallel wrapper) and FSDP2
(fully_shard + DTensor), because torch.distributed.checkpoint.state_dict
abstracts over them.
"""

from typing import Optional

import torch.distributed as dist

from dlrover_amd.common.log import logger
from dlrover_amd.trainer.flash_checkpoint.checkpointer import (
    Checkpointer,
    StorageType,
)
from dlrover_amd.trainer.flash_checkpoint.engine import ShardedCheckpointEngine


class FsdpShardCheckpointer(Checkpointer):
    def __init__(self, checkpoint_dir: str, model=None, optimizer=None, storage=None):
        self.checkpoint_dir = checkpoint_dir
        self.model = model
        self.optimizer = optimizer
        self.engine = ShardedCheckpointEngine(checkpoint_dir, storage=storage)

    def save_checkpoint(
        self,
        step: int,
        state_dict: Optional[dict] = None,
        path: str = "",
        storage_type: int = StorageType.DISK,
    ) -> float:
        if state_dict is None:
            state_dict = self.engine.gather_state_dict(self.model, self.optimizer)
        state_dict["step"] = step
        if storage_type == StorageType.MEMORY:
            return self.engine.save_to_memory(step, state_dict, path=path)
        return self.engine.save_to_storage(step, state_dict, path=path)

    def load_checkpoint(self, resume_path: str = "") -> Optional[dict]:
        import os

        world = dist.get_world_size() if dist.is_initialized() else 1
        sd = self.engine.load(resume_path)
        if sd is not None and sd.get("world_size", world) != world:
            sd = None  # shard is from a different world size: must reshard
        if sd is None:
            # UCP path: the checkpoint on disk was written at another world
            # size (elastic scale) — reshard it (ref: UCP hook,
            # training.py:1548)
            from dlrover_amd.common.storage import read_tracker_step
            from dlrover_amd.trainer.flash_checkpoint import ucp

            path = resume_path
            if not path:
                step = read_tracker_step(self.checkpoint_dir)
                if step < 0:
                    return None
                path = os.path.join(self.checkpoint_dir, str(step))
            meta = ucp.load_resharded(self.engine, self.model, self.optimizer, path)
            if meta is None:
                return None
            if dist.is_available() and dist.is_initialized():
                dist.barrier()
            logger.info("UCP-resharded FSDP checkpoint step=%s", meta.get("step"))
            return meta
        if self.model is not None:
            self.engine.load_into(self.model, self.optimizer, sd)
        if dist.is_available() and dist.is_initialized():
            dist.barrier()
        logger.info("restored FSDP shard checkpoint step=%s", sd.get("step"))
        return sd

    def wait_latest_checkpoint(self, timeout: int = 600):
        self.engine.wait_saving()

    def close(self):
        self.engine.close()
