"""Process-group topology for DP x PP x TP over RCCL/xGMI.

The reference delegates TP/PP to Megatron; on MI355X we
supply them natively. Rank layout follows the Megatron convention the
checkpoint shard naming depends on: tp is the FASTEST-varying dimension, then
pp, then dp —
    global_rank = dp_rank * (pp * tp) + pp_rank * tp + tp_rank
so TP groups are consecutive ranks. On one 8-GPU MI355X node this keeps each
TP group inside the fully-connected xGMI mesh (7 links/GPU): the per-layer TP
all-reduces are small and latency-bound, so locality beats ring length.
"""

from dataclasses import dataclass
from typing import Optional

import torch.distributed as dist


@dataclass
class ParallelDims:
    tp: int = 1
    pp: int = 1
    dp: int = 1

    @property
    def world(self) -> int:
        return self.tp * self.pp * self.dp

    @classmethod
    def infer(cls, world: int, tp: int = 1, pp: int = 1) -> "ParallelDims":
        if world % (tp * pp) != 0:
            raise ValueError(f"world {world} not divisible by tp*pp={tp * pp}")
        return cls(tp=tp, pp=pp, dp=world // (tp * pp))


class ParallelGroups:
    def __init__(self, dims: ParallelDims, rank: Optional[int] = None, backend=None):
        if not dist.is_initialized():
            raise RuntimeError("init_process_group first")
        world = dist.get_world_size()
        if world != dims.world:
            raise ValueError(f"world {world} != dims {dims}")
        self.dims = dims
        self.rank = dist.get_rank() if rank is None else rank
        tp, pp, dp = dims.tp, dims.pp, dims.dp

        self.tp_rank = self.rank % tp
        self.pp_rank = (self.rank // tp) % pp
        self.dp_rank = self.rank // (tp * pp)

        self.tp_group = None
        self.pp_group = None
        self.dp_group = None
        # every rank must participate in every new_group call
        for d in range(dp):
            for p in range(pp):
                ranks = [d * pp * tp + p * tp + t for t in range(tp)]
                g = dist.new_group(ranks, backend=backend) if tp > 1 else None
                if self.rank in ranks:
                    self.tp_group = g
        for d in range(dp):
            for t in range(tp):
                ranks = [d * pp * tp + p * tp + t for p in range(pp)]
                g = dist.new_group(ranks, backend=backend) if pp > 1 else None
                if self.rank in ranks:
                    self.pp_group = g
                    self.pp_ranks = ranks
        for p in range(pp):
            for t in range(tp):
                ranks = [d * pp * tp + p * tp + t for d in range(dp)]
                g = dist.new_group(ranks, backend=backend) if dp > 1 else None
                if self.rank in ranks:
                    self.dp_group = g

    # -- pipeline neighbors --------------------------------------------------

    @property
    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.dims.pp - 1

    @property
    def prev_stage_rank(self) -> Optional[int]:
        return None if self.is_first_stage else self.pp_ranks[self.pp_rank - 1]

    @property
    def next_stage_rank(self) -> Optional[int]:
        return None if self.is_last_stage else self.pp_ranks[self.pp_rank + 1]

    def describe(self) -> str:
        return (
            f"rank{self.rank}: tp={self.tp_rank}/{self.dims.tp} "
            f"pp={self.pp_rank}/{self.dims.pp} dp={self.dp_rank}/{self.dims.dp}"
        )
