"""ElasticDistributedSampler: a DistributedSampler that can (a) resume from a
sample offset after restart and (b) re-shard cleanly when the world size
changes (ref: dlrover/trainer/torch/elastic/sampler.py:25).
"""

import math
from typing import Iterator, Optional

import torch
import torch.distributed as dist
from torch.utils.data import Sampler


class ElasticDistributedSampler(Sampler):
    def __init__(
        self,
        dataset,
        num_replicas: Optional[int] = None,
        rank: Optional[int] = None,
        shuffle: bool = True,
        seed: int = 0,
        drop_last: bool = False,
    ):
        if num_replicas is None:
            num_replicas = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        # completed samples in THIS epoch across ALL replicas (resume point)
        self.completed_num = 0
        n = len(self.dataset)
        if self.drop_last and n % self.num_replicas != 0:
            self.num_samples = n // self.num_replicas
        else:
            self.num_samples = math.ceil(n / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas

    def __iter__(self) -> Iterator[int]:
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(len(self.dataset), generator=g).tolist()
        else:
            indices = list(range(len(self.dataset)))
        if not self.drop_last:
            padding = self.total_size - len(indices)
            if padding > 0:
                indices += (indices * math.ceil(padding / max(len(indices), 1)))[:padding]
        indices = indices[: self.total_size]
        # skip what the job already consumed this epoch, then shard round-robin
        start = self.completed_num
        indices = indices[start:]
        # after an elastic world-size change completed_num is generally NOT a
        # multiple of the new num_replicas: pad the remainder back to one so
        # every rank yields exactly the same count (unequal per-rank lengths
        # hang DDP at epoch end). Mirrors the reference sampler's re-pad.
        rem = len(indices) % self.num_replicas
        if rem:
            pad = self.num_replicas - rem
            pool = indices if indices else list(range(len(self.dataset)))
            indices += pool[:pad]
        return iter(indices[self.rank :: self.num_replicas])

    def __len__(self) -> int:
        remaining = self.total_size - self.completed_num
        return max(0, math.ceil(remaining / self.num_replicas))

    def set_epoch(self, epoch: int):
        self.epoch = epoch
        self.completed_num = 0

    # -- elastic resume (ref: sampler state_dict/load_state_dict) ----------------

    def state_dict(self, step: int = 0, batch_size: int = 0) -> dict:
        return {
            "epoch": self.epoch,
            "completed_num": self.completed_num + step * batch_size * self.num_replicas,
        }

    def load_state_dict(self, state: dict):
        self.epoch = int(state.get("epoch", 0))
        self.completed_num = int(state.get("completed_num", 0))
