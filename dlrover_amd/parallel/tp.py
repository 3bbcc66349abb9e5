"""Tensor-parallel layers over RCCL (xGMI all-reduce).

Megatron-style sharding, implemented fresh for MI355X: column-parallel keeps
the GEMM output sharded (no comm forward; all-reduce of input grads
backward), row-parallel all-reduces the GEMM output forward. A transformer
block needs exactly two all-reduces per direction (after o_proj and after
down_proj) — the xGMI-friendly minimum.
"""


import torch
import torch.distributed as dist
import torch.nn as nn


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce grads backward (input of column-parallel)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        # group None == tp degenerate (pgroups convention): NO collective.
        # Falling through to the default group would all-reduce across pp/dp
        # ranks — wrong gradients and a deadlock across pipeline stages.
        if ctx.group is not None:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward; identity backward (output of row-parallel)."""

    @staticmethod
    def forward(ctx, x, group):
        if group is None:  # tp degenerate: nothing to reduce
            return x
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def copy_to_tp(x, group):
    return _CopyToTP.apply(x, group)


def reduce_from_tp(x, group):
    return _ReduceFromTP.apply(x, group)


class ColumnParallelLinear(nn.Module):
    """Y = X @ W^T with W row-sharded (output features split across TP)."""

    def __init__(self, in_features: int, out_features: int, tp_size: int,
                 tp_group=None, bias: bool = False, input_is_parallel_input: bool = True):
        super().__init__()
        if out_features % tp_size != 0:
            raise ValueError(f"out_features {out_features} % tp {tp_size} != 0")
        self.tp_size = tp_size
        self.tp_group = tp_group
        self.out_per_rank = out_features // tp_size
        self.weight = nn.Parameter(torch.empty(self.out_per_rank, in_features))
        self.bias = nn.Parameter(torch.zeros(self.out_per_rank)) if bias else None
        self._copy_input = input_is_parallel_input

    def forward(self, x):
        if self.tp_size > 1 and self._copy_input:
            x = copy_to_tp(x, self.tp_group)
        return torch.nn.functional.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """Y = X @ W^T with W column-sharded (input features split across TP);
    output all-reduced."""

    def __init__(self, in_features: int, out_features: int, tp_size: int,
                 tp_group=None, bias: bool = False):
        super().__init__()
        if in_features % tp_size != 0:
            raise ValueError(f"in_features {in_features} % tp {tp_size} != 0")
        self.tp_size = tp_size
        self.tp_group = tp_group
        self.in_per_rank = in_features // tp_size
        self.weight = nn.Parameter(torch.empty(out_features, self.in_per_rank))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None

    def forward(self, x):
        y = torch.nn.functional.linear(x, self.weight)
        if self.tp_size > 1:
            y = reduce_from_tp(y, self.tp_group)
        if self.bias is not None:
            y = y + self.bias
        return y


def shard_full_weight(full: torch.Tensor, tp_rank: int, tp_size: int, dim: int
                      ) -> torch.Tensor:
    """Slice a replicated weight into this rank's TP shard."""
    chunk = full.shape[dim] // tp_size
    return full.narrow(dim, tp_rank * chunk, chunk).contiguous()
