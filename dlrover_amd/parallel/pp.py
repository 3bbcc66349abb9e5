"""Pipeline-parallel GPipe schedule over RCCL point-to-point.

Fill-drain schedule: all microbatch forwards stage-by-stage, then all
backwards in reverse. Activations move with dist.send/recv (batch_isend_irecv
on RCCL maps to xGMI P2P within a node). Simple and correct; 1F1B is a later
optimization (the bubble at pp=2, micro>=4 is already <20%).
"""

from typing import Callable, List, Optional

import torch
import torch.distributed as dist

from dlrover_amd.parallel.pgroups import ParallelGroups


def _send(t: torch.Tensor, dst: int, group):
    dist.send(t.contiguous(), dst=dst, group=group)


def _recv(shape, dtype, device, src: int, group) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src=src, group=group)
    return t


class PipelineRunner:
    """Drives one optimizer step of a staged model across the PP group."""

    def __init__(self, stage, groups: ParallelGroups, hidden_size: int,
                 act_dtype: torch.dtype = torch.float32):
        self.stage = stage
        self.g = groups
        self.hidden = hidden_size
        self.act_dtype = act_dtype

    def _stage_device(self):
        p = next(self.stage.parameters(), None)
        return p.device if p is not None else torch.device("cpu")

    def train_step(
        self,
        micro_inputs: List[torch.Tensor],
        micro_labels: List[torch.Tensor],
    ) -> Optional[torch.Tensor]:
        """GPipe fill-drain. Returns mean loss on the LAST stage, None
        elsewhere. Caller owns optimizer.step()/zero_grad()."""
        g = self.g
        device = self._stage_device()
        n_micro = len(micro_inputs)
        fwd_inputs: List[Optional[torch.Tensor]] = []
        fwd_outputs: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []

        # ---- forward fill ----
        for m in range(n_micro):
            if g.is_first_stage:
                x_in = micro_inputs[m].to(device)
                h = None
            else:
                B, S = micro_inputs[m].shape[:2]
                h = _recv(
                    (B, S, self.hidden), self.act_dtype, device,
                    g.prev_stage_rank, g.pp_group,
                ).requires_grad_(True)
                x_in = h
            if g.is_last_stage:
                out = self.stage(x_in, labels=micro_labels[m].to(device))
                losses.append(out)
            else:
                out = self.stage(x_in)
                _send(out.detach(), g.next_stage_rank, g.pp_group)
            fwd_inputs.append(h)
            fwd_outputs.append(out)

        # ---- backward drain (reverse order) ----
        for m in reversed(range(n_micro)):
            if g.is_last_stage:
                (losses[m] / n_micro).backward()
            else:
                grad = _recv(
                    fwd_outputs[m].shape, self.act_dtype, device,
                    g.next_stage_rank, g.pp_group,
                )
                fwd_outputs[m].backward(grad)
            if not g.is_first_stage:
                _send(fwd_inputs[m].grad, g.prev_stage_rank, g.pp_group)

        if g.is_last_stage and losses:
            return torch.stack([l.detach() for l in losses]).mean()
        return None
