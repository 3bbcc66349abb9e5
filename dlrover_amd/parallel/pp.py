"""Pipeline-parallel schedules over RCCL point-to-point.

Two schedules (ref behavior: Megatron-style PP the reference's checkpoint
layout assumes, dlrover/python/common/constants.py:CheckpointConstant):

- "1f1b" (default): each stage runs (pp-1-rank) warmup forwards, then
  alternates one-forward-one-backward, then drains remaining backwards.
  Peak live activations per stage is bounded by the pipeline depth rather
  than the microbatch count — the right trade on 288 GB HBM3E where the
  constraint is large models, not activation count per se, but it still
  halves peak at micro=2*pp.
- "gpipe": all forwards, then all backwards (kept for tests/reference).

Activations move with dist.send/recv (RCCL maps these to xGMI P2P within a
node). Both schedules produce bit-identical gradients — only ordering and
liveness differ.
"""

from typing import List, Optional

import torch
import torch.distributed as dist

from dlrover_amd.parallel.pgroups import ParallelGroups


def _send(t: torch.Tensor, dst: int, group):
    dist.send(t.contiguous(), dst=dst, group=group)


def _isend(t: torch.Tensor, dst: int, group, state):
    """Non-blocking send; the buffer and work handle are parked in
    state["sends"] until the end of the step. 1F1B needs this: in steady
    state both neighbors send (activation down, gradient up) before either
    receives, which deadlocks with blocking sends."""
    buf = t.contiguous()
    state["sends"].append((dist.isend(buf, dst=dst, group=group), buf))


def _recv(shape, dtype, device, src: int, group) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src=src, group=group)
    return t


class PipelineRunner:
    """Drives one optimizer step of a staged model across the PP group."""

    def __init__(self, stage, groups: ParallelGroups, hidden_size: int,
                 act_dtype: Optional[torch.dtype] = None):
        self.stage = stage
        self.g = groups
        self.hidden = hidden_size
        # activation dtype must match what neighboring stages SEND — infer
        # from the stage parameters (a bf16 stage exchanging into fp32 recv
        # buffers would corrupt the pipeline)
        if act_dtype is None:
            p = next(stage.parameters(), None)
            act_dtype = p.dtype if p is not None else torch.float32
        self.act_dtype = act_dtype

    def _stage_device(self):
        p = next(self.stage.parameters(), None)
        return p.device if p is not None else torch.device("cpu")

    def train_step(
        self,
        micro_inputs: List[torch.Tensor],
        micro_labels: List[torch.Tensor],
        schedule: str = "1f1b",
    ) -> Optional[torch.Tensor]:
        """One optimizer-step worth of microbatches. schedule:
        "1f1b" (default — warmup fwds, steady 1-fwd-1-bwd, cooldown bwds;
        peak activation memory bounded by the stage depth instead of the
        microbatch count) or "gpipe" (all-forward then all-backward).
        Returns mean loss on the LAST stage, None elsewhere."""
        if schedule == "1f1b":
            return self._train_step_1f1b(micro_inputs, micro_labels)
        return self._train_step_gpipe(micro_inputs, micro_labels)

    # -- 1F1B ------------------------------------------------------------------

    def _fwd_micro(self, m, micro_inputs, micro_labels, device, state):
        g = self.g
        if g.is_first_stage:
            x_in, h = micro_inputs[m].to(device), None
        else:
            B, S = micro_inputs[m].shape[:2]
            h = _recv(
                (B, S, self.hidden), self.act_dtype, device,
                g.prev_stage_rank, g.pp_group,
            ).requires_grad_(True)
            x_in = h
        if g.is_last_stage:
            out = self.stage(x_in, labels=micro_labels[m].to(device))
            state["losses"].append(out)
        else:
            out = self.stage(x_in)
            _isend(out.detach(), g.next_stage_rank, g.pp_group, state)
        state["fwd_in"][m] = h
        state["fwd_out"][m] = out

    def _bwd_micro(self, m, n_micro, device, state):
        g = self.g
        out = state["fwd_out"].pop(m)
        h = state["fwd_in"].pop(m)
        if g.is_last_stage:
            (state["losses"][m] / n_micro).backward()
        else:
            grad = _recv(out.shape, self.act_dtype, device,
                         g.next_stage_rank, g.pp_group)
            out.backward(grad)
        if not g.is_first_stage:
            _isend(h.grad, g.prev_stage_rank, g.pp_group, state)

    def _train_step_1f1b(self, micro_inputs, micro_labels):
        g = self.g
        device = self._stage_device()
        n_micro = len(micro_inputs)
        # this stage runs (pp - 1 - pp_rank) warmup forwards before steady state
        warmup = min(g.dims.pp - 1 - g.pp_rank, n_micro)
        state = {"fwd_in": {}, "fwd_out": {}, "losses": [], "sends": []}
        for m in range(warmup):
            self._fwd_micro(m, micro_inputs, micro_labels, device, state)
        fwd_next, bwd_next = warmup, 0
        while fwd_next < n_micro:
            self._fwd_micro(fwd_next, micro_inputs, micro_labels, device, state)
            fwd_next += 1
            self._bwd_micro(bwd_next, n_micro, device, state)
            bwd_next += 1
        while bwd_next < n_micro:
            self._bwd_micro(bwd_next, n_micro, device, state)
            bwd_next += 1
        for work, _buf in state["sends"]:
            work.wait()
        if g.is_last_stage and state["losses"]:
            return torch.stack([l.detach() for l in state["losses"]]).mean()
        return None

    # -- GPipe (fill-drain) ------------------------------------------------------

    def _train_step_gpipe(
        self,
        micro_inputs: List[torch.Tensor],
        micro_labels: List[torch.Tensor],
    ) -> Optional[torch.Tensor]:
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
