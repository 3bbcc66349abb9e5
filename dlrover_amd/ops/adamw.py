"""FusedAdamW: bf16 training with fp32 master weights, one HIP kernel per
param, optionally replayed as a single hipGraph.

MI355X-native replacement for the fused Adam the reference delegates to
Megatron/Apex (BASELINE.json north star). Design:
  - model params stay bf16 (what forward/backward/collectives touch);
  - this optimizer owns fp32 master params + fp32 moments;
  - each step launches adamw_kernel once per param; the kernel updates
    master/m/v AND writes the bf16 param in the same pass;
  - with ``capture_graph=True`` the per-param launch sequence is captured
    into a hipGraph after the first step and replayed thereafter (launch
    overhead of ~300 small kernels -> 1 graph launch). Requires stable grad
    pointers (true under DDP bucket views and our Llama trainer).

State-dict format is torch-optimizer-compatible (state[param] has step /
exp_avg / exp_avg_sq / master_param) so flash checkpoint handles it like any
optimizer.
"""

from typing import Iterable, Optional

import torch

from dlrover_amd.ops.api import fused_adamw_step

try:
    from torch.distributed.tensor import DTensor
except ImportError:  # pragma: no cover
    DTensor = ()


def _local(t: torch.Tensor) -> torch.Tensor:
    """FSDP2 params/grads are DTensors; the kernel operates on the local
    shard (each rank owns its shard's optimizer state — ZeRO-3 style)."""
    if DTensor and isinstance(t, DTensor):
        return t.to_local()
    return t


class FusedAdamW(torch.optim.Optimizer):
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-4,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.1,
        capture_graph: bool = False,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._capture_graph = capture_graph and torch.cuda.is_available()
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._graph_ptrs = None
        self._lr_at_capture = None

    def _init_state(self, p: torch.Tensor):
        state = self.state[p]
        state["step"] = 0
        master = _local(p.detach()).float().clone()
        state["master_param"] = master
        state["exp_avg"] = torch.zeros_like(master)
        state["exp_avg_sq"] = torch.zeros_like(master)

    def _one_step(self):
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "master_param" not in state:
                    self._init_state(p)
                state["step"] += 1
                is_bf16 = p.dtype == torch.bfloat16
                p_data = _local(p.data)
                fused_adamw_step(
                    state["master_param"],
                    _local(p.grad).contiguous(),
                    state["exp_avg"],
                    state["exp_avg_sq"],
                    p_data if is_bf16 else None,
                    group["lr"],
                    beta1,
                    beta2,
                    group["eps"],
                    group["weight_decay"],
                    state["step"],
                )
                if not is_bf16:
                    # fp32 params: master IS the param storage
                    p_data.copy_(state["master_param"])

    def _grad_ptrs(self):
        return tuple(
            p.grad.data_ptr()
            for g in self.param_groups
            for p in g["params"]
            if p.grad is not None
        )

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        if not self._capture_graph:
            self._one_step()
            return loss

        ptrs = self._grad_ptrs()
        lr = self.param_groups[0]["lr"]
        if self._graph is None or ptrs != self._graph_ptrs or lr != self._lr_at_capture:
            # warm-up step on a side stream, then capture the next one.
            # NOTE: `step` increments inside the graph are host-side, so the
            # kernel's bias correction uses a step snapshot; graphs are only
            # exact when bias correction has converged (step >> 1/(1-beta)).
            # We therefore run eagerly for the first 100 steps.
            min_step = min(
                (self.state[p].get("step", 0))
                for g in self.param_groups
                for p in g["params"]
                if p.grad is not None
            )
            if min_step < 100:
                self._one_step()
                return loss
            torch.cuda.synchronize()
            self._graph = torch.cuda.CUDAGraph()
            # capture records the kernels WITHOUT executing them — the
            # step counters bumped inside _one_step() then describe work
            # that has not run yet. Replay immediately (capture-then-replay)
            # so the capture iteration's gradients are actually applied.
            with torch.cuda.graph(self._graph):
                self._one_step()
            self._graph.replay()
            self._graph_ptrs = ptrs
            self._lr_at_capture = lr
            return loss
        # replay path: bump host-side step counters to keep state dict honest
        for g in self.param_groups:
            for p in g["params"]:
                if p.grad is not None:
                    self.state[p]["step"] += 1
        self._graph.replay()
        return loss

    def load_state_dict(self, state_dict):
        """Override: torch's default casts every floating state tensor to the
        PARAM dtype — which would narrow our fp32 master weights to bf16.
        Restore state by position with device moves only."""
        groups = self.param_groups
        saved_groups = state_dict["param_groups"]
        params = [p for g in groups for p in g["params"]]
        saved_ids = [pid for g in saved_groups for pid in g["params"]]
        if len(params) != len(saved_ids):
            raise ValueError(
                f"optimizer param count mismatch: {len(params)} vs {len(saved_ids)}"
            )
        new_state = {}
        for sid, p in zip(saved_ids, params):
            if sid in state_dict["state"]:
                dev = _local(p).device
                cur = self.state.get(p, {})
                entry = {}
                for k, v in state_dict["state"][sid].items():
                    if torch.is_tensor(v):
                        # copy INTO existing state storage when shapes match:
                        # a fresh .to(dev) would double-allocate ~96 GB of
                        # fp32 optimizer state mid-restore (OOM on 8B @ N=1)
                        old = cur.get(k)
                        if (
                            torch.is_tensor(old)
                            and old.shape == v.shape
                            and old.dtype == v.dtype
                            and old.device == dev
                        ):
                            old.copy_(v, non_blocking=True)
                            entry[k] = old
                        else:
                            entry[k] = v.to(dev)
                    else:
                        entry[k] = v
                new_state[p] = entry
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.state.clear()
        self.state.update(new_state)
        for g, sg in zip(groups, saved_groups):
            g.update({k: v for k, v in sg.items() if k != "params"})
        self._graph = None  # state tensors moved: any captured graph is stale

    def zero_grad(self, set_to_none: bool = True):
        # torch semantics by default: dropping grads lets backward ASSIGN the
        # first accumulation instead of add_-ing into zeroed storage (measured
        # ~2% of an 8B step in fills + fan-in adds). Graph capture is the one
        # case that needs stable grad storage — zero in place there.
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    if set_to_none and not self._capture_graph:
                        p.grad = None
                    else:
                        p.grad.detach_()
                        p.grad.zero_()
