"""Autograd-wrapped ops: HIP kernels on GPU, torch fp32 references on CPU."""

from typing import Optional, Tuple

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


class ExtensionMissingError(RuntimeError):
    pass


def hip_ops():
    """Return the compiled extension, importing it lazily. Raises loudly on a
    GPU box if the .so is missing (the driver checks native code is loaded)."""
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from dlrover_amd.ops import _hip_ops as ext  # built in-tree

            _EXT = ext
        except ImportError as e:  # pragma: no cover - build problem
            _EXT_ERR = str(e)
    if _EXT is None:
        raise ExtensionMissingError(
            "dlrover_amd._hip_ops is not built — run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {_EXT_ERR})"
        )
    return _EXT


def hip_ops_available() -> bool:
    try:
        hip_ops()
        return True
    except ExtensionMissingError:
        return False


def _use_hip(*tensors: torch.Tensor) -> bool:
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if on_gpu:
        hip_ops()  # raises if missing: no silent eager fallback on GPU
        return True
    return False


# ---------------------------------------------------------------------------
# reference implementations (CPU path + GPU numerics oracle)
# ---------------------------------------------------------------------------


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


def swiglu_ref(gate_up: torch.Tensor) -> torch.Tensor:
    gate, up = gate_up.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(gate_up.dtype)


def rope_ref(
    x: torch.Tensor, pos: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor
) -> torch.Tensor:
    """x [..., n_heads, D]; pos [n_tokens]; cos/sin [max_pos, D/2] fp32."""
    d = x.shape[-1]
    xf = x.float()
    x1, x2 = xf[..., : d // 2], xf[..., d // 2 :]
    shape = [1] * (x.dim() - 3) + [-1, 1, d // 2]
    c = cos[pos].view(*shape)
    s = sin[pos].view(*shape)
    o1 = x1 * c - x2 * s
    o2 = x2 * c + x1 * s
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


def causal_softmax_ref(
    scores: torch.Tensor, scale: float, q_offset: int = 0, q_len: int = 0
) -> torch.Tensor:
    s = scores.float() * scale
    n_rows, k_len = s.shape[-2], s.shape[-1]
    q_len = q_len or n_rows
    # row r is query position (r % q_len) — supports the GQA grouped layout
    # where rep query-head blocks are folded into the row dimension
    qpos = (torch.arange(n_rows, device=s.device) % q_len).unsqueeze(-1) + q_offset
    kpos = torch.arange(k_len, device=s.device).unsqueeze(0)
    s = s.masked_fill(kpos > qpos, float("-inf"))
    return torch.softmax(s, dim=-1).to(scores.dtype)


# ---------------------------------------------------------------------------
# autograd functions
# ---------------------------------------------------------------------------


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        if _use_hip(x):
            y, invrms = hip_ops().rmsnorm_fwd(x.contiguous(), w.contiguous(), eps)
            ctx.save_for_backward(x, w, invrms)
            ctx.use_hip = True
        else:
            xf = x.float()
            invrms = torch.rsqrt(xf.pow(2).mean(-1) + eps).reshape(-1)
            y = rmsnorm_ref(x, w, eps)
            ctx.save_for_backward(x, w, invrms)
            ctx.use_hip = False
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, invrms = ctx.saved_tensors
        if ctx.use_hip:
            dx, dw = hip_ops().rmsnorm_bwd(dy.contiguous(), x, w, invrms, None)
            return dx, dw, None
        h = x.shape[-1]
        xf, dyf, wf = x.float(), dy.float(), w.float()
        inv = invrms.view(*x.shape[:-1], 1)
        dot = (dyf * wf * xf).sum(-1, keepdim=True)
        dx = dyf * wf * inv - xf * (dot * inv.pow(3) / h)
        dw = (dyf * xf * inv).reshape(-1, h).sum(0)
        return dx.to(x.dtype), dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    return _RMSNorm.apply(x, w, eps)


class _RMSNormAdd(torch.autograd.Function):
    """Fused h = x + resid; y = rmsnorm(h) * w. Two outputs: (y, h) — h is
    the continuing residual stream, so its incoming gradient folds into the
    norm backward in the same kernel pass."""

    @staticmethod
    def forward(ctx, x, resid, w, eps):
        if _use_hip(x):
            h, y, invrms = hip_ops().rmsnorm_add_fwd(
                x.contiguous(), resid.contiguous(), w.contiguous(), eps
            )
            ctx.save_for_backward(h, w, invrms)
            ctx.use_hip = True
            return y, h
        hf = x.float() + resid.float()
        invrms = torch.rsqrt(hf.pow(2).mean(-1) + eps).reshape(-1)
        h = hf.to(x.dtype)
        y = rmsnorm_ref(h, w, eps)
        ctx.save_for_backward(h, w, invrms)
        ctx.use_hip = False
        return y, h

    @staticmethod
    def backward(ctx, dy, dh):
        h, w, invrms = ctx.saved_tensors
        if ctx.use_hip:
            dh_c = dh.contiguous() if dh is not None else None
            dx, dw = hip_ops().rmsnorm_bwd(dy.contiguous(), h, w, invrms, dh_c)
            return dx, dx, dw, None
        hid = h.shape[-1]
        hf, dyf, wf = h.float(), dy.float(), w.float()
        inv = invrms.view(*h.shape[:-1], 1)
        dot = (dyf * wf * hf).sum(-1, keepdim=True)
        dxf = dyf * wf * inv - hf * (dot * inv.pow(3) / hid)
        if dh is not None:
            dxf = dxf + dh.float()
        dw = (dyf * hf * inv).reshape(-1, hid).sum(0)
        dx = dxf.to(h.dtype)
        return dx, dx, dw.to(w.dtype), None


def rmsnorm_add(x: torch.Tensor, resid: torch.Tensor, w: torch.Tensor,
                eps: float = 1e-5):
    """(normed, new_residual) = fused residual-add + RMSNorm."""
    return _RMSNormAdd.apply(x, resid, w, eps)


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate_up):
        ctx.save_for_backward(gate_up)
        if _use_hip(gate_up):
            return hip_ops().swiglu_fwd(gate_up.contiguous())
        return swiglu_ref(gate_up)

    @staticmethod
    def backward(ctx, dy):
        (gate_up,) = ctx.saved_tensors
        if _use_hip(gate_up):
            return hip_ops().swiglu_bwd(dy.contiguous(), gate_up.contiguous())
        gate, up = gate_up.float().chunk(2, dim=-1)
        sig = torch.sigmoid(gate)
        dyf = dy.float()
        dg = dyf * up * sig * (1 + gate * (1 - sig))
        du = dyf * gate * sig
        return torch.cat([dg, du], dim=-1).to(gate_up.dtype)


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    """out = silu(gate_up[..., :I]) * gate_up[..., I:]"""
    return _SwiGLU.apply(gate_up)


def _rope_addressable(x: torch.Tensor) -> torch.Tensor:
    """The oop rope kernel addresses [tokens(strided), heads, head_dim] with
    heads*head_dim contiguous and a uniform token stride; anything else
    (rare) goes through one contiguous copy."""
    hd = x.size(-1)
    if x.stride(-1) != 1 or x.stride(-2) != hd:
        return x.contiguous()
    if x.dim() >= 3:
        tok_stride = x.stride(-3)
        rows = x.size(-3)
        for d in range(x.dim() - 4, -1, -1):
            if x.stride(d) != rows * tok_stride:
                return x.contiguous()
            rows *= x.size(d)
    return x


class _RoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, pos, cos, sin):
        ctx.save_for_backward(pos, cos, sin)
        if _use_hip(x):
            # out-of-place kernel reads the strided view (e.g. a q/k slice
            # of the fused QKV projection) directly — no clone pass
            return hip_ops().rope_rotate_oop(
                _rope_addressable(x), pos.int(), cos, sin, False
            )
        return rope_ref(x, pos, cos, sin)

    @staticmethod
    def backward(ctx, dy):
        pos, cos, sin = ctx.saved_tensors
        if _use_hip(dy):
            dx = hip_ops().rope_rotate_oop(
                _rope_addressable(dy), pos.int(), cos, sin, True
            )
            return dx, None, None, None
        # inverse rotation
        return rope_ref(dy, pos, cos, -sin), None, None, None


def rope_rotate(
    x: torch.Tensor, pos: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor
) -> torch.Tensor:
    """Apply rotate-half RoPE. x: [..., n_tokens, n_heads, head_dim]."""
    return _RoPE.apply(x, pos, cos, sin)


def build_rope_cache(
    max_pos: int, head_dim: int, base: float = 500000.0, device="cpu"
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed fp32 cos/sin tables [max_pos, head_dim/2]
    (on-device trig would turn RoPE VALU-bound — guide Appendix B)."""
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(max_pos, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().to(device), freqs.sin().to(device)


class _CausalSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, scale, q_offset, q_len):
        q_len = q_len or scores.shape[-2]
        if _use_hip(scores):
            probs = scores.contiguous()
            hip_ops().causal_softmax_fwd(probs, q_len, q_offset, scale)
        else:
            probs = causal_softmax_ref(scores, scale, q_offset, q_len)
        ctx.save_for_backward(probs)
        ctx.scale = scale
        return probs

    @staticmethod
    def backward(ctx, dprobs):
        (probs,) = ctx.saved_tensors
        if _use_hip(probs):
            ds = dprobs.contiguous().clone()
            hip_ops().causal_softmax_bwd(ds, probs, ctx.scale)
            return ds, None, None, None
        pf, df = probs.float(), dprobs.float()
        dot = (pf * df).sum(-1, keepdim=True)
        return (pf * (df - dot) * ctx.scale).to(probs.dtype), None, None, None


def causal_softmax(
    scores: torch.Tensor, scale: float, q_offset: int = 0, q_len: int = 0
) -> torch.Tensor:
    """In one fused pass: probs = softmax(scale * scores + causal_mask).

    q_len: the true sequence length when rows fold multiple query-head
    blocks (GQA grouped layout) — row r is query position r % q_len.
    NOTE (GPU path): consumes ``scores`` in place — do not reuse it.
    """
    return _CausalSoftmax.apply(scores, scale, q_offset, q_len)


def flash_attention_ref(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: float
) -> torch.Tensor:
    """fp32 composite reference: q [B,H,S,D], k/v [B,HKV,S,D], causal."""
    B, H, S, D = q.shape
    rep = H // k.shape[1]
    kf = k.float().repeat_interleave(rep, 1)
    vf = v.float().repeat_interleave(rep, 1)
    s_ = torch.matmul(q.float(), kf.transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
    s_ = s_.masked_fill(mask, float("-inf"))
    p = torch.softmax(s_, -1)
    return torch.matmul(p, vf).to(q.dtype)


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        if _use_hip(q):
            # strided [B,H,S,D] views are fine (kernels are stride-aware);
            # only the head dim must be contiguous
            if q.stride(-1) != 1:
                q = q.contiguous()
            if k.stride(-1) != 1:
                k = k.contiguous()
            if v.stride(-1) != 1:
                v = v.contiguous()
            out, lse = hip_ops().flash_attn_fwd(q, k, v, scale)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.scale = scale
            ctx.use_hip = True
            return out
        ctx.use_hip = False
        q32 = q.detach().clone().requires_grad_(True)
        k32 = k.detach().clone().requires_grad_(True)
        v32 = v.detach().clone().requires_grad_(True)
        with torch.enable_grad():
            out = flash_attention_ref(q32, k32, v32, scale)
        ctx.saved_ref = (q32, k32, v32, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        if ctx.use_hip:
            q, k, v, out, lse = ctx.saved_tensors
            dq, dk, dv = hip_ops().flash_attn_bwd(
                q, k, v, out, dout.contiguous(), lse, ctx.scale
            )
            return dq, dk, dv, None
        q32, k32, v32, out = ctx.saved_ref
        torch.autograd.backward(out, dout)
        return q32.grad, k32.grad, v32.grad, None


def flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: float
) -> torch.Tensor:
    """Fused causal GQA attention. q [B,H,S,D], k/v [B,HKV,S,D], D=128,
    S % 64 == 0. Never materializes the S^2 score matrix (hand-written MFMA
    kernels, ops/csrc/attention*.hip)."""
    return _FlashAttention.apply(q, k, v, scale)


class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        n_valid = int((targets != ignore_index).sum())
        grad_scale = 1.0 / max(n_valid, 1)
        if _use_hip(logits):
            work = logits.contiguous()
            losses = hip_ops().cross_entropy_fwd_bwd(
                work, targets.int(), ignore_index, grad_scale, True
            )
            ctx.save_for_backward(work)  # now holds dlogits (pre-scaled)
            ctx.use_hip = True
        else:
            lf = logits.float()
            losses = torch.nn.functional.cross_entropy(
                lf.view(-1, lf.shape[-1]),
                targets.view(-1).long(),
                ignore_index=ignore_index,
                reduction="none",
            )
            dl = torch.softmax(lf, dim=-1)
            t = targets.view(-1)
            valid = t != ignore_index
            onehot = torch.zeros_like(dl.view(-1, dl.shape[-1]))
            onehot[valid, t[valid].long()] = 1.0
            dl = (dl.view(-1, dl.shape[-1]) - onehot) * grad_scale
            dl[~valid] = 0
            ctx.save_for_backward(dl.to(logits.dtype).view_as(logits))
            ctx.use_hip = False
        ctx.n_valid = n_valid
        return losses.sum() * grad_scale

    @staticmethod
    def backward(ctx, dloss):
        (dlogits,) = ctx.saved_tensors
        return dlogits * dloss, None, None


def cross_entropy_loss(
    logits: torch.Tensor, targets: torch.Tensor, ignore_index: int = -100
) -> torch.Tensor:
    """Mean CE over non-ignored tokens. GPU path fuses loss+grad in one
    kernel and reuses the logits buffer for dlogits (no extra V-sized
    allocation). The logits tensor is consumed."""
    return _CrossEntropy.apply(logits, targets, ignore_index)


def fused_adamw_step(
    param_f32: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    param_bf16: Optional[torch.Tensor],
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    grad_scale: float = 1.0,
):
    """Single fused AdamW update; see FusedAdamW for the optimizer class."""
    if param_f32.is_cuda:
        hip_ops().adamw_step(
            param_f32,
            grad.contiguous(),
            exp_avg,
            exp_avg_sq,
            param_bf16,
            lr,
            beta1,
            beta2,
            eps,
            weight_decay,
            step,
            grad_scale,
        )
        return
    # CPU reference
    g = grad.float() * grad_scale
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    param_f32.add_(
        exp_avg / bc1 / denom + weight_decay * param_f32, alpha=-lr
    )
    if param_bf16 is not None:
        param_bf16.copy_(param_f32.to(param_bf16.dtype))
