"""CPU-path numerics: custom autograd backward formulas vs torch autograd
through an independent fp32 composition of the same op."""

import math

import pytest
import torch

from dlrover_amd.ops.api import (
    build_rope_cache,
    causal_softmax,
    causal_softmax_ref,
    cross_entropy_loss,
    fused_adamw_step,
    rmsnorm,
    rope_ref,
    rope_rotate,
    swiglu,
)

torch.manual_seed(0)


def _autograd_oracle(fn, *inputs):
    ins = [i.detach().clone().requires_grad_(i.is_floating_point()) for i in inputs]
    out = fn(*ins)
    out.sum().backward()
    return out.detach(), [i.grad for i in ins]


def test_rmsnorm_fwd_bwd_matches_autograd():
    x = torch.randn(4, 6, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y = rmsnorm(x, w, 1e-5)
    y.sum().backward()

    def ref(x_, w_):
        inv = torch.rsqrt(x_.pow(2).mean(-1, keepdim=True) + 1e-5)
        return x_ * inv * w_

    y2, (gx, gw) = _autograd_oracle(ref, x.detach(), w.detach())
    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(x.grad, gx, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(w.grad, gw, rtol=1e-4, atol=1e-4)


def test_swiglu_fwd_bwd_matches_autograd():
    gu = torch.randn(8, 32, requires_grad=True)
    out = swiglu(gu)
    out.sum().backward()

    def ref(gu_):
        g, u = gu_.chunk(2, -1)
        return torch.nn.functional.silu(g) * u

    o2, (ggu,) = _autograd_oracle(ref, gu.detach())
    torch.testing.assert_close(out, o2, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(gu.grad, ggu, rtol=1e-4, atol=1e-5)


def test_rope_roundtrip_and_grad():
    cos, sin = build_rope_cache(32, 16, base=10000.0)
    x = torch.randn(2, 8, 4, 16, requires_grad=True)  # [B, T, H, D]
    pos = torch.arange(8, dtype=torch.int32)
    y = rope_rotate(x, pos, cos, sin)
    # rotation preserves pair norms
    xf, yf = x.detach().float(), y.float()
    nx = xf[..., :8].pow(2) + xf[..., 8:].pow(2)
    ny = yf[..., :8].pow(2) + yf[..., 8:].pow(2)
    torch.testing.assert_close(nx, ny, rtol=1e-4, atol=1e-5)
    # gradient: rotation is orthogonal, so grad = inverse rotation of dy
    y.sum().backward()
    ones = torch.ones_like(x)
    expected = rope_ref(ones, pos, cos, -sin)
    torch.testing.assert_close(x.grad, expected, rtol=1e-4, atol=1e-5)


def test_causal_softmax_matches_masked_softmax():
    s = torch.randn(2, 3, 8, 8, requires_grad=True)
    p = causal_softmax(s, scale=0.5)
    assert torch.all(p[..., 0, 1:] == 0)  # causal zeros
    torch.testing.assert_close(
        p.sum(-1), torch.ones(2, 3, 8), rtol=1e-4, atol=1e-5
    )
    p.pow(2).sum().backward()

    def ref(s_):
        return causal_softmax_ref(s_, 0.5)

    s2 = s.detach().clone().requires_grad_(True)
    ref(s2).pow(2).sum().backward()
    torch.testing.assert_close(s.grad, s2.grad, rtol=1e-4, atol=1e-5)


def test_cross_entropy_matches_torch():
    logits = torch.randn(12, 37, requires_grad=True)
    targets = torch.randint(0, 37, (12,))
    targets[3] = -100
    loss = cross_entropy_loss(logits, targets)
    loss.backward()
    l2 = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(l2, targets.long(), ignore_index=-100)
    ref.backward()
    torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(logits.grad, l2.grad, rtol=1e-4, atol=1e-6)


def test_fused_adamw_matches_torch_adamw():
    torch.manual_seed(1)
    p_ref = torch.randn(33)
    p_mine = p_ref.clone()
    m = torch.zeros(33)
    v = torch.zeros(33)
    ref_param = p_ref.clone().requires_grad_(True)
    opt = torch.optim.AdamW(
        [ref_param], lr=1e-2, betas=(0.9, 0.99), eps=1e-8, weight_decay=0.1
    )
    for step in range(1, 6):
        g = torch.randn(33)
        ref_param.grad = g.clone()
        opt.step()
        fused_adamw_step(
            p_mine, g, m, v, None, 1e-2, 0.9, 0.99, 1e-8, 0.1, step
        )
    torch.testing.assert_close(p_mine, ref_param.detach(), rtol=1e-5, atol=1e-6)


def test_fused_adamw_optimizer_class():
    from dlrover_amd.ops import FusedAdamW

    torch.manual_seed(2)
    model = torch.nn.Linear(8, 8)
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    x = torch.randn(4, 8)
    for _ in range(3):
        loss = model(x).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
    st = opt.state[model.weight]
    assert st["step"] == 3 and "master_param" in st


def test_rmsnorm_add_matches_unfused():
    from dlrover_amd.ops import rmsnorm_add

    torch.manual_seed(3)
    x = torch.randn(4, 6, 64, requires_grad=True)
    r = torch.randn(4, 6, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y, h = rmsnorm_add(x, r, w, 1e-5)
    # h is used downstream too: grads flow through BOTH outputs
    (y.pow(2).sum() + h.sum() * 0.5).backward()

    x2 = x.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    h2 = x2 + r2
    inv = torch.rsqrt(h2.pow(2).mean(-1, keepdim=True) + 1e-5)
    y2 = h2 * inv * w2
    (y2.pow(2).sum() + h2.sum() * 0.5).backward()
    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(h, h2, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(r.grad, r2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(w.grad, w2.grad, rtol=1e-4, atol=1e-4)
