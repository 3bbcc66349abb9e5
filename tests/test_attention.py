"""Attention module vs a naive per-head fp32 oracle — guards the zero-copy
GQA grouped layout (q folded to [B, nkv, rep*S, D])."""

import math

import torch

from dlrover_amd.models.llama import Attention, LlamaConfig
from dlrover_amd.ops.api import build_rope_cache, rope_ref


def naive_attention(q, k, v, nkv):
    """q [B,S,H,D], k/v [B,S,Hkv,D] -> [B,S,H*D], straightforward loops."""
    B, S, H, D = q.shape
    rep = H // nkv
    out = torch.zeros(B, S, H, D)
    scale = 1.0 / math.sqrt(D)
    mask = torch.tril(torch.ones(S, S, dtype=torch.bool))
    for h in range(H):
        g = h // rep
        s = torch.einsum("bsd,btd->bst", q[:, :, h].float(), k[:, :, g].float())
        s = (s * scale).masked_fill(~mask, float("-inf"))
        p = torch.softmax(s, dim=-1)
        out[:, :, h] = torch.einsum("bst,btd->bsd", p, v[:, :, g].float())
    return out.reshape(B, S, H * D)


def test_attention_matches_naive_oracle():
    torch.manual_seed(0)
    cfg = LlamaConfig.tiny()  # H=4, Hkv=2, D=32
    attn = Attention(cfg)
    B, S = 2, 16
    x = torch.randn(B, S, cfg.hidden_size)
    pos = torch.arange(S, dtype=torch.int32)
    cos, sin = build_rope_cache(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)

    out = attn(x, pos, cos, sin)

    # oracle path: same projections, rope_ref, naive per-head attention
    with torch.no_grad():
        qkv = attn.qkv_proj(x)
        hd, nh, nkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
        q, k, v = qkv.split([nh * hd, nkv * hd, nkv * hd], dim=-1)
        q = rope_ref(q.view(B, S, nh, hd), pos, cos, sin)
        k = rope_ref(k.view(B, S, nkv, hd), pos, cos, sin)
        v = v.view(B, S, nkv, hd)
        ref = attn.o_proj(naive_attention(q, k, v, nkv).to(x.dtype))

    torch.testing.assert_close(out, ref, rtol=1e-3, atol=1e-4)


def test_attention_backward_flows():
    torch.manual_seed(1)
    cfg = LlamaConfig.tiny()
    attn = Attention(cfg)
    x = torch.randn(1, 8, cfg.hidden_size, requires_grad=True)
    pos = torch.arange(8, dtype=torch.int32)
    cos, sin = build_rope_cache(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
    attn(x, pos, cos, sin).sum().backward()
    assert x.grad is not None and x.grad.isfinite().all()
    assert attn.qkv_proj.weight.grad.isfinite().all()
