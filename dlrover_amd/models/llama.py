"""Llama-3 style decoder built on the dlrover_amd HIP ops.

The flagship training model for the benchmark configs (BASELINE.json:
"Llama-3 8B FSDP bf16 on 8xMI355X"). MI355X-native choices:
  - all projections are packed bf16 linears (QKV in one GEMM, gate+up in one
    GEMM) so hipBLASLt sees few, large GEMMs;
  - RMSNorm / RoPE / SwiGLU / causal-softmax / cross-entropy are our fused
    HIP kernels (dlrover_amd.ops), eliminating eager elementwise passes;
  - attention v1 = hipBLASLt batched GEMM (QK^T, PV) + fused
    scale+mask+softmax kernel; 288 GB HBM3E holds the S^2 scores at the
    bench shapes, trading memory for library-GEMM MFMA throughput.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn

from dlrover_amd.ops import (
    causal_softmax,
    cross_entropy_loss,
    rmsnorm,
    rmsnorm_add,
    rope_rotate,
    swiglu,
)
from dlrover_amd.ops.api import build_rope_cache


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    max_seq_len: int = 8192
    rope_base: float = 500000.0
    norm_eps: float = 1e-5
    tie_embeddings: bool = False
    # "auto": hand-written flash kernel on GPU when D==128 and S%64==0,
    # composite (hipBLASLt GEMM + fused softmax) otherwise
    attn_impl: str = "auto"
    # qkv projection bias (Qwen2-style checkpoints; Llama-3 uses none)
    attn_bias: bool = False
    # recompute each block in backward instead of saving activations
    # (non-reentrant torch.utils.checkpoint — composes with FSDP2); trades
    # ~30% step time for ~n_layers x less activation memory, the right move
    # for long-seq / big-batch on 288 GB HBM3E only when activations would
    # not otherwise fit
    activation_checkpointing: bool = False

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.n_heads

    @classmethod
    def llama3_8b(cls, max_seq_len: int = 8192) -> "LlamaConfig":
        return cls(max_seq_len=max_seq_len)

    @classmethod
    def tiny(cls) -> "LlamaConfig":
        """CPU-test scale."""
        return cls(
            vocab_size=512,
            hidden_size=128,
            intermediate_size=256,
            n_layers=2,
            n_heads=4,
            n_kv_heads=2,
            max_seq_len=128,
            rope_base=10000.0,
        )

    @classmethod
    def qwen2_7b(cls, max_seq_len: int = 8192) -> "LlamaConfig":
        """Qwen2-7B geometry: Llama architecture + qkv bias + tied-free
        151k vocab (the HIP kernel stack is identical)."""
        return cls(
            vocab_size=152064,
            hidden_size=3584,
            intermediate_size=18944,
            n_layers=28,
            n_heads=28,
            n_kv_heads=4,
            max_seq_len=max_seq_len,
            rope_base=1000000.0,
            attn_bias=True,
        )

    @classmethod
    def small_1b(cls, max_seq_len: int = 4096) -> "LlamaConfig":
        return cls(
            vocab_size=32000,
            hidden_size=2048,
            intermediate_size=5632,
            n_layers=22,
            n_heads=32,
            n_kv_heads=8,
            max_seq_len=max_seq_len,
        )


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.eps = eps

    def forward(self, x):
        return rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        d, hd = cfg.hidden_size, cfg.head_dim
        self.qkv_proj = nn.Linear(
            d, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd, bias=cfg.attn_bias
        )
        self.o_proj = nn.Linear(cfg.n_heads * hd, d, bias=False)

    def forward(self, x, pos, cos, sin):
        cfg = self.cfg
        B, S, _ = x.shape
        hd, nh, nkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
        qkv = self.qkv_proj(x)  # one hipBLASLt GEMM
        q, k, v = qkv.split([nh * hd, nkv * hd, nkv * hd], dim=-1)
        q = q.view(B, S, nh, hd)
        k = k.view(B, S, nkv, hd)
        v = v.view(B, S, nkv, hd)
        q = rope_rotate(q, pos, cos, sin)
        k = rope_rotate(k, pos, cos, sin)
        use_flash = (
            cfg.attn_impl in ("auto", "flash")
            and x.is_cuda
            and hd == 128
            and S % 64 == 0
        )
        if use_flash:
            from dlrover_amd.ops import flash_attention

            # permuted VIEWS: the stride-aware kernels read [B,S,H,D] storage
            # directly — zero transpose copies in or out
            out = flash_attention(
                q.transpose(1, 2),
                k.transpose(1, 2),
                v.transpose(1, 2),
                1.0 / math.sqrt(hd),
            )
            out = out.transpose(1, 2).reshape(B, S, nh * hd)
            return self.o_proj(out)
        # grouped-query attention with ZERO kv copies: fold the query heads
        # of each kv group into the row dimension — q [B, nkv, rep*S, D]
        # against k/v [B, nkv, S, D]. (repeat_interleave showed up as 11% of
        # step time as __amd_rocclr_copyBuffer; see profiles/.)
        rep = nh // nkv
        q = q.view(B, S, nkv, rep, hd).permute(0, 2, 3, 1, 4)  # [B,nkv,rep,S,D]
        q = q.reshape(B, nkv, rep * S, hd)
        k = k.transpose(1, 2)  # [B, nkv, S, D]
        v = v.transpose(1, 2)
        scores = torch.matmul(q, k.transpose(-1, -2))  # [B,nkv,rep*S,S] MFMA
        # causal pattern: row r of the rep*S rows is query position r % S —
        # exactly the kernel's q_len=S row mapping
        probs = causal_softmax(scores, scale=1.0 / math.sqrt(hd), q_len=S)
        out = torch.matmul(probs, v)  # [B, nkv, rep*S, D] MFMA
        out = out.view(B, nkv, rep, S, hd).permute(0, 3, 1, 2, 4)
        out = out.reshape(B, S, nh * hd)
        return self.o_proj(out)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_up_proj = nn.Linear(
            cfg.hidden_size, 2 * cfg.intermediate_size, bias=False
        )
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_up_proj(x)))


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.mlp_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.mlp = MLP(cfg)

    def forward(self, x, pos, cos, sin, resid=None):
        """Two forms. Plain (resid=None): the classic pre-norm block.
        Residual-stream (resid given): x is the previous sublayer's BRANCH
        output (None on the first block); both residual adds fuse into the
        following RMSNorm kernel; returns (branch, resid). Must be entered
        through __call__ so FSDP's unshard hooks fire."""
        if resid is None:
            x = x + self.attn(self.attn_norm(x), pos, cos, sin)
            x = x + self.mlp(self.mlp_norm(x))
            return x
        if x is None:
            normed = self.attn_norm(resid)
        else:
            normed, resid = rmsnorm_add(x, resid, self.attn_norm.weight,
                                        self.attn_norm.eps)
        a = self.attn(normed, pos, cos, sin)
        normed, resid = rmsnorm_add(a, resid, self.mlp_norm.weight,
                                    self.mlp_norm.eps)
        return self.mlp(normed), resid


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed.weight
        cos, sin = build_rope_cache(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.apply(self._init_weights)
        # scaled init for residual-out projections (depth-aware)
        std = 0.02 / math.sqrt(2 * cfg.n_layers)
        for blk in self.blocks:
            nn.init.normal_(blk.attn.o_proj.weight, std=std)
            nn.init.normal_(blk.mlp.down_proj.weight, std=std)

    def _apply(self, fn, recurse=True):
        # .bfloat16()/.half() must not narrow the RoPE tables: the HIP kernel
        # (and the reference impl) take fp32 cos/sin
        super()._apply(fn, recurse)
        self.rope_cos = self.rope_cos.float()
        self.rope_sin = self.rope_sin.float()
        return self

    @staticmethod
    def _init_weights(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor = None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device, dtype=torch.int32)
        x = self.embed(input_ids)
        ckpt = (self.cfg.activation_checkpointing and self.training
                and torch.is_grad_enabled())
        if ckpt:
            from torch.utils.checkpoint import checkpoint
        if input_ids.is_cuda:
            # fused residual-stream form (identical math, fewer HBM passes);
            # blocks entered via __call__ so FSDP unshard hooks fire
            branch, resid = None, x
            for blk in self.blocks:
                if ckpt:
                    branch, resid = checkpoint(
                        lambda b, r, _blk=blk: _blk(
                            b, pos, self.rope_cos, self.rope_sin, resid=r),
                        branch, resid, use_reentrant=False,
                    )
                else:
                    branch, resid = blk(
                        branch, pos, self.rope_cos, self.rope_sin, resid=resid
                    )
            x, _ = rmsnorm_add(branch, resid, self.final_norm.weight,
                               self.final_norm.eps)
        else:
            for blk in self.blocks:
                if ckpt:
                    x = checkpoint(
                        lambda h, _blk=blk: _blk(
                            h, pos, self.rope_cos, self.rope_sin),
                        x, use_reentrant=False,
                    )
                else:
                    x = blk(x, pos, self.rope_cos, self.rope_sin)
            x = self.final_norm(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        return cross_entropy_loss(
            logits.view(-1, self.cfg.vocab_size), labels.view(-1)
        )

    def num_params(self) -> int:
        n = sum(p.numel() for p in self.parameters())
        if self.cfg.tie_embeddings:
            n -= self.embed.weight.numel()
        return n
