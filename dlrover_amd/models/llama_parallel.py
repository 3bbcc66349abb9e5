"""Tensor+pipeline parallel Llama over RCCL.

The reference integrates Megatron for TP/PP and only supplies the checkpoint
engine; our MI355X build owns the parallelism too:
  - TP: attention heads and MLP inner dim split across the TP group; two
    all-reduces per block per direction (tp.py);
  - PP: contiguous block ranges per stage (pp.py GPipe schedule);
  - embeddings + final norm + lm_head live on the first/last stages.

Weight layout per TP rank for the packed projections:
  qkv_proj.weight rows = [my q heads | my k heads | my v heads]
  gate_up_proj.weight rows = [my gate slice | my up slice]
so a full (tp=1) checkpoint reshards by row slicing (megatron engine).
"""

import math
from typing import Optional

import torch
import torch.nn as nn

from dlrover_amd.models.llama import LlamaConfig, RMSNorm
from dlrover_amd.ops import causal_softmax, cross_entropy_loss, rope_rotate, swiglu
from dlrover_amd.ops.api import build_rope_cache
from dlrover_amd.parallel.pgroups import ParallelGroups
from dlrover_amd.parallel.tp import ColumnParallelLinear, RowParallelLinear


class ParallelAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, groups: ParallelGroups):
        super().__init__()
        tp = groups.dims.tp
        if cfg.n_heads % tp or cfg.n_kv_heads % tp:
            raise ValueError("n_heads and n_kv_heads must divide tp")
        self.cfg = cfg
        self.groups = groups
        self.nh = cfg.n_heads // tp
        self.nkv = cfg.n_kv_heads // tp
        hd = cfg.head_dim
        self.qkv_proj = ColumnParallelLinear(
            cfg.hidden_size, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd, tp,
            groups.tp_group,
        )
        self.o_proj = RowParallelLinear(
            cfg.n_heads * hd, cfg.hidden_size, tp, groups.tp_group
        )

    def forward(self, x, pos, cos, sin):
        cfg = self.cfg
        B, S, _ = x.shape
        hd, nh, nkv = cfg.head_dim, self.nh, self.nkv
        qkv = self.qkv_proj(x)
        q, k, v = qkv.split([nh * hd, nkv * hd, nkv * hd], dim=-1)
        q = rope_rotate(q.view(B, S, nh, hd), pos, cos, sin)
        k = rope_rotate(k.view(B, S, nkv, hd), pos, cos, sin)
        v = v.view(B, S, nkv, hd)
        if x.is_cuda and hd == 128 and S % 64 == 0:
            # hand-written MFMA flash path (same conditions as llama.py —
            # the kernels are GQA/stride-aware, per-TP-shard head counts)
            from dlrover_amd.ops import flash_attention

            out = flash_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                1.0 / math.sqrt(hd),
            )
            out = out.transpose(1, 2).reshape(B, S, nh * hd)
            return self.o_proj(out)
        rep = nh // nkv
        q = q.view(B, S, nkv, rep, hd).permute(0, 2, 3, 1, 4).reshape(
            B, nkv, rep * S, hd
        )
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        scores = torch.matmul(q, k.transpose(-1, -2))
        probs = causal_softmax(scores, scale=1.0 / math.sqrt(hd), q_len=S)
        out = torch.matmul(probs, v)
        out = out.view(B, nkv, rep, S, hd).permute(0, 3, 1, 2, 4).reshape(
            B, S, nh * hd
        )
        return self.o_proj(out)


class ParallelMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, groups: ParallelGroups):
        super().__init__()
        tp = groups.dims.tp
        self.gate_up_proj = ColumnParallelLinear(
            cfg.hidden_size, 2 * cfg.intermediate_size, tp, groups.tp_group
        )
        self.down_proj = RowParallelLinear(
            cfg.intermediate_size, cfg.hidden_size, tp, groups.tp_group
        )

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_up_proj(x)))


class ParallelBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, groups: ParallelGroups):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.attn = ParallelAttention(cfg, groups)
        self.mlp_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.mlp = ParallelMLP(cfg, groups)

    def forward(self, x, pos, cos, sin):
        x = x + self.attn(self.attn_norm(x), pos, cos, sin)
        x = x + self.mlp(self.mlp_norm(x))
        return x


def stage_layer_range(n_layers: int, pp_rank: int, pp: int):
    per = n_layers // pp
    extra = n_layers % pp
    start = pp_rank * per + min(pp_rank, extra)
    end = start + per + (1 if pp_rank < extra else 0)
    return start, end


class LlamaStage(nn.Module):
    """One pipeline stage: [embed]? + blocks[lo:hi] + [norm + lm_head]?."""

    def __init__(self, cfg: LlamaConfig, groups: ParallelGroups):
        super().__init__()
        self.cfg = cfg
        self.groups = groups
        lo, hi = stage_layer_range(cfg.n_layers, groups.pp_rank, groups.dims.pp)
        self.layer_range = (lo, hi)
        self.embed = (
            nn.Embedding(cfg.vocab_size, cfg.hidden_size)
            if groups.is_first_stage
            else None
        )
        self.blocks = nn.ModuleList(
            ParallelBlock(cfg, groups) for _ in range(hi - lo)
        )
        self.final_norm = (
            RMSNorm(cfg.hidden_size, cfg.norm_eps) if groups.is_last_stage else None
        )
        self.lm_head = (
            nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
            if groups.is_last_stage
            else None
        )
        cos, sin = build_rope_cache(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.apply(self._init)

    def _apply(self, fn, recurse=True):
        super()._apply(fn, recurse)
        self.rope_cos = self.rope_cos.float()
        self.rope_sin = self.rope_sin.float()
        return self

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, ColumnParallelLinear, RowParallelLinear)):
            nn.init.normal_(m.weight, std=0.02)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, x, labels: Optional[torch.Tensor] = None):
        """x: input_ids on the first stage, hidden states elsewhere."""
        S = x.shape[1]
        pos = torch.arange(S, device=x.device, dtype=torch.int32)
        if self.embed is not None:
            x = self.embed(x)
        for blk in self.blocks:
            x = blk(x, pos, self.rope_cos, self.rope_sin)
        if self.final_norm is None:
            return x
        x = self.final_norm(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        return cross_entropy_loss(
            logits.reshape(-1, self.cfg.vocab_size), labels.reshape(-1)
        )
