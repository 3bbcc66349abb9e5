"""nanoGPT-scale GPT-2 style model for the CPU/gloo plumbing config
(BASELINE.json config #1: "nanoGPT DDP world_size=2 on CPU/gloo").

Uses plain torch modules so it runs anywhere (CPU tests, smoke jobs); the
MI355X hot path lives in models/llama.py.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class GPTConfig:
    vocab_size: int = 50304
    block_size: int = 256
    n_layer: int = 6
    n_head: int = 6
    n_embd: int = 384
    dropout: float = 0.0

    @classmethod
    def tiny(cls) -> "GPTConfig":
        return cls(vocab_size=512, block_size=64, n_layer=2, n_head=2, n_embd=64)


class _Block(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.ln1 = nn.LayerNorm(cfg.n_embd)
        self.attn = nn.MultiheadAttention(
            cfg.n_embd, cfg.n_head, dropout=cfg.dropout, batch_first=True
        )
        self.ln2 = nn.LayerNorm(cfg.n_embd)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.n_embd, 4 * cfg.n_embd),
            nn.GELU(),
            nn.Linear(4 * cfg.n_embd, cfg.n_embd),
        )
        mask = torch.triu(torch.ones(cfg.block_size, cfg.block_size, dtype=torch.bool), 1)
        self.register_buffer("causal_mask", mask, persistent=False)

    def forward(self, x):
        h = self.ln1(x)
        S = x.shape[1]
        a, _ = self.attn(h, h, h, attn_mask=self.causal_mask[:S, :S], need_weights=False)
        x = x + a
        x = x + self.mlp(self.ln2(x))
        return x


class NanoGPT(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.pos_emb = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.blocks = nn.ModuleList(_Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f = nn.LayerNorm(cfg.n_embd)
        self.head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)

    def forward(self, idx: torch.Tensor, labels: torch.Tensor = None):
        B, S = idx.shape
        pos = torch.arange(S, device=idx.device)
        x = self.tok_emb(idx) + self.pos_emb(pos)
        for blk in self.blocks:
            x = blk(x)
        logits = self.head(self.ln_f(x))
        if labels is None:
            return logits
        return F.cross_entropy(
            logits.view(-1, self.cfg.vocab_size), labels.view(-1), ignore_index=-100
        )
