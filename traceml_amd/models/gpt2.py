"""GPT-2 decoder (in-repo, random init) — the lightning_minimal workload
(BASELINE config 5). Standard pre-LN transformer with learned positions;
sizes: gpt2 (124M) default, or scaled-down for CPU tests."""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


class Block(nn.Module):
    def __init__(self, d_model: int, n_heads: int):
        super().__init__()
        self.ln1 = nn.LayerNorm(d_model)
        self.attn = nn.MultiheadAttention(d_model, n_heads, batch_first=True)
        self.ln2 = nn.LayerNorm(d_model)
        self.mlp = nn.Sequential(
            nn.Linear(d_model, 4 * d_model),
            nn.GELU(),
            nn.Linear(4 * d_model, d_model),
        )

    def forward(self, x, attn_mask):
        h = self.ln1(x)
        a, _ = self.attn(h, h, h, attn_mask=attn_mask, need_weights=False)
        x = x + a
        return x + self.mlp(self.ln2(x))


class GPT2(nn.Module):
    def __init__(
        self,
        vocab_size: int = 50257,
        n_layers: int = 12,
        d_model: int = 768,
        n_heads: int = 12,
        max_seq: int = 1024,
    ):
        super().__init__()
        self.tok = nn.Embedding(vocab_size, d_model)
        self.pos = nn.Embedding(max_seq, d_model)
        self.blocks = nn.ModuleList(Block(d_model, n_heads) for _ in range(n_layers))
        self.ln_f = nn.LayerNorm(d_model)
        self.head = nn.Linear(d_model, vocab_size, bias=False)
        self.head.weight = self.tok.weight  # tied
        self.max_seq = max_seq
        for p in self.parameters():
            if p.dim() > 1:
                nn.init.normal_(p, std=0.02 / math.sqrt(2 * n_layers))

    def forward(self, input_ids, labels=None):
        b, t = input_ids.shape
        pos = torch.arange(t, device=input_ids.device)
        x = self.tok(input_ids) + self.pos(pos)
        mask = torch.triu(
            torch.full((t, t), float("-inf"), device=input_ids.device), diagonal=1
        )
        for block in self.blocks:
            x = block(x, mask)
        logits = self.head(self.ln_f(x))
        if labels is not None:
            loss = F.cross_entropy(
                logits[:, :-1].reshape(-1, logits.size(-1)),
                labels[:, 1:].reshape(-1),
            )
            return {"loss": loss, "logits": logits}
        return {"logits": logits}


def gpt2_small() -> GPT2:
    return GPT2()


def gpt2_tiny() -> GPT2:
    """CPU-test scale."""
    return GPT2(vocab_size=512, n_layers=2, d_model=64, n_heads=4, max_seq=128)
