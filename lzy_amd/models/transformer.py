"""Decoder-only transformer LM — clean-room torch implementation.

A compact GPT-style architecture for workflow examples and gang-DDP
smoke runs (the reference framework ships no models at all; this and
resnet.py exist so the runtime's multi-GPU paths have realistic
workloads without torchvision/transformers checkpoints).  bf16-friendly:
pre-norm blocks, SDPA attention (rocBLAS/MIOpen-backed on ROCm).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


class Block(nn.Module):
    def __init__(self, d_model: int, n_heads: int, mlp_ratio: int = 4):
        super().__init__()
        self.n_heads = n_heads
        self.ln1 = nn.LayerNorm(d_model)
        self.qkv = nn.Linear(d_model, 3 * d_model, bias=False)
        self.proj = nn.Linear(d_model, d_model, bias=False)
        self.ln2 = nn.LayerNorm(d_model)
        self.mlp = nn.Sequential(
            nn.Linear(d_model, mlp_ratio * d_model, bias=False),
            nn.GELU(),
            nn.Linear(mlp_ratio * d_model, d_model, bias=False),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, s, d = x.shape
        h = self.ln1(x)
        q, k, v = self.qkv(h).chunk(3, dim=-1)

        def heads(t):
            return t.view(b, s, self.n_heads, d // self.n_heads).transpose(1, 2)

        a = F.scaled_dot_product_attention(
            heads(q), heads(k), heads(v), is_causal=True
        )
        x = x + self.proj(a.transpose(1, 2).reshape(b, s, d))
        return x + self.mlp(self.ln2(x))


class TransformerLM(nn.Module):
    """Small decoder-only LM: tied embedding/unembedding, learned
    positions, causal SDPA attention."""

    def __init__(self, vocab: int = 32000, d_model: int = 512,
                 n_layers: int = 8, n_heads: int = 8, max_seq: int = 2048):
        super().__init__()
        self.tok = nn.Embedding(vocab, d_model)
        self.pos = nn.Embedding(max_seq, d_model)
        self.blocks = nn.ModuleList(
            Block(d_model, n_heads) for _ in range(n_layers)
        )
        self.ln_f = nn.LayerNorm(d_model)
        for p in self.parameters():
            if p.dim() > 1:
                nn.init.normal_(p, std=0.02 / math.sqrt(2 * n_layers))

    def forward(self, idx: torch.Tensor) -> torch.Tensor:
        b, s = idx.shape
        x = self.tok(idx) + self.pos(torch.arange(s, device=idx.device))
        for blk in self.blocks:
            x = blk(x)
        return self.ln_f(x) @ self.tok.weight.t()  # tied unembedding

    def loss(self, idx: torch.Tensor) -> torch.Tensor:
        logits = self(idx[:, :-1])
        return F.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(),
            idx[:, 1:].reshape(-1),
        )
