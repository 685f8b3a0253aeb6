"""GPT-2-small (Radford et al. 2019) for BASELINE config 5
(AsySG-InCon stale-gradient path)."""

from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.attn import fused_sdpa_qkv
from ..ops.ce import fused_cross_entropy
from ..ops.linear import FusedLinear
from ..ops.ln import FusedLayerNorm


class CausalBlock(nn.Module):
    def __init__(self, dim, heads):
        super().__init__()
        self.n1 = FusedLayerNorm(dim)
        self.qkv = FusedLinear(dim, 3 * dim)
        self.proj = FusedLinear(dim, dim)
        self.n2 = FusedLayerNorm(dim)
        self.fc1 = FusedLinear(dim, 4 * dim)
        self.fc2 = FusedLinear(4 * dim, dim)
        self.heads = heads

    def forward(self, x):
        B, T, D = x.shape
        qkv = self.qkv(self.n1(x)).view(B, T, 3, self.heads, D // self.heads)
        y = fused_sdpa_qkv(qkv, is_causal=True)  # [B, H, T, Dh]
        y = y.transpose(1, 2).reshape(B, T, D)
        x = x + self.proj(y)
        x = x + self.fc2(F.gelu(self.fc1(self.n2(x))))
        return x


class GPT2(nn.Module):
    def __init__(self, vocab=50257, ctx=1024, dim=768, depth=12, heads=12):
        super().__init__()
        self.wte = nn.Embedding(vocab, dim)
        self.wpe = nn.Embedding(ctx, dim)
        self.blocks = nn.ModuleList(CausalBlock(dim, heads)
                                    for _ in range(depth))
        self.norm = FusedLayerNorm(dim)
        self.ctx = ctx
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.02)

    def forward(self, idx):
        B, T = idx.shape
        pos = torch.arange(T, device=idx.device)
        x = self.wte(idx) + self.wpe(pos)[None]
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        # tied output head
        return x @ self.wte.weight.t()

    def loss(self, idx, targets):
        # CE directly on the bf16 logits: torch's log_softmax accumulates in
        # fp32 internally; materializing a fp32 [tokens, vocab] copy cost
        # ~12 ms/step at batch 96 (9.9 GB of extra traffic).
        logits = self.forward(idx)
        return fused_cross_entropy(
            logits.reshape(-1, logits.size(-1)).contiguous(),
            targets.reshape(-1))


def gpt2_small(vocab=50304, ctx=1024):
    # vocab padded 50257 -> 50304 (multiple of 64) for GEMM tile alignment
    return GPT2(vocab=vocab, ctx=ctx)
