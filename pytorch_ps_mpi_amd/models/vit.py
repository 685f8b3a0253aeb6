"""ViT-B/16 (Dosovitskiy et al. 2020) for BASELINE config 4."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.attn import fused_sdpa_qkv
from ..ops.linear import FusedLinear
from ..ops.ln import FusedLayerNorm


class Block(nn.Module):
    def __init__(self, dim, heads, mlp_ratio=4.0):
        super().__init__()
        self.n1 = FusedLayerNorm(dim)
        self.qkv = FusedLinear(dim, dim * 3)
        self.proj = FusedLinear(dim, dim)
        self.heads = heads
        self.n2 = FusedLayerNorm(dim)
        h = int(dim * mlp_ratio)
        self.fc1 = FusedLinear(dim, h)
        self.fc2 = FusedLinear(h, dim)

    def forward(self, x):
        B, N, D = x.shape
        qkv = self.qkv(self.n1(x)).view(B, N, 3, self.heads, D // self.heads)
        y = fused_sdpa_qkv(qkv)  # [B, H, N, Dh]
        y = y.transpose(1, 2).reshape(B, N, D)
        x = x + self.proj(y)
        x = x + self.fc2(F.gelu(self.fc1(self.n2(x))))
        return x


class ViT(nn.Module):
    def __init__(self, img=224, patch=16, dim=768, depth=12, heads=12,
                 num_classes=1000):
        super().__init__()
        # patch embedding as unfold + Linear: with stride == kernel the
        # "conv" is a pure reshape + GEMM, which keeps it on hipBLASLt
        # (MIOpen's stride-16 bf16 conv fell back to naive/im2col kernels —
        # 60% of the step in the round-1 profile)
        self.patch = patch
        self.patch_embed = nn.Linear(3 * patch * patch, dim)
        n = (img // patch) ** 2
        self.cls = nn.Parameter(torch.zeros(1, 1, dim))
        self.pos = nn.Parameter(torch.randn(1, n + 1, dim) * 0.02)
        self.blocks = nn.ModuleList(Block(dim, heads) for _ in range(depth))
        self.norm = FusedLayerNorm(dim)
        self.head = nn.Linear(dim, num_classes)

    def _patchify(self, x):
        B, C, H, W = x.shape
        p = self.patch
        x = x.view(B, C, H // p, p, W // p, p)
        x = x.permute(0, 2, 4, 1, 3, 5).reshape(B, (H // p) * (W // p),
                                                C * p * p)
        return x

    def forward(self, x):
        x = self.patch_embed(self._patchify(x))
        cls = self.cls.expand(x.shape[0], -1, -1)
        x = torch.cat([cls, x], dim=1) + self.pos
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.norm(x)[:, 0])


def vit_b16(num_classes=1000):
    return ViT(dim=768, depth=12, heads=12, num_classes=num_classes)
