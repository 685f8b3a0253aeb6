"""Fused flash attention (ops/csrc/attn_kernels.hip).

Correctness-first flash fwd+bwd, one wave per row, D == 64 — hardware-
validated against torch SDPA (tests/test_gpu_attn.py) but not yet
perf-competitive with aotriton, so the models keep torch SDPA.  ROADMAP.md
item 2 is the MFMA-tiled rewrite of the inner loops.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from . import HAVE_EXT, _EXT


class _FusedSDPA(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        B, H, N, D = q.shape
        scale = 1.0 / math.sqrt(D)
        o = torch.empty_like(q)
        lse = torch.empty(B * H * N, dtype=torch.float32, device=q.device)
        _EXT.attn_fwd(q, k, v, o, lse, N, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        B, H, N, D = q.shape
        scale = 1.0 / math.sqrt(D)
        dout = dout.contiguous()
        delta = torch.empty_like(lse)
        dq = torch.empty_like(q)
        dk = torch.empty_like(k)
        dv = torch.empty_like(v)
        _EXT.attn_bwd(q, k, v, o, dout, lse, delta, dq, dk, dv, N, scale,
                      ctx.causal)
        return dq, dk, dv, None


def fused_sdpa(q, k, v, is_causal=False):
    """Like F.scaled_dot_product_attention for [B,H,N,D] with D==64."""
    if (HAVE_EXT and q.is_cuda and q.dtype == torch.bfloat16
            and q.dim() == 4 and q.shape[-1] == 64
            and q.is_contiguous() and k.is_contiguous()
            and v.is_contiguous() and q.shape == k.shape == v.shape):
        return _FusedSDPA.apply(q, k, v, bool(is_causal))
    return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal)
