"""Fused flash attention (MFMA-tiled, ops/csrc/mfma_attn_kernels.hip).

Two in-tree implementations, selected by PS_AMD_ATTN:
  "mfma" (default) — 16x16x32-MFMA-tiled flash fwd+bwd, 128-row blocks,
                     transposed-operand LDS staging (round 2);
  "ref"            — the one-wave-per-row correctness kernels
                     (attn_kernels.hip) — the on-GPU oracle;
  "torch"          — torch SDPA (aotriton).

Shape contract for the fused paths: [B, H, N, D] bf16 contiguous, D == 64
(GPT-2-small and ViT-B/16 head dim).  Anything else falls back to torch
SDPA.
"""

from __future__ import annotations

import math
import os

import torch
import torch.nn.functional as F

from . import HAVE_EXT, _EXT


def _impl():
    return os.environ.get("PS_AMD_ATTN", "mfma")


class _FusedSDPA(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, use_mfma):
        B, H, N, D = q.shape
        scale = 1.0 / math.sqrt(D)
        # explicit shape: empty_like would inherit q's (possibly strided)
        # layout, and o is always written contiguous
        o = torch.empty((B, H, N, D), dtype=q.dtype, device=q.device)
        lse = torch.empty(B * H * N, dtype=torch.float32, device=q.device)
        if use_mfma:
            _EXT.fa_fwd(q, k, v, o, lse, N, scale, causal)
        else:
            _EXT.attn_fwd(q, k, v, o, lse, N, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.use_mfma = use_mfma
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        B, H, N, D = q.shape
        scale = 1.0 / math.sqrt(D)
        dout = dout.contiguous()
        delta = torch.empty_like(lse)
        dq = torch.empty((B, H, N, D), dtype=q.dtype, device=q.device)
        dk = torch.empty_like(dq)
        dv = torch.empty_like(dq)
        if ctx.use_mfma:
            _EXT.fa_bwd(q, k, v, o, dout, lse, delta, dq, dk, dv, N, scale,
                        ctx.causal)
        else:
            _EXT.attn_bwd(q, k, v, o, dout, lse, delta, dq, dk, dv, N, scale,
                          ctx.causal)
        return dq, dk, dv, None, None


class _FusedSDPAQkv(torch.autograd.Function):
    """Attention on a FUSED [B,T,3,H,D] qkv tensor: forward slices q/k/v as
    strided views (no copies); backward writes dq/dk/dv STRIDED into one
    dqkv buffer, so the gradient of the qkv projection arrives assembled —
    autograd's CatArrayBatchedCopy pass (~2 ms/step on GPT-2-small,
    profiles/gpt2_steady_r02.md) disappears."""

    @staticmethod
    def forward(ctx, qkv, causal):
        B, T, three, H, D = qkv.shape
        q, k, v = (qkv[:, :, i].permute(0, 2, 1, 3) for i in range(3))
        scale = 1.0 / math.sqrt(D)
        o = torch.empty((B, H, T, D), dtype=qkv.dtype, device=qkv.device)
        lse = torch.empty(B * H * T, dtype=torch.float32, device=qkv.device)
        _EXT.fa_fwd(q, k, v, o, lse, T, scale, causal)
        ctx.save_for_backward(qkv, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, dout):
        qkv, o, lse = ctx.saved_tensors
        B, T, three, H, D = qkv.shape
        q, k, v = (qkv[:, :, i].permute(0, 2, 1, 3) for i in range(3))
        scale = 1.0 / math.sqrt(D)
        dout = dout.contiguous()
        delta = torch.empty_like(lse)
        dqkv = torch.empty_like(qkv)
        dq, dk, dv = (dqkv[:, :, i].permute(0, 2, 1, 3) for i in range(3))
        _EXT.fa_bwd(q, k, v, o, dout, lse, delta, dq, dk, dv, T, scale,
                    ctx.causal)
        return dqkv, None


def fused_sdpa_qkv(qkv, is_causal=False):
    """Attention for a fused qkv projection output [B, T, 3, H, D] (D==64).
    Returns [B, H, T, D].  Falls back to slicing + fused_sdpa."""
    if (_impl() == "mfma" and HAVE_EXT and qkv.is_cuda
            and qkv.dtype == torch.bfloat16 and qkv.dim() == 5
            and qkv.shape[2] == 3 and qkv.shape[-1] == 64
            and qkv.is_contiguous()):
        return _FusedSDPAQkv.apply(qkv, bool(is_causal))
    q, k, v = qkv.permute(2, 0, 3, 1, 4)
    return fused_sdpa(q, k, v, is_causal=is_causal)


def fused_sdpa(q, k, v, is_causal=False):
    """Like F.scaled_dot_product_attention for [B,H,N,D] with D==64."""
    impl = _impl()
    if (impl in ("mfma", "ref") and HAVE_EXT and q.is_cuda
            and q.dtype == torch.bfloat16
            and q.dim() == 4 and q.shape[-1] == 64
            and q.shape == k.shape == v.shape):
        mfma = impl == "mfma"
        if mfma and (q.stride(-1) == 1 and q.stride(0) % 8 == 0
                     and q.stride(1) % 8 == 0 and q.stride(2) % 8 == 0
                     and q.stride() == k.stride() == v.stride()):
            # the MFMA kernels read strided q/k/v natively (e.g. head
            # slices of a fused qkv projection) — no copies
            return _FusedSDPA.apply(q, k, v, bool(is_causal), True)
        return _FusedSDPA.apply(q.contiguous(), k.contiguous(),
                                v.contiguous(), bool(is_causal), mfma)
    return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal)
