"""Linear with a hand-written bias-gradient reduction.

torch's Linear backward computes db with a generic `reduce_kernel` over the
token axis — measured 2.4 ms/step on GPT-2-small and 4.3 ms/step on
ViT-B/16 (profiles/{gpt2,vit}_steady_r02.md).  The GEMMs (dx, dW) stay on
hipBLASLt exactly as autograd issues them; only db moves to k_colsum
(16B loads, wave-owned 64-column tiles — the LN-dgb reduction shape).

Fast path: CUDA + bf16 + out_features % 64 == 0; anything else falls back
to nn.Linear's own autograd (bitwise the stock semantics).
"""

from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import HAVE_EXT, _EXT


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return F.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy2 @ w).reshape(x.shape)
        dw = dy2.t() @ x2
        db32 = torch.empty(dy2.shape[-1], dtype=torch.float32,
                           device=dy.device)
        _EXT.colsum(dy2, db32)
        return dx, dw, db32.to(w.dtype)


class FusedLinear(nn.Linear):
    def forward(self, x):
        if (HAVE_EXT and x.is_cuda and x.dtype == torch.bfloat16
                and self.bias is not None
                and self.out_features % 64 == 0
                and os.environ.get("PS_AMD_FUSED_LINEAR", "1") != "0"):
            return _LinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)
