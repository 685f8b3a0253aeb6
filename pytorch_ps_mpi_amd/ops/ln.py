"""Fused row-wise LayerNorm on CDNA4 HIP kernels (ops/csrc/ln_kernels.hip).

Drop-in for nn.LayerNorm over the last dimension.  Fast path: CUDA + bf16 +
1-D normalized_shape with D % 256 == 0, D <= 4096 (one wave per row, row in
registers).  Anything else falls back to F.layer_norm, so the same model
runs on CPU for the gloo tests.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import HAVE_EXT, _EXT


class _FusedLN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        shape = x.shape
        D = shape[-1]
        rows = x.numel() // D
        xc = x.contiguous()
        y = torch.empty_like(xc)
        mean = torch.empty(rows, dtype=torch.float32, device=x.device)
        rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
        _EXT.ln_fwd(xc, y, weight, bias, mean, rstd, rows, D, eps)
        ctx.save_for_backward(xc, weight, mean, rstd)
        ctx.dims = (rows, D)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        rows, D = ctx.dims
        dyc = dy.contiguous()
        dx = torch.empty_like(x)
        _EXT.ln_bwd_dx(x, dyc, dx, weight, mean, rstd, rows, D)
        dg32 = torch.empty(D, dtype=torch.float32, device=x.device)
        db32 = torch.empty(D, dtype=torch.float32, device=x.device)
        _EXT.ln_bwd_dgb(x, dyc, mean, rstd, dg32, db32, rows, D)
        return dx, dg32.to(weight.dtype), db32.to(weight.dtype), None


class FusedLayerNorm(nn.LayerNorm):
    def _fast_ok(self, x):
        return (HAVE_EXT and x.is_cuda and x.dtype == torch.bfloat16
                and len(self.normalized_shape) == 1
                and self.normalized_shape[0] % 256 == 0
                and self.normalized_shape[0] <= 4096
                and self.elementwise_affine
                and self.weight is not None and self.bias is not None
                and self.weight.dtype == torch.bfloat16)

    def forward(self, x):
        if self._fast_ok(x):
            return _FusedLN.apply(x, self.weight, self.bias, self.eps)
        return F.layer_norm(x, self.normalized_shape, self.weight, self.bias,
                            self.eps)
