"""Fused NHWC BatchNorm (+residual add +ReLU) on CDNA4 HIP kernels.

Why this exists: on MI355X, PyTorch's native `batch_norm_*_channels_last`
kernels were 57% of a ResNet-50 bf16 training step (see
profiles/resnet50_steady_r01.md) at ~0.5 TB/s effective bandwidth.  The
kernels in ops/csrc/bn_kernels.hip run the same math at HBM line rate and
fuse the residual add and ReLU (masks are recomputed from the saved
per-channel scale/shift, so backward reads no extra tensors).

Fast path requirements: CUDA + bf16 + 4D channels_last + C % 64 == 0.
Anything else falls back to torch.nn.functional.batch_norm (bitwise the
reference semantics), so the same model runs on CPU for the gloo tests.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import HAVE_EXT, _EXT

_CL = torch.channels_last


class _FusedBN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, z, weight, bias, rmean, rvar, training, momentum,
                eps, relu):
        N, C, H, W = x.shape
        rows = N * H * W
        dev = x.device
        f32 = dict(dtype=torch.float32, device=dev)
        scale = torch.empty(C, **f32)
        shift = torch.empty(C, **f32)
        if training:
            stats = torch.empty(2 * C, **f32)  # contiguous: one fill launch
            psum, psumsq = stats[:C], stats[C:]
            _EXT.bn_fwd_stats(x, psum, psumsq, rows, C)
            mean = torch.empty(C, **f32)
            invstd = torch.empty(C, **f32)
            _EXT.bn_finalize(psum, psumsq, weight, bias, rmean, rvar, mean,
                             invstd, scale, shift, float(rows), eps, momentum)
        else:
            _EXT.bn_eval_coef(weight, bias, rmean, rvar, scale, shift, eps)
            mean = rmean.float()
            invstd = torch.rsqrt(rvar.float() + eps)
        y = torch.empty_like(x)
        _EXT.bn_normalize(x, z, y, scale, shift, rows, C, relu)
        ctx.save_for_backward(x, z if z is not None else x.new_empty(0),
                              weight if weight is not None else x.new_empty(0),
                              mean, invstd, scale, shift)
        ctx.has_z = z is not None
        ctx.relu = relu
        ctx.training = training
        ctx.dims = (rows, C)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, z, weight, mean, invstd, scale, shift = ctx.saved_tensors
        z = z if ctx.has_z else None
        weight = weight if weight.numel() else None
        rows, C = ctx.dims
        if not dy.is_contiguous(memory_format=_CL):
            dy = dy.contiguous(memory_format=_CL)
        dev = x.device
        f32 = dict(dtype=torch.float32, device=dev)
        dstats = torch.empty(2 * C, **f32)  # contiguous: one fill launch
        dsum, dxsum = dstats[:C], dstats[C:]
        _EXT.bn_bwd_stats(x, dy, z, scale, shift, dsum, dxsum, rows, C,
                          ctx.relu)
        wd = weight.dtype if weight is not None else x.dtype
        dgamma = torch.empty(C, dtype=wd, device=dev)
        dbeta = torch.empty(C, dtype=wd, device=dev)
        ca = torch.empty(C, **f32)
        cbx = torch.empty(C, **f32)
        cc = torch.empty(C, **f32)
        _EXT.bn_bwd_coef(dsum, dxsum, mean, invstd, weight, dgamma, dbeta,
                         ca, cbx, cc, float(rows), ctx.training)
        dx = torch.empty_like(x)
        dz = torch.empty_like(x) if ctx.has_z else None
        _EXT.bn_bwd_dx(x, dy, z, dx, dz, ca, cbx, cc, scale, shift, rows, C,
                       ctx.relu)
        return (dx, dz, dgamma if weight is not None else None,
                dbeta if weight is not None else None,
                None, None, None, None, None, None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d with optional fused residual-add + ReLU.

    forward(x, z=None): y = bn(x) [+ z] [-> relu].  Uses the CDNA4 HIP
    kernels when the fast-path conditions hold, torch otherwise.
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, relu=False):
        super().__init__(num_features, eps=eps, momentum=momentum,
                         affine=affine,
                         track_running_stats=track_running_stats)
        self.relu = relu

    def _fast_ok(self, x):
        return (HAVE_EXT and x.is_cuda and x.dtype == torch.bfloat16
                and x.dim() == 4 and x.shape[1] % 64 == 0
                and x.is_contiguous(memory_format=_CL)
                and (self.weight is None
                     or self.weight.dtype == torch.bfloat16)
                and self.track_running_stats)

    def forward(self, x, z=None):
        if self._fast_ok(x):
            if self.training and self.track_running_stats \
                    and self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            mom = self.momentum if self.momentum is not None else 0.1
            return _FusedBN.apply(x, z, self.weight, self.bias,
                                  self.running_mean, self.running_var,
                                  self.training, mom, self.eps, self.relu)
        y = F.batch_norm(
            x, self.running_mean, self.running_var, self.weight, self.bias,
            self.training or not self.track_running_stats,
            self.momentum if self.momentum is not None else 0.0, self.eps)
        if z is not None:
            y = y + z
        if self.relu:
            y = F.relu(y)
        return y
