"""Fused NHWC BatchNorm kernels vs plain-torch fp32 reference (@gpu)."""

import pytest
import torch
import torch.nn.functional as F

from pytorch_ps_mpi_amd.ops.bn import FusedBatchNorm2d

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def _mk(shape, seed=0):
    torch.manual_seed(seed)
    x = torch.randn(shape, device="cuda:0")
    return x


def _ref_bn(x32, w32, b32, relu, z32=None, eps=1e-5):
    """fp32 eager reference with batch stats. Returns (y, mean, var, y_lin)
    where y_lin is the pre-ReLU value (for boundary-aware comparisons)."""
    mean = x32.mean(dim=(0, 2, 3))
    var = x32.var(dim=(0, 2, 3), unbiased=False)
    y = F.batch_norm(x32, None, None, w32, b32, True, 0.0, eps)
    if z32 is not None:
        y = y + z32
    lin = y
    if relu:
        y = F.relu(y)
    return y, mean, var, lin


@pytest.mark.parametrize("C,HW,relu,has_z", [
    (64, 32, False, False),
    (128, 16, True, False),
    (256, 8, True, True),
])
def test_fused_bn_forward(C, HW, relu, has_z):
    N = 8
    x32 = _mk((N, C, HW, HW))
    z32 = _mk((N, C, HW, HW), seed=5) if has_z else None
    bn = FusedBatchNorm2d(C, relu=relu).to("cuda:0", torch.bfloat16)
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)
    x = x32.bfloat16().contiguous(memory_format=CL).requires_grad_(True)
    z = z32.bfloat16().contiguous(memory_format=CL).requires_grad_(True) \
        if has_z else None
    bn.train()
    y = bn(x, z=z) if has_z else bn(x)
    assert y.is_contiguous(memory_format=CL)
    ref, mean, var, _ = _ref_bn(x32, bn.weight.float(), bn.bias.float(),
                                relu, z32)
    err = (y.float() - ref).abs().max().item()
    assert err < 0.08, f"forward err {err}"
    # running stats updated toward batch stats
    rm_err = (bn.running_mean.float() - 0.1 * mean).abs().max().item()
    assert rm_err < 0.02, rm_err


@pytest.mark.parametrize("relu,has_z", [(False, False), (True, False),
                                        (True, True)])
def test_fused_bn_backward(relu, has_z):
    N, C, HW = 8, 64, 16
    # bf16-rounded inputs for BOTH arms so the fp32 reference sees the same
    # values the kernel does (otherwise ReLU masks flip at the boundary)
    x32 = _mk((N, C, HW, HW), seed=1).bfloat16().float().requires_grad_(True)
    z32 = _mk((N, C, HW, HW), seed=2).bfloat16().float().requires_grad_(True) \
        if has_z else None
    w32 = (torch.rand(C, device="cuda:0") + 0.5).bfloat16().float()
    b32 = (torch.randn(C, device="cuda:0") * 0.3).bfloat16().float()
    w32.requires_grad_(True)
    b32.requires_grad_(True)
    ref, _, _, y_lin = _ref_bn(x32, w32, b32, relu, z32)
    gout = _mk((N, C, HW, HW), seed=3).bfloat16().float()
    ref.backward(gout)

    bn = FusedBatchNorm2d(C, relu=relu).to("cuda:0", torch.bfloat16)
    with torch.no_grad():
        bn.weight.copy_(w32.detach())
        bn.bias.copy_(b32.detach())
    bn.train()
    x = x32.detach().bfloat16().contiguous(memory_format=CL).requires_grad_(True)
    z = z32.detach().bfloat16().contiguous(memory_format=CL).requires_grad_(True) \
        if has_z else None
    y = bn(x, z=z) if has_z else bn(x)
    y.backward(gout.bfloat16().contiguous(memory_format=CL))

    # Near the ReLU boundary (|y|≈0) the bf16 kernel and the fp32 reference
    # can disagree on the mask, flipping whole dy elements — compare away
    # from the boundary and bound how many boundary elements exist.
    if relu:
        interior = y_lin.detach().abs() > 0.05
        frac = 1.0 - interior.float().mean().item()
        assert frac < 0.10, f"too many boundary elems {frac}"
    else:
        interior = torch.ones_like(ref, dtype=torch.bool)
    dx_err = ((x.grad.float() - x32.grad).abs() * interior).max().item()
    assert dx_err < 0.05, f"dx err {dx_err}"
    if has_z:
        dz_err = ((z.grad.float() - z32.grad).abs() * interior).max().item()
        assert dz_err < 0.05, f"dz err {dz_err}"
    dg_err = (bn.weight.grad.float() - w32.grad).abs().max().item()
    db_err = (bn.bias.grad.float() - b32.grad).abs().max().item()
    scale = w32.grad.abs().max().item() + 1.0
    assert dg_err / scale < 0.06, f"dgamma err {dg_err}"
    assert db_err / (b32.grad.abs().max().item() + 1.0) < 0.06, db_err


def test_fused_bn_eval_mode():
    N, C, HW = 4, 64, 8
    bn = FusedBatchNorm2d(C).to("cuda:0", torch.bfloat16)
    x32 = _mk((N, C, HW, HW), seed=7)
    with torch.no_grad():
        bn.running_mean.copy_(torch.randn(C) * 0.1)
        bn.running_var.copy_(torch.rand(C) + 0.5)
    bn.eval()
    x = x32.bfloat16().contiguous(memory_format=CL)
    with torch.no_grad():
        y = bn(x)
    ref = F.batch_norm(x32, bn.running_mean.float(), bn.running_var.float(),
                       bn.weight.float(), bn.bias.float(), False, 0.0, bn.eps)
    err = (y.float() - ref).abs().max().item()
    assert err < 0.08, err


def test_fused_bn_fallback_matches_fast():
    """CPU fallback and GPU fast path must implement the same function."""
    N, C, HW = 4, 64, 8
    x32 = _mk((N, C, HW, HW), seed=9)
    z32 = _mk((N, C, HW, HW), seed=10)
    bn_gpu = FusedBatchNorm2d(C, relu=True).to("cuda:0", torch.bfloat16)
    bn_cpu = FusedBatchNorm2d(C, relu=True)
    with torch.no_grad():
        bn_cpu.weight.copy_(bn_gpu.weight.float().cpu())
        bn_cpu.bias.copy_(bn_gpu.bias.float().cpu())
    y_gpu = bn_gpu(x32.bfloat16().contiguous(memory_format=CL),
                   z=z32.bfloat16().contiguous(memory_format=CL))
    y_cpu = bn_cpu(x32.cpu(), z=z32.cpu())
    err = (y_gpu.float().cpu() - y_cpu).abs().max().item()
    assert err < 0.08, err
