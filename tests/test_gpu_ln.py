"""Fused LayerNorm kernels vs fp32 torch reference (@gpu)."""

import pytest
import torch
import torch.nn.functional as F

from pytorch_ps_mpi_amd.ops.ln import FusedLayerNorm

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("rows,D", [(1024, 768), (333, 1024), (64, 256)])
def test_fused_ln_forward_backward(rows, D):
    torch.manual_seed(0)
    dev = "cuda:0"
    x32 = torch.randn(rows, D, device=dev).bfloat16().float()
    w32 = (torch.rand(D, device=dev) + 0.5).bfloat16().float()
    b32 = (torch.randn(D, device=dev) * 0.2).bfloat16().float()
    x32.requires_grad_(True)
    w32.requires_grad_(True)
    b32.requires_grad_(True)
    ref = F.layer_norm(x32, (D,), w32, b32, 1e-5)
    g = torch.randn(rows, D, device=dev).bfloat16().float()
    ref.backward(g)

    ln = FusedLayerNorm(D).to(dev, torch.bfloat16)
    with torch.no_grad():
        ln.weight.copy_(w32.detach())
        ln.bias.copy_(b32.detach())
    x = x32.detach().bfloat16().requires_grad_(True)
    assert ln._fast_ok(x)
    y = ln(x)
    y.backward(g.bfloat16())

    assert (y.float() - ref).abs().max().item() < 0.05
    assert (x.grad.float() - x32.grad).abs().max().item() < 0.05
    scale = w32.grad.abs().max().item() + 1.0
    assert (ln.weight.grad.float() - w32.grad).abs().max().item() / scale \
        < 0.02
    assert (ln.bias.grad.float() - b32.grad).abs().max().item() \
        / (b32.grad.abs().max().item() + 1.0) < 0.02


def test_fused_ln_3d_shape():
    dev = "cuda:0"
    torch.manual_seed(1)
    ln = FusedLayerNorm(768).to(dev, torch.bfloat16)
    x = torch.randn(4, 197, 768, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    y = ln(x)
    y.sum().backward()
    ref = F.layer_norm(x.detach().float(), (768,), ln.weight.float(),
                       ln.bias.float(), ln.eps)
    assert (y.float() - ref).abs().max().item() < 0.05
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()


def test_fused_ln_fallback_non_multiple():
    """D not multiple of 256 -> torch fallback, still correct."""
    dev = "cuda:0"
    ln = FusedLayerNorm(100).to(dev, torch.bfloat16)
    x = torch.randn(8, 100, device=dev, dtype=torch.bfloat16)
    assert not ln._fast_ok(x)
    y = ln(x)
    assert torch.isfinite(y.float()).all()
