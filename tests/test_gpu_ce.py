"""Fused cross-entropy vs F.cross_entropy fp32 reference (@gpu)."""

import pytest
import torch
import torch.nn.functional as F

from pytorch_ps_mpi_amd.ops.ce import fused_cross_entropy

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("T,V", [(128, 1000), (512, 50304), (3, 256)])
def test_fused_ce(T, V):
    torch.manual_seed(0)
    dev = "cuda:0"
    logits32 = (torch.randn(T, V, device=dev) * 3).bfloat16().float()
    targets = torch.randint(0, V, (T,), device=dev)
    logits32.requires_grad_(True)
    ref = F.cross_entropy(logits32, targets)
    ref.backward()

    lb = logits32.detach().bfloat16().requires_grad_(True)
    loss = fused_cross_entropy(lb, targets)
    loss.backward()

    assert abs(loss.item() - ref.item()) < 2e-2 * max(1.0, abs(ref.item()))
    derr = (lb.grad.float() - logits32.grad).abs().max().item()
    scale = logits32.grad.abs().max().item()
    assert derr < 0.05 * scale + 1e-5, derr


def test_fused_ce_upstream_grad_scaling():
    torch.manual_seed(1)
    dev = "cuda:0"
    logits = torch.randn(64, 512, device=dev,
                         dtype=torch.bfloat16).requires_grad_(True)
    targets = torch.randint(0, 512, (64,), device=dev)
    loss = fused_cross_entropy(logits, targets)
    (3.0 * loss).backward()
    l2 = logits.detach().clone().requires_grad_(True)
    loss2 = fused_cross_entropy(l2, targets)
    loss2.backward()
    assert torch.allclose(logits.grad.float(), 3.0 * l2.grad.float(),
                          atol=3e-2, rtol=3e-2)


def test_fused_ce_fallback():
    dev = "cuda:0"
    logits = torch.randn(16, 1001, device=dev, dtype=torch.bfloat16)  # V%8!=0
    targets = torch.randint(0, 1001, (16,), device=dev)
    loss = fused_cross_entropy(logits, targets)
    assert torch.isfinite(loss.float())
