"""Fused flash-attention fwd+bwd vs torch SDPA (@gpu).

Hardware-validated (causal + non-causal, odd N, all three grads); the
MFMA-tiled perf rewrite is ROADMAP item 2 — models keep torch SDPA until
this kernel beats it.
"""

import pytest
import torch
import torch.nn.functional as F

from pytorch_ps_mpi_amd.ops.attn import fused_sdpa

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("B,H,N,causal", [
    (2, 2, 64, False), (2, 2, 67, False), (1, 3, 128, True), (2, 1, 33, True),
])
def test_fused_sdpa_fwd_bwd(B, H, N, causal):
    torch.manual_seed(0)
    dev = "cuda:0"
    D = 64
    q32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    k32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    v32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    for t in (q32, k32, v32):
        t.requires_grad_(True)
    ref = F.scaled_dot_product_attention(q32, k32, v32, is_causal=causal)
    g = torch.randn_like(ref).bfloat16().float()
    ref.backward(g)

    q = q32.detach().bfloat16().requires_grad_(True)
    k = k32.detach().bfloat16().requires_grad_(True)
    v = v32.detach().bfloat16().requires_grad_(True)
    o = fused_sdpa(q, k, v, is_causal=causal)
    o.backward(g.bfloat16())

    assert (o.float() - ref).abs().max().item() < 0.05
    for got, want, name in [(q.grad, q32.grad, "dq"), (k.grad, k32.grad, "dk"),
                            (v.grad, v32.grad, "dv")]:
        err = (got.float() - want).abs().max().item()
        scale = want.abs().max().item() + 1.0
        assert err < 0.05 * scale, f"{name} err {err}"


def test_fused_sdpa_fallback_d128():
    dev = "cuda:0"
    q = torch.randn(1, 2, 16, 128, device=dev, dtype=torch.bfloat16)
    o = fused_sdpa(q, q, q)
    assert torch.isfinite(o.float()).all()
