"""Fused flash-attention fwd+bwd vs torch SDPA (@gpu).

Covers BOTH in-tree implementations (PS_AMD_ATTN=mfma|ref) against the
fp32 torch oracle, at the model shapes (GPT-2 N=512 causal, ViT N=197
non-causal) and edge shapes (odd N, tiny N), plus the 16x16x32 MFMA
fragment-layout self-check that pins the A/B/C lane maps the MFMA kernels
assume.
"""

import pytest
import torch
import torch.nn.functional as F

from pytorch_ps_mpi_amd import ops
from pytorch_ps_mpi_amd.ops.attn import fused_sdpa

pytestmark = pytest.mark.gpu


def test_mfma_fragment_layouts():
    """C[16,16] = A[16,32] @ B[32,16] through one mfma_f32_16x16x32_bf16
    with the lane maps the attention kernels assume.  Asymmetric operands
    (catches transposed layouts, guide §5.4 rule 16)."""
    dev = "cuda:0"
    torch.manual_seed(0)
    a = torch.randn(16, 32, device=dev).bfloat16()
    b = torch.randn(32, 16, device=dev).bfloat16()
    c = torch.zeros(16, 16, dtype=torch.float32, device=dev)
    ops._EXT.fa_selfcheck(a.contiguous(), b.contiguous(), c)
    torch.cuda.synchronize()
    ref = a.float() @ b.float()
    err = (c - ref).abs().max().item()
    assert err < 0.1, f"fragment layout mismatch: max err {err}"


@pytest.mark.parametrize("impl", ["mfma", "ref"])
@pytest.mark.parametrize("B,H,N,causal", [
    (2, 2, 64, False), (2, 2, 67, False), (1, 3, 128, True), (2, 1, 33, True),
    (2, 2, 197, False),   # ViT-B/16 sequence (odd, multi-block)
    (1, 2, 512, True),    # GPT-2 sequence
])
def test_fused_sdpa_fwd_bwd(impl, B, H, N, causal, monkeypatch):
    monkeypatch.setenv("PS_AMD_ATTN", impl)
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


def test_mfma_vs_ref_spike_row(monkeypatch):
    """A spiked K row forces large online-softmax rescales mid-sequence
    (guide §5.4 rule 26: bounded random data never exercises that branch)."""
    monkeypatch.setenv("PS_AMD_ATTN", "mfma")
    torch.manual_seed(1)
    dev = "cuda:0"
    B, H, N, D = 1, 2, 160, 64
    q32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    k32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    k32[:, :, 140] *= 8.0  # spike near the end -> max jumps at a late tile
    v32 = torch.randn(B, H, N, D, device=dev).bfloat16().float()
    ref = F.scaled_dot_product_attention(q32, k32, v32)
    o = fused_sdpa(q32.bfloat16(), k32.bfloat16(), v32.bfloat16())
    assert (o.float() - ref).abs().max().item() < 0.06


def test_fused_sdpa_fallback_d128():
    dev = "cuda:0"
    q = torch.randn(1, 2, 16, 128, device=dev, dtype=torch.bfloat16)
    o = fused_sdpa(q, q, q)
    assert torch.isfinite(o.float()).all()


def test_fused_sdpa_strided_qkv(monkeypatch):
    """Head-slices of a fused qkv projection (the model path) hit the MFMA
    kernels WITHOUT .contiguous() copies — strides are consumed natively."""
    monkeypatch.setenv("PS_AMD_ATTN", "mfma")
    torch.manual_seed(0)
    dev = "cuda:0"
    B, H, N, D = 2, 3, 197, 64
    qkv = torch.randn(B, N, 3, H, D, device=dev).bfloat16()
    q, k, v = qkv.permute(2, 0, 3, 1, 4)  # [B,H,N,D] non-contiguous views
    assert not q.is_contiguous()
    ref = F.scaled_dot_product_attention(q.float(), k.float(), v.float())
    o = fused_sdpa(q, k, v)
    assert (o.float() - ref).abs().max().item() < 0.05
    # causal too
    ref = F.scaled_dot_product_attention(q.float(), k.float(), v.float(),
                                         is_causal=True)
    o = fused_sdpa(q, k, v, is_causal=True)
    assert (o.float() - ref).abs().max().item() < 0.05


def test_fused_sdpa_qkv_function(monkeypatch):
    """qkv-level Function: strided slices in, assembled dqkv out — fwd and
    grads vs the fp32 autograd oracle."""
    monkeypatch.setenv("PS_AMD_ATTN", "mfma")
    torch.manual_seed(2)
    dev = "cuda:0"
    B, T, H, D = 2, 160, 3, 64
    from pytorch_ps_mpi_amd.ops.attn import fused_sdpa_qkv
    qkv32 = torch.randn(B, T, 3, H, D, device=dev).bfloat16().float()
    qkv32.requires_grad_(True)
    q, k, v = qkv32.permute(2, 0, 3, 1, 4)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    g = torch.randn_like(ref).bfloat16().float()
    ref.backward(g)

    qkv = qkv32.detach().bfloat16().requires_grad_(True)
    o = fused_sdpa_qkv(qkv, is_causal=True)
    o.backward(g.bfloat16())
    assert (o.float() - ref).abs().max().item() < 0.05
    err = (qkv.grad.float() - qkv32.grad).abs().max().item()
    scale = qkv32.grad.abs().max().item() + 1.0
    assert err < 0.05 * scale, f"dqkv err {err}"
