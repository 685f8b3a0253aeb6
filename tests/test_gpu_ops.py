"""HIP kernel numerics vs the plain-torch fp32 CPU references (@gpu).

Every kernel in ops/csrc/ps_kernels.hip is compared against the CPU
implementation in ops/__init__.py (itself validated against torch.optim in
test_ops_cpu.py)."""

import pytest
import torch

from pytorch_ps_mpi_amd import ops

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda:0")


def test_extension_loaded():
    assert ops.HAVE_EXT, f"HIP extension missing: {ops._EXT_ERR!r}"


def test_fails_loudly_without_ext(monkeypatch):
    monkeypatch.setattr(ops, "_EXT", None)
    monkeypatch.setattr(ops, "HAVE_EXT", False)
    t = torch.zeros(8, device=_dev())
    with pytest.raises(RuntimeError, match="HIP extension"):
        ops.reduce_accum(t, [t])


@pytest.mark.parametrize("momentum,nesterov,wd", [
    (0.0, False, 0.0), (0.9, False, 0.01), (0.9, True, 0.0)])
def test_fused_sgd_gpu(momentum, nesterov, wd):
    torch.manual_seed(0)
    n = 100_003
    p_c = torch.randn(n)
    g_c = torch.randn(n)
    buf_c = torch.zeros(n)
    p_g = p_c.to(_dev())
    buf_g = buf_c.to(_dev())
    for step in range(3):
        g = g_c * (step + 1)
        ops.fused_sgd(p_c, buf_c, g, None, lr=0.1, momentum=momentum,
                      wd=wd, nesterov=nesterov, mom_init=(step == 0))
        ops.fused_sgd(p_g, buf_g, g.to(_dev()), None, lr=0.1,
                      momentum=momentum, wd=wd, nesterov=nesterov,
                      mom_init=(step == 0))
    assert torch.allclose(p_c, p_g.cpu(), atol=1e-5)
    if momentum:
        assert torch.allclose(buf_c, buf_g.cpu(), atol=1e-5)


def test_fused_sgd_bf16_out():
    torch.manual_seed(1)
    n = 4096
    p = torch.randn(n, device=_dev())
    g = torch.randn(n, device=_dev())
    out = torch.zeros(n, dtype=torch.bfloat16, device=_dev())
    ops.fused_sgd(p, None, g, out, lr=0.1)
    assert torch.allclose(out.float(), p, atol=0.01, rtol=0.01)


def test_fused_sgd_deterministic():
    torch.manual_seed(5)
    n = 1 << 20
    p0 = torch.randn(n, device=_dev())
    g = torch.randn(n, device=_dev())
    buf = torch.zeros(n, device=_dev())
    p1, b1 = p0.clone(), buf.clone()
    p2, b2 = p0.clone(), buf.clone()
    ops.fused_sgd(p1, b1, g, None, lr=0.1, momentum=0.9, mom_init=True)
    ops.fused_sgd(p2, b2, g, None, lr=0.1, momentum=0.9, mom_init=True)
    assert torch.equal(p1, p2) and torch.equal(b1, b2)


@pytest.mark.parametrize("amsgrad", [False, True])
def test_fused_adam_gpu(amsgrad):
    torch.manual_seed(2)
    n = 54_321
    p_c = torch.randn(n)
    g_c = torch.randn(n)
    m1_c, m2_c, vm_c = (torch.zeros(n) for _ in range(3))
    p_g, m1_g, m2_g, vm_g = (t.to(_dev()) for t in (p_c, m1_c, m2_c, vm_c))
    for step in range(1, 4):
        ops.fused_adam(p_c, m1_c, m2_c, vm_c, g_c, None, lr=1e-2, beta1=0.9,
                       beta2=0.99, eps=1e-8, wd=0.01, step=step,
                       amsgrad=amsgrad)
        ops.fused_adam(p_g, m1_g, m2_g, vm_g, g_c.to(_dev()), None, lr=1e-2,
                       beta1=0.9, beta2=0.99, eps=1e-8, wd=0.01, step=step,
                       amsgrad=amsgrad)
    assert torch.allclose(p_c, p_g.cpu(), atol=1e-5)
    assert torch.allclose(m2_c, m2_g.cpu(), atol=1e-6)


@pytest.mark.parametrize("dtype,n", [
    (torch.bfloat16, 1 << 16),      # vectorized path (n % 8 == 0)
    (torch.bfloat16, 65531),        # scalar path
    (torch.float32, 40_000),
])
def test_reduce_accum_gpu(dtype, n):
    torch.manual_seed(3)
    srcs_c = [torch.randn(n).to(dtype) for _ in range(8)]
    dst_c = torch.zeros(n)
    acc = torch.zeros(n)
    for s in srcs_c:
        acc += s.float()
    srcs_g = [s.to(_dev()) for s in srcs_c]
    dst_g = torch.zeros(n, device=_dev())
    ops.reduce_accum(dst_g, srcs_g, scale=0.5)
    assert torch.allclose(dst_g.cpu(), 0.5 * acc, atol=1e-4)
    # beta accumulate
    ops.reduce_accum(dst_g, srcs_g[:2], scale=1.0, beta=1.0)
    ref = 0.5 * acc + srcs_c[0].float() + srcs_c[1].float()
    assert torch.allclose(dst_g.cpu(), ref, atol=1e-4)


def test_casts_gpu():
    torch.manual_seed(4)
    n = 12345
    f = torch.randn(n, device=_dev())
    b = torch.zeros(n, dtype=torch.bfloat16, device=_dev())
    ops.f32_to_bf16(f, b)
    assert torch.equal(b, f.to(torch.bfloat16))
    f2 = torch.zeros(n, device=_dev())
    ops.bf16_to_f32(b, f2)
    assert torch.equal(f2, b.float())


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_quant8_gpu_vs_cpu(dtype):
    torch.manual_seed(6)
    n = 10_000
    x = torch.randn(n).to(dtype)
    nc = ops.quant8_nscales(n)
    s_c = torch.zeros(nc)
    q_c = torch.zeros(n, dtype=torch.int8)
    ops.quant8_encode(x.float() if dtype == torch.bfloat16 else x, s_c, q_c)
    x_g = x.to(_dev())
    s_g = torch.zeros(nc, device=_dev())
    q_g = torch.zeros(n, dtype=torch.int8, device=_dev())
    ops.quant8_encode(x_g, s_g, q_g)
    assert torch.allclose(s_c, s_g.cpu(), atol=1e-6)
    assert (q_c.int() - q_g.cpu().int()).abs().max() <= 1
    # decode+reduce on GPU vs dequant on CPU
    dst = torch.zeros(n, device=_dev())
    ops.quant8_reduce(dst, [s_g, s_g], [q_g, q_g], gscale=0.5)
    ref = torch.zeros(n)
    ops.quant8_reduce(ref, [s_g.cpu(), s_g.cpu()],
                      [q_g.cpu(), q_g.cpu()], gscale=0.5)
    assert torch.allclose(dst.cpu(), ref, atol=1e-5)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_topk_gpu_invariants(dtype):
    torch.manual_seed(7)
    n = 200_000
    k = 2000
    x = torch.randn(n).to(dtype).to(_dev())
    ws = ops.topk_workspace(_dev())
    idx = torch.zeros(k, dtype=torch.int32, device=_dev())
    val = torch.zeros(k, dtype=dtype, device=_dev())
    ops.topk_encode(x, k, ws, idx, val)
    idx_c = idx.cpu().long()
    val_c = val.cpu()
    assert len(set(idx_c.tolist())) == k, "indices must be unique"
    x_c = x.cpu()
    assert torch.equal(val_c, x_c[idx_c]), "values must match source"
    # bin-threshold invariant: every selected key >= every unselected key
    keys = (x_c.float().abs().view(torch.int32) >> 21) & 0x7FF
    sel_mask = torch.zeros(n, dtype=torch.bool)
    sel_mask[idx_c] = True
    assert keys[sel_mask].min() >= keys[~sel_mask].max()


def test_topk_scatter_gpu():
    torch.manual_seed(8)
    n = 50_000
    k = 512
    x = torch.randn(n, device=_dev())
    ws = ops.topk_workspace(_dev())
    idx = torch.zeros(k, dtype=torch.int32, device=_dev())
    val = torch.zeros(k, device=_dev())
    ops.topk_encode(x, k, ws, idx, val)
    dst = torch.zeros(n, device=_dev())
    ops.topk_scatter(dst, idx, val, k, gscale=2.0)
    ref = torch.zeros(n)
    ref[idx.cpu().long()] = 2.0 * val.cpu()
    assert torch.allclose(dst.cpu(), ref)


def test_topk_thresh_gpu_invariants():
    """Variable-k threshold select: k_used is content-dependent, lands in
    the device header, and the selected set obeys the key-bin threshold."""
    torch.manual_seed(11)
    n = 200_000
    kmax = 20_000
    x = (torch.randn(n) * 0.01)
    spikes = torch.randperm(n)[:500]
    x[spikes] = torch.sign(torch.randn(500)) * (1.0 + torch.rand(500))
    x = x.bfloat16().to(_dev())
    ws = ops.topk_workspace(_dev())
    hdr = torch.zeros(1, dtype=torch.int32, device=_dev())
    idx = torch.zeros(kmax, dtype=torch.int32, device=_dev())
    val = torch.zeros(kmax, dtype=torch.bfloat16, device=_dev())
    off = 16  # alpha = 2^-2 = 0.25
    ops.topk_thresh_encode(x, off, kmax, ws, hdr, idx, val)
    k = int(hdr.item())
    assert 500 <= k < kmax, k
    idx_c = idx[:k].cpu().long()
    assert len(set(idx_c.tolist())) == k
    x_c = x.float().cpu()
    assert torch.equal(val[:k].float().cpu(), x_c[idx_c])
    keys = (x_c.abs().view(torch.int32) >> 21) & 0x7FF
    sel = torch.zeros(n, dtype=torch.bool)
    sel[idx_c] = True
    assert keys[sel].min() >= keys[~sel].max()
    # k matches the key-threshold census (cap not hit here)
    thr0 = max(0, int(keys.max()) - off)
    assert k == min(int((keys >= thr0).sum()), kmax)
    # scatter_var reads exactly k entries
    dst = torch.zeros(n, dtype=torch.float32, device=_dev())
    ops.topk_scatter_var(dst, hdr, idx, val, kmax, gscale=2.0)
    dc = dst.cpu()
    nz = dc.nonzero().flatten()
    assert len(nz) == k
    assert torch.allclose(dc[nz], 2.0 * x_c[nz])


def test_topk_thresh_gpu_kmax_cap():
    torch.manual_seed(12)
    n = 50_000
    kmax = 100
    x = torch.randn(n, device=_dev())  # smooth -> census >> kmax
    ws = ops.topk_workspace(_dev())
    hdr = torch.zeros(1, dtype=torch.int32, device=_dev())
    idx = torch.zeros(kmax, dtype=torch.int32, device=_dev())
    val = torch.zeros(kmax, dtype=torch.float32, device=_dev())
    ops.topk_thresh_encode(x, 80, kmax, ws, hdr, idx, val)
    assert int(hdr.item()) == kmax


def test_reduce_accum_many_sources_gpu():
    torch.manual_seed(0)
    n = 8192
    srcs = [torch.randn(n).bfloat16().to(_dev()) for _ in range(12)]
    dst = torch.full((n,), 2.0, device=_dev())
    ops.reduce_accum(dst, srcs, scale=0.5, beta=0.25)
    ref = 0.25 * torch.full((n,), 2.0) + 0.5 * sum(s.float().cpu() for s in srcs)
    assert torch.allclose(dst.cpu(), ref, atol=1e-2)


def test_fused_linear_parity():
    """FusedLinear (custom bias-grad colsum, GEMMs unchanged) matches
    nn.Linear fwd and all three grads."""
    import torch.nn as nn
    from pytorch_ps_mpi_amd.ops.linear import FusedLinear
    torch.manual_seed(5)
    dev = _dev()
    ref = nn.Linear(256, 512).to(dev, torch.bfloat16)
    fl = FusedLinear(256, 512).to(dev, torch.bfloat16)
    with torch.no_grad():
        fl.weight.copy_(ref.weight)
        fl.bias.copy_(ref.bias)
    x1 = torch.randn(4, 37, 256, device=dev, dtype=torch.bfloat16,
                     requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    g = torch.randn(4, 37, 512, device=dev, dtype=torch.bfloat16)
    y1 = ref(x1)
    y2 = fl(x2)
    assert torch.equal(y1, y2)
    y1.backward(g)
    y2.backward(g)
    assert torch.equal(x1.grad, x2.grad)
    assert torch.allclose(fl.weight.grad.float(), ref.weight.grad.float(),
                          atol=1e-2, rtol=1e-2)
    assert torch.allclose(fl.bias.grad.float(), ref.bias.grad.float(),
                          atol=1e-2, rtol=1e-2)
