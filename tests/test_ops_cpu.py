"""CPU reference-op semantics: the fused op layer must match torch.optim math
(the numerics oracle the GPU kernels are tested against in test_gpu_ops)."""

import torch

from pytorch_ps_mpi_amd import ops


def test_fused_sgd_matches_torch():
    torch.manual_seed(0)
    n = 1000
    p = torch.randn(n)
    g = torch.randn(n)
    buf = torch.zeros(n)
    p2 = p.clone().requires_grad_(True)
    opt = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=0.01,
                          dampening=0.2)
    for step in range(4):
        gg = g * (step + 1)
        ops.fused_sgd(p, buf, gg, None, lr=0.1, momentum=0.9, dampening=0.2,
                      wd=0.01, nesterov=False, mom_init=(step == 0))
        p2.grad = gg.clone()
        opt.step()
    assert torch.allclose(p, p2.detach(), atol=1e-6)


def test_fused_sgd_nesterov():
    torch.manual_seed(1)
    n = 257
    p = torch.randn(n)
    g = torch.randn(n)
    buf = torch.zeros(n)
    p2 = p.clone().requires_grad_(True)
    opt = torch.optim.SGD([p2], lr=0.05, momentum=0.8, nesterov=True)
    for step in range(3):
        ops.fused_sgd(p, buf, g, None, lr=0.05, momentum=0.8, nesterov=True,
                      mom_init=(step == 0))
        p2.grad = g.clone()
        opt.step()
    assert torch.allclose(p, p2.detach(), atol=1e-6)


def test_fused_adam_matches_torch():
    torch.manual_seed(2)
    n = 513
    p = torch.randn(n)
    g = torch.randn(n)
    m1 = torch.zeros(n)
    m2 = torch.zeros(n)
    vmax = torch.zeros(n)
    p2 = p.clone().requires_grad_(True)
    opt = torch.optim.Adam([p2], lr=1e-2, betas=(0.9, 0.99), eps=1e-8,
                           amsgrad=True)
    for step in range(1, 5):
        gg = g * step
        ops.fused_adam(p, m1, m2, vmax, gg, None, lr=1e-2, beta1=0.9,
                       beta2=0.99, eps=1e-8, step=step, amsgrad=True)
        p2.grad = gg.clone()
        opt.step()
    assert torch.allclose(p, p2.detach(), atol=1e-6)


def test_reduce_accum():
    torch.manual_seed(3)
    srcs = [torch.randn(100) for _ in range(5)]
    dst = torch.zeros(100)
    ops.reduce_accum(dst, srcs, scale=2.0)
    assert torch.allclose(dst, 2.0 * sum(srcs), atol=1e-6)
    ops.reduce_accum(dst, srcs[:1], scale=1.0, beta=1.0)
    assert torch.allclose(dst, 2.0 * sum(srcs) + srcs[0], atol=1e-6)


def test_quant8_roundtrip_error():
    torch.manual_seed(4)
    n = 1000
    x = torch.randn(n)
    nc = ops.quant8_nscales(n)
    scales = torch.zeros(nc)
    q = torch.zeros(n, dtype=torch.int8)
    ops.quant8_encode(x, scales, q)
    dst = torch.zeros(n)
    ops.quant8_reduce(dst, [scales], [q])
    # error bounded by half a quantization step per chunk
    step = scales.repeat_interleave(256)[:n]
    assert ((dst - x).abs() <= step * 0.5 + 1e-7).all()


def test_topk_cpu():
    torch.manual_seed(5)
    n = 500
    k = 50
    x = torch.randn(n)
    idx = torch.zeros(k, dtype=torch.int32)
    val = torch.zeros(k)
    ops.topk_encode(x, k, None, idx, val)
    assert len(set(idx.tolist())) == k
    sel = set(idx.tolist())
    thresh = x.abs().topk(k).values.min()
    assert all(abs(x[i]) >= thresh - 1e-7 for i in sel)
    dst = torch.zeros(n)
    ops.topk_scatter(dst, idx, val, k)
    for i in sel:
        assert abs(dst[i] - x[i]) < 1e-6
    assert dst.abs().sum() > 0
    assert (dst[[i for i in range(n) if i not in sel]] == 0).all()


def test_reduce_accum_many_sources():
    """No 8-source cliff: 12 sources chunk into kernel passes (world > 8)."""
    torch.manual_seed(0)
    n = 1024
    srcs = [torch.randn(n) for _ in range(12)]
    dst = torch.full((n,), 2.0)
    ops.reduce_accum(dst, srcs, scale=0.5, beta=0.25)
    ref = 0.25 * torch.full((n,), 2.0) + 0.5 * sum(srcs)
    assert torch.allclose(dst, ref, atol=1e-5)


def test_quant8_reduce_many_sources():
    torch.manual_seed(1)
    n = 700
    nc = ops.quant8_nscales(n)
    scales, qs, ref = [], [], torch.zeros(n)
    for _ in range(11):
        src = torch.randn(n)
        s = torch.zeros(nc)
        q = torch.zeros(n, dtype=torch.int8)
        ops.quant8_encode(src, s, q)
        scales.append(s)
        qs.append(q)
        ref += s.repeat_interleave(256)[:n] * q.float()
    dst = torch.zeros(n)
    ops.quant8_reduce(dst, scales, qs, gscale=2.0)
    assert torch.allclose(dst, 2.0 * ref, atol=1e-4)
