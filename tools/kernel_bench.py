#!/usr/bin/env python3
"""Microbenchmark of the framework's HIP kernels (GPU box).

Times each hot kernel with hip events over large flat buffers and reports
effective HBM bandwidth (read+write bytes / time).  Roofline: ~6.3 TB/s
achievable on MI355X (MI355X_MICROARCH.md).

Usage: python tools/kernel_bench.py [--n 100000000] [--iters 50]
"""

import argparse
import json
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pytorch_ps_mpi_amd import ops


def timeit(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e-3  # seconds


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=100_000_000)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available() and ops.HAVE_EXT
    dev = torch.device("cuda:0")
    n = args.n
    torch.manual_seed(0)

    p = torch.randn(n, device=dev)
    g = torch.randn(n, device=dev)
    buf = torch.zeros(n, device=dev)
    bout = torch.zeros(n, dtype=torch.bfloat16, device=dev)
    results = {}

    def rec(name, t, bytes_moved):
        gbs = bytes_moved / t / 1e9
        results[name] = {"ms": t * 1e3, "GBps": gbs}
        print(f"{name:28s} {t*1e3:9.3f} ms   {gbs:8.1f} GB/s")

    # fused SGD: read p,g,buf + write p,buf,bout = 4f32+1f32... r(p,g,buf)=12B/el, w(p,buf)=8B/el, w(bout)=2B/el
    t = timeit(lambda: ops.fused_sgd(p, buf, g, bout, lr=1e-4, momentum=0.9,
                                     wd=1e-4, mom_init=False), args.iters)
    rec("fused_sgd(+bf16 out)", t, n * (12 + 8 + 2))

    m1 = torch.zeros(n, device=dev)
    m2 = torch.zeros(n, device=dev)
    t = timeit(lambda: ops.fused_adam(p, m1, m2, None, g, bout, lr=1e-4,
                                      step=10), args.iters)
    rec("fused_adam(+bf16 out)", t, n * (16 + 12 + 2))

    # multi-source reduce, 8 bf16 sources
    srcs = [torch.randn(n, device=dev).bfloat16() for _ in range(8)]
    dst = torch.zeros(n, device=dev)
    t = timeit(lambda: ops.reduce_accum(dst, srcs), args.iters)
    rec("reduce8_bf16->f32", t, n * (8 * 2 + 4))
    t = timeit(lambda: ops.reduce_accum(dst, srcs[:1]), args.iters)
    rec("reduce1_bf16->f32(cast)", t, n * (2 + 4))

    gb = g.bfloat16()
    nc = ops.quant8_nscales(n)
    scales = torch.zeros(nc, device=dev)
    q = torch.zeros(n, dtype=torch.int8, device=dev)
    t = timeit(lambda: ops.quant8_encode(gb, scales, q), args.iters)
    rec("quant8_encode(bf16)", t, n * (2 + 1))
    t = timeit(lambda: ops.quant8_reduce(dst, [scales] * 8, [q] * 8),
               args.iters)
    rec("quant8_reduce x8", t, n * (8 * 1 + 4))

    k = n // 100
    ws = ops.topk_workspace(dev)
    idx = torch.zeros(k, dtype=torch.int32, device=dev)
    val = torch.zeros(k, dtype=torch.bfloat16, device=dev)
    t = timeit(lambda: ops.topk_encode(gb, k, ws, idx, val), args.iters)
    rec("topk_encode 1% (bf16)", t, n * 2 * 2)  # two passes over src
    t = timeit(lambda: ops.topk_scatter(dst, idx, val, k), args.iters)
    rec("topk_scatter 1%", t, k * (4 + 2 + 8))

    t = timeit(lambda: ops.f32_to_bf16(p, bout), args.iters)
    rec("f32_to_bf16", t, n * 6)

    # fused BN vs torch native BN (channels_last bf16), ResNet-50 hot shape
    import torch.nn as nn
    from pytorch_ps_mpi_amd.ops.bn import FusedBatchNorm2d
    N, C, HW = 256, 256, 56
    cl = torch.channels_last
    xb = torch.randn(N, C, HW, HW, device=dev).bfloat16().contiguous(
        memory_format=cl)
    gy = torch.randn_like(xb).contiguous(memory_format=cl)
    el = N * C * HW * HW
    for name, mod in [
        ("fused_bn", FusedBatchNorm2d(C, relu=True).to(dev, torch.bfloat16)),
        ("torch_bn", nn.Sequential(nn.BatchNorm2d(C), nn.ReLU()).to(
            dev, torch.bfloat16)),
    ]:
        mod.train()

        def fwdbwd():
            x = xb.detach().requires_grad_(True)
            y = mod(x)
            y.backward(gy)

        t = timeit(fwdbwd, max(5, args.iters // 5))
        # fwd: r x, w y; bwd: B1 r x,dy; B3 r x,dy w dx  => 2B*(2+2+3+... )
        rec(f"{name}+relu fwd+bwd", t, el * 2 * 8)

    with open("gpurun_out/kernel_bench.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    os.makedirs("gpurun_out", exist_ok=True)
    main()
