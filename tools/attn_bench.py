#!/usr/bin/env python3
"""Attention implementation A/B at the model shapes (GPU box).

Times fwd and fwd+bwd for torch SDPA (aotriton), the one-wave-per-row
reference kernels, and the MFMA-tiled kernels, on GPT-2-small
(B=96,H=12,N=512,causal) and ViT-B/16 (B=512,H=12,N=197) shapes.
Reports ms and achieved TFLOP/s (2*2*B*H*N^2*D flops fwd, x2.5 bwd,
causal halves).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, iters=20, warmup=5):
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
    return s.elapsed_time(e) / iters * 1e-3


def bench_shape(name, B, H, N, causal, iters):
    from pytorch_ps_mpi_amd.ops import attn as attn_mod
    dev = "cuda:0"
    D = 64
    torch.manual_seed(0)
    q = torch.randn(B, H, N, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    g = torch.randn_like(q)
    fwd_flops = 4.0 * B * H * N * N * D * (0.5 if causal else 1.0)
    tot_flops = fwd_flops * 3.5
    print(f"== {name}: B={B} H={H} N={N} D=64 causal={causal}")
    for impl in ("torch", "ref", "mfma"):
        os.environ["PS_AMD_ATTN"] = impl
        qq = q.clone().requires_grad_(True)
        kk = k.clone().requires_grad_(True)
        vv = v.clone().requires_grad_(True)

        def fwd():
            with torch.no_grad():
                return attn_mod.fused_sdpa(q, k, v, is_causal=causal)

        def fwdbwd():
            if qq.grad is not None:
                qq.grad = None
                kk.grad = None
                vv.grad = None
            o = attn_mod.fused_sdpa(qq, kk, vv, is_causal=causal)
            o.backward(g)

        tf = timeit(fwd, iters)
        tb = timeit(fwdbwd, max(5, iters // 2))
        print(f"{impl:6s} fwd {tf*1e3:8.3f} ms ({fwd_flops/tf/1e12:6.1f} TF) "
              f"fwd+bwd {tb*1e3:8.3f} ms ({tot_flops/tb/1e12:6.1f} TF)",
              flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--skip-ref", action="store_true")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    bench_shape("gpt2-small", 96, 12, 512, True, args.iters)
    bench_shape("vit-b16", 512, 12, 197, False, args.iters)


if __name__ == "__main__":
    main()
