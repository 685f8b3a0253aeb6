#!/usr/bin/env python3
"""Same-box interleaved A/B: in-model step time for attention impl x
contiguity policy.  PS_AMD_ATTN_FORCE_CONTIG=1 makes fused_sdpa copy."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def bench(model_name, kw, steps=8, warmup=3):
    from pytorch_ps_mpi_amd import SGD, Adam, models
    torch.manual_seed(1234)
    dev = torch.device("cuda:0")
    model = models.build_model(model_name, device=dev, dtype=torch.bfloat16)
    opt = Adam(model.named_parameters(), lr=1e-4, mode="async", grad_scale="mean")
    x, y = models.synthetic_batch(model_name, kw["batch"], device=dev,
                                  dtype=torch.bfloat16, seed=1, seq_len=512)
    def step():
        opt.zero_grad()
        models.loss_fn(model_name, model, x, y).backward()
        opt.step()
    for _ in range(warmup): step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps): step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3

res = {}
for rnd in range(2):
    for impl in ("torch", "mfma"):
        os.environ["PS_AMD_ATTN"] = impl
        for mn, kw in [("gpt2_small", {"batch": 96}), ("vit_b16", {"batch": 512})]:
            ms = bench(mn, kw)
            key = (mn, impl)
            res.setdefault(key, []).append(ms)
            print(f"round{rnd} {mn:10s} {impl}: {ms:7.2f} ms", flush=True)
print("== medians ==")
for k, v in sorted(res.items()):
    print(k, round(sorted(v)[len(v)//2], 2))
