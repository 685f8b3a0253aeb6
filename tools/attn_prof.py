#!/usr/bin/env python3
"""Run fwd / fwd+bwd attention at the GPT-2 shape for rocprofv3."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ps_mpi_amd.ops import attn as attn_mod

os.environ["PS_AMD_ATTN"] = "mfma"
dev = "cuda:0"
B, H, N, D = 96, 12, 512, 64
torch.manual_seed(0)
q = torch.randn(B, H, N, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
k = torch.randn_like(q, requires_grad=True)
v = torch.randn_like(q, requires_grad=True)
g = torch.randn_like(q)
for _ in range(5):
    o = attn_mod.fused_sdpa(q, k, v, is_causal=True)
    o.backward(g)
    q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
print("done")
