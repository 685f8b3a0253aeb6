#!/usr/bin/env python3
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ps_mpi_amd import Adam, models
dev = torch.device("cuda:0")
torch.manual_seed(0)
model = models.build_model("vit_b16", device=dev, dtype=torch.bfloat16)
opt = Adam(model.named_parameters(), lr=1e-4, mode="async", grad_scale="mean")
x, y = models.synthetic_batch("vit_b16", 512, device=dev, dtype=torch.bfloat16, seed=1)
for _ in range(6):
    opt.zero_grad()
    models.loss_fn("vit_b16", model, x, y).backward()
    opt.step()
torch.cuda.synchronize()
print("done")
