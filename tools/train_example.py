#!/usr/bin/env python3
"""End-to-end usage example — the L4 user loop the reference implies
(SURVEY §1): drop-in optimizer, codec, checkpointing, metrics summary.

Single process:
    python tools/train_example.py --model mlp --steps 20
One 8-GPU node (async PS):
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
        tools/train_example.py --model resnet50 --mode async --codec topk:0.01
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pytorch_ps_mpi_amd import SGD, init_distributed, models
from pytorch_ps_mpi_amd.utils import checkpoint
from pytorch_ps_mpi_amd.utils.metrics import print_summary


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="mlp")
    ap.add_argument("--mode", default="async")
    ap.add_argument("--codec", default=None)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--lr", type=float, default=0.05)
    ap.add_argument("--ckpt", default=None, help="save path (rank 0)")
    ap.add_argument("--resume", default=None)
    args = ap.parse_args()

    device = init_distributed()
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    torch.manual_seed(0)
    model = models.build_model(args.model, device=device, dtype=dtype)
    opt = SGD(model.named_parameters(), lr=args.lr, momentum=0.9,
              mode=args.mode, code=args.codec, grad_scale="mean")
    start = 0
    if args.resume:
        start = checkpoint.load(args.resume, opt).get("step", 0)

    rank = opt.comm.rank
    x, y = models.synthetic_batch(args.model, args.batch, device=device,
                                  dtype=dtype, seed=100 + rank)
    history = []
    for step in range(start, start + args.steps):
        opt.zero_grad()
        loss = models.loss_fn(args.model, model, x, y)
        loss.backward()
        loss, metrics = opt.step(loss=loss)
        history.append(metrics)
        if rank == 0 and step % 5 == 0:
            print(f"step {step:4d}  loss {float(loss.detach()):.4f}")
    opt.finish()
    if args.ckpt:
        checkpoint.save(args.ckpt, opt, extra={"step": start + args.steps})
    if rank == 0:
        print_summary(history)


if __name__ == "__main__":
    main()
