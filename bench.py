#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 async-PS training throughput on MI355X.

BASELINE.json metric: "SGD steps/sec + samples/sec, ResNet-50 async-PS at
1/2/4/8 MI355X".  Synthetic ImageNet-shaped data, random-init weights, bf16
compute with fp32 master (the reference publishes no numbers — BASELINE.md —
so this harness IS the baseline series).

Run (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 bench.py ...

Per-GPU work is fixed (weak scaling).  Rank 0 prints ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=1024,
                    help="per-rank batch size")
    ap.add_argument("--mode", default="async",
                    choices=["async", "replicated", "ps"])
    ap.add_argument("--codec", default=None,
                    help="identity|topk:D|quant8 (default identity)")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--optim", default="sgd", choices=["sgd", "adam"])
    ap.add_argument("--lr", type=float, default=0.0125)
    ap.add_argument("--bucket-mb", type=float, default=None,
                    help="bucket size; default 8 MB for async (ResNet-50 -> "
                    "~7 buckets so hook-pipelined pushes overlap backward, "
                    "docs/DESIGN.md scaling model), 50 MB otherwise")
    ap.add_argument("--dedicated-ps", action="store_true")
    ap.add_argument("--window", type=int, default=2)
    ap.add_argument("--max-stale", type=int, default=8)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--no-channels-last", action="store_true",
                    help="disable NHWC layout for conv models")
    ap.add_argument("--benchmark-find", action="store_true",
                    help="enable exhaustive MIOpen kernel search")
    return ap.parse_args()


def main():
    args = parse_args()
    if os.environ.get("BENCH_DEBUG_HANG"):
        import faulthandler
        faulthandler.dump_traceback_later(
            int(os.environ["BENCH_DEBUG_HANG"]), exit=True)
    import torch.distributed as dist

    from pytorch_ps_mpi_amd import SGD, Adam, init_distributed, models

    device = init_distributed()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = device.type == "cuda"
    dtype = torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else \
        torch.float32

    if on_gpu:
        from pytorch_ps_mpi_amd import ops
        assert ops.HAVE_EXT, "HIP extension must be loaded on GPU"

    torch.manual_seed(1234)
    if on_gpu and args.benchmark_find:
        # exhaustive MIOpen find: same steady-state perf as immediate mode on
        # this workload (measured 4063 vs 4066 samples/s) but ~5 min of
        # cold-start search per fresh box — off by default.
        torch.backends.cudnn.benchmark = True
    model = models.build_model(args.model, device=device, dtype=dtype)
    image_model = args.model in ("resnet18", "resnet50", "vit_b16")
    channels_last = (on_gpu and image_model and not args.no_channels_last
                     and args.model != "vit_b16")
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    if args.bucket_mb is None:
        args.bucket_mb = 8.0 if args.mode == "async" else 50.0
    opt_cls = {"sgd": SGD, "adam": Adam}[args.optim]
    opt_kw = dict(mode=args.mode, code=args.codec,
                  bucket_mb=args.bucket_mb, grad_scale="mean",
                  window=args.window, max_stale=args.max_stale,
                  dedicated_ps=args.dedicated_ps)
    if args.optim == "sgd":
        opt = opt_cls(model.named_parameters(), lr=args.lr, momentum=0.9,
                      **opt_kw)
    else:
        opt = opt_cls(model.named_parameters(), lr=args.lr, **opt_kw)

    x, y = models.synthetic_batch(args.model, args.batch, device=device,
                                  dtype=dtype, seed=100 + rank,
                                  seq_len=args.seq_len)
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)

    # MIOpen find warm-up: with N ranks on one box every process would
    # compile the same conv solutions concurrently (8x contention, many
    # minutes).  Rank 0 populates the shared find/binary cache with
    # comm-free forward+backward passes; the rest wait at a barrier.
    # Safe in async mode: the PS posts nothing before its first step().
    if on_gpu and world > 1 and dist.is_initialized():
        if rank == 0:
            for _ in range(2):
                # flat.zero_grad (NOT opt.zero_grad): must not arm the
                # hook-overlap engine — its collectives would have no
                # matching calls on the ranks waiting at the barrier
                opt.flat.zero_grad()
                models.loss_fn(args.model, model, x, y).backward()
            opt.flat.zero_grad()
            torch.cuda.synchronize()
        dist.barrier()

    is_serving_ps = (args.mode == "async" and args.dedicated_ps and
                     world > 1 and rank == 0)
    n_train = world - 1 if (args.mode == "async" and args.dedicated_ps
                            and world > 1) else world

    def one_step():
        opt.zero_grad()
        loss = models.loss_fn(args.model, model, x, y)
        loss.backward()
        opt.step(loss=loss)
        return loss

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    if is_serving_ps:
        # dedicated PS: no barriers (workers would stall waiting for
        # replies) — serve until every worker sent its stop marker.
        opt.serve()
        opt.finish()
        elapsed = torch.tensor([0.0], dtype=torch.float64)
    else:
        for _ in range(args.warmup):
            one_step()
        # async mode: NO barrier here — the colocated PS must keep serving
        # while workers run (a PS parked in a collective deadlocks a worker
        # that hit its staleness/window bound).  Warmup aligns the ranks.
        if args.mode != "async":
            barrier_sync()
        elif on_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            one_step()
        if on_gpu:
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
        if args.mode == "async":
            opt.finish()  # drain replies, send stop (before any collective)

    # max over training ranks (dedicated PS contributes 0)
    if dist.is_initialized():
        if on_gpu:
            elapsed = elapsed.to(device)  # nccl needs device tensors
        dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
        elapsed = elapsed.cpu()
    barrier_sync()

    t = float(elapsed.item())
    steps_per_sec = args.steps / t
    samples_per_sec = n_train * args.batch * args.steps / t

    if rank == 0:
        out = {
            "metric": "samples_per_sec",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": 1000.0 * t / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "steps_per_sec": steps_per_sec,
            "config": {
                "model": args.model,
                "global_batch": n_train * args.batch,
                "parallelism": (
                    f"async-ps({'dedicated' if args.dedicated_ps else 'colocated'},"
                    f"dp{world})" if args.mode == "async"
                    else f"{args.mode}(dp{world})"),
                "optim": args.optim,
                "codec": opt.codec.name,
                "bucket_mb": args.bucket_mb,
            },
        }
        if args.mode == "async" and world > 1:
            # BASELINE config 5: staleness distribution (PS-side histogram
            # of ps_version - version_used_by_worker per served push)
            hist = dict(getattr(opt.engine, "staleness_hist", {}))
            out["config"]["staleness_hist"] = \
                {str(k): v for k, v in sorted(hist.items())}
        print(json.dumps(out))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
