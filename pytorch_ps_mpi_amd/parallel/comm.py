"""Communication bootstrap and topology helpers.

One process per GPU, `torch.distributed` with backend "nccl" (= RCCL over
xGMI on ROCm) on GPU and "gloo" on CPU — replacing the reference's mpi4py
transport (mpi_comms.py:11-13, README.md:23-27).  The PS gather pattern maps
to grouped send/recv over the node's fully-connected xGMI links (7 p2p links
per GPU), which serves gather-to-rank-0 better than a ring.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def env_rank():
    return int(os.environ.get("RANK", "0"))


def env_world():
    return int(os.environ.get("WORLD_SIZE", "1"))


def init_distributed(backend=None, device=None, timeout_s=1800):
    """Initialise torch.distributed from torchrun/torch.distributed.run env.

    Returns the torch.device this rank should use.  Safe to call when
    WORLD_SIZE==1 (no process group is created).
    """
    world = env_world()
    rank = env_rank()
    use_cuda = torch.cuda.is_available()
    if device is None:
        if use_cuda:
            local = int(os.environ.get("LOCAL_RANK", rank))
            device = torch.device("cuda", local % torch.cuda.device_count())
        else:
            device = torch.device("cpu")
    else:
        device = torch.device(device)
    if device.type == "cuda":
        torch.cuda.set_device(device)
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if device.type == "cuda" else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return device


class Comm:
    """Thin wrapper: rank/world info plus per-peer pair groups for the
    async-PS p2p channels (each (PS, worker) pair gets its own process group
    so its sends/recvs order independently of other peers')."""

    def __init__(self, ps_rank=0, make_pair_groups=False):
        self.initialized = dist.is_available() and dist.is_initialized()
        self.rank = dist.get_rank() if self.initialized else 0
        self.world = dist.get_world_size() if self.initialized else 1
        self.ps_rank = ps_rank
        # two groups per (PS, worker) pair: the grad-push channel and the
        # param-reply channel.  With RCCL each group owns its own internal
        # stream, so a worker's next push never queues behind a pending
        # reply recv (p2p ops within one communicator are stream-ordered).
        self.push_groups = {}
        self.reply_groups = {}
        if make_pair_groups and self.initialized:
            # every rank must create every group, in the same order
            for w in range(self.world):
                if w == ps_rank:
                    continue
                self.push_groups[w] = dist.new_group([ps_rank, w])
                self.reply_groups[w] = dist.new_group([ps_rank, w])

    @property
    def is_ps(self):
        return self.rank == self.ps_rank

    def push_group(self, worker_rank):
        return self.push_groups[worker_rank]

    def reply_group(self, worker_rank):
        return self.reply_groups[worker_rank]

    def barrier(self):
        if self.initialized:
            dist.barrier()
