"""Checkpoint / resume.

The reference had none — optimizer state lived implicitly in
torch.optim state.  Here: rank 0 saves the fp32 master weights + flat
optimizer state + per-parameter layout; load restores onto any device and
re-broadcasts params so every rank resumes consistently.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def save(path, optimizer, extra=None):
    if optimizer.comm.rank != 0:
        return
    payload = {
        "optimizer": optimizer.state_dict(),
        "codec": optimizer.codec.name,
        "mode": optimizer.mode,
        "extra": extra or {},
    }
    torch.save(payload, path)


def load(path, optimizer):
    payload = torch.load(path, map_location="cpu", weights_only=False)
    optimizer.load_state_dict(payload["optimizer"])
    if optimizer.comm.initialized:
        dist.broadcast(optimizer.flat.flat_param, src=0)
        optimizer.flat.sync_master_from_param()
    return payload.get("extra", {})
