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
        # Broadcast the fp32 MASTER (not the bf16 model copy) and re-derive
        # flat_param from it: broadcasting bf16 then syncing master FROM it
        # would round the master through bf16 and a resumed run would diverge
        # from an uninterrupted one (advisor round-1 finding).
        dist.broadcast(optimizer.flat.master, src=0)
        optimizer.flat.sync_param_from_master()
    return payload.get("extra", {})
