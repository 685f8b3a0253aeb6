#!/usr/bin/env python3
"""Probe RCCL with TWO RANKS SHARING ONE GPU: are same-device collectives
and/or same-device pair-group p2p possible at all?  Each sub-probe records
its own ok/error; results dump to gpurun_out BEFORE any further op so a
failure cannot mask earlier findings."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _dump(rank, res, out_dir):
    res["rank"] = rank
    line = json.dumps(res)
    print(line, flush=True)
    if out_dir:
        with open(os.path.join(out_dir, f"nccl_probe_r{rank}.json"), "w") as f:
            f.write(line)


def worker(rank, port, out_dir):
    res = {}
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    torch.cuda.set_device(0)
    dev = torch.device("cuda", 0)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=2)
        res["init"] = "ok"
    except Exception as e:
        res["init"] = repr(e)
        _dump(rank, res, out_dir)
        return
    # 1. pair-group p2p FIRST (the async engine's only dependency)
    try:
        g1 = dist.new_group([0, 1])
        if rank == 0:
            t = torch.zeros(4, device=dev)
            dist.irecv(t, src=1, group=g1).wait()
            torch.cuda.synchronize()
            res["p2p_pair"] = "ok" if float(t[0]) == 5.0 else f"wrong {t.tolist()}"
        else:
            dist.isend(torch.full((4,), 5.0, device=dev), dst=0, group=g1).wait()
            res["p2p_pair"] = "ok-send"
    except Exception as e:
        res["p2p_pair"] = repr(e)
    _dump(rank, res, out_dir)
    # 2. default-group collective
    try:
        t = torch.ones(4, device=dev) * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        res["allreduce"] = "ok" if float(t[0]) == 3.0 else f"wrong {t.tolist()}"
    except Exception as e:
        res["allreduce"] = repr(e)[:300]
    _dump(rank, res, out_dir)
    time.sleep(2)
    os._exit(0)


if __name__ == "__main__":
    out_dir = sys.argv[1] if len(sys.argv) > 1 else ""
    mp.spawn(worker, args=(29541, out_dir), nprocs=2, join=True)
    print("probe done")
