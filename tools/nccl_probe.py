#!/usr/bin/env python3
"""Probe RCCL behaviors the async engine depends on, with TWO RANKS SHARING
ONE GPU (for `pytest -m gpu` world>1 tests on a 1-GPU box):

  1. does RCCL accept two ranks on the same device in one communicator?
  2. do pair process groups + isend/irecv work rank<->rank on one device?
  3. does the content-tag arrival detection (tail message) observe arrival
     from the host without wait()?
  4. does Work.is_completed() ever turn true for RCCL p2p (gloo's never
     does on this torch build)?

Prints one JSON line per finding.  Exit 0 even on failures — this is a
probe, not a test.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def worker(rank, port, out_dir):
    res = {}
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    torch.cuda.set_device(0)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=2)
        res["init"] = "ok"
    except Exception as e:
        res["init"] = repr(e)
        _dump(rank, res, out_dir)
        return
    dev = torch.device("cuda", 0)
    try:
        t = torch.ones(4, device=dev) * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        res["allreduce"] = "ok" if float(t[0]) == 3.0 else f"wrong {t.tolist()}"
    except Exception as e:
        res["allreduce"] = repr(e)
    try:
        g1 = dist.new_group([0, 1])
        g2 = dist.new_group([0, 1])
        if rank == 0:
            hdr = torch.zeros(2, dtype=torch.int64, device=dev)
            wire = torch.zeros(1 << 20, device=dev)
            tail = torch.zeros(1, dtype=torch.int64, device=dev)
            r = [dist.irecv(hdr, src=1, group=g1),
                 dist.irecv(wire, src=1, group=g1),
                 dist.irecv(tail, src=1, group=g1)]
            t0 = time.time()
            seen_tag = seen_completed = None
            while time.time() - t0 < 20:
                if seen_tag is None and int(tail.item()) == 7:
                    seen_tag = time.time() - t0
                if seen_completed is None and all(x.is_completed() for x in r):
                    seen_completed = time.time() - t0
                if seen_tag is not None and seen_completed is not None:
                    break
                time.sleep(0.01)
            for x in r:
                x.wait()
            res["p2p_tag_seen_s"] = seen_tag
            res["p2p_is_completed_seen_s"] = seen_completed
            res["p2p_payload"] = "ok" if float(wire[123]) == 1.0 and \
                hdr.tolist() == [5, 6] else "wrong"
            # reply channel back
            dist.isend(torch.full((1,), 9, dtype=torch.int64, device=dev),
                       dst=1, group=g2).wait()
        else:
            time.sleep(1.0)
            dist.isend(torch.tensor([5, 6], dtype=torch.int64, device=dev),
                       dst=0, group=g1).wait()
            dist.isend(torch.ones(1 << 20, device=dev), dst=0, group=g1).wait()
            dist.isend(torch.full((1,), 7, dtype=torch.int64, device=dev),
                       dst=0, group=g1).wait()
            back = torch.zeros(1, dtype=torch.int64, device=dev)
            dist.irecv(back, src=0, group=g2).wait()
            res["reply_back"] = int(back.item())
        res["pair_groups"] = "ok"
    except Exception as e:
        res["pair_groups"] = repr(e)
    dist.barrier()
    _dump(rank, res, out_dir)
    dist.destroy_process_group()


def _dump(rank, res, out_dir):
    res["rank"] = rank
    line = json.dumps(res)
    print(line, flush=True)
    if out_dir:
        with open(os.path.join(out_dir, f"nccl_probe_r{rank}.json"), "w") as f:
            f.write(line)


if __name__ == "__main__":
    out_dir = sys.argv[1] if len(sys.argv) > 1 else ""
    mp.spawn(worker, args=(29541, out_dir), nprocs=2, join=True)
    print("probe done")
