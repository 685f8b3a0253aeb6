"""World>1 RCCL tests on ONE GPU (`pytest -m gpu`).

Two ranks share cuda:0: RCCL rejects same-device COLLECTIVES ("Duplicate GPU
detected") but serves same-device p2p through the per-peer pair communicators
(tools/nccl_probe.py) — and the async engine's data plane is pure p2p, so the
flagship AsySG-InCon path gets real-RCCL world-2 coverage on a 1-GPU box:
pair process groups, deferred recv-ring posting, content-tag arrival
detection, sharded replies, staleness accounting.  The collective engines
(replicated / sync-PS) need distinct devices and are covered by the gloo
world-2/4 suites plus the driver's multi-GPU runs.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, port, codec, dedicated, out_file):
    import torch.distributed as dist
    import torch.nn as nn
    import torch.nn.functional as F

    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from pytorch_ps_mpi_amd import SGD, ops

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=2)
    assert ops.HAVE_EXT, "HIP extension must be loaded on GPU"
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = nn.Sequential(*[nn.Sequential(nn.Linear(256, 256), nn.ReLU())
                            for _ in range(4)], nn.Linear(256, 10))
    model = model.to(dev, torch.bfloat16)
    g = torch.Generator().manual_seed(rank + 1)
    x = torch.randn(64, 256, generator=g).to(dev, torch.bfloat16)
    y = torch.randint(0, 10, (64,), generator=g).to(dev)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9, mode="async",
              code=codec, bucket_mb=0.2, window=2, max_stale=4,
              dedicated_ps=dedicated)
    eng = opt.engine
    steps = 10
    if dedicated and rank == 0:
        opt.serve()
        served = sum(eng.staleness_hist.values())
        assert served == steps, f"PS served {served}, expected {steps}"
        opt.finish(barrier=False)
    else:
        assert len(opt.flat.buckets) >= 2
        losses = []
        for _ in range(steps):
            opt.zero_grad()
            loss = F.cross_entropy(model(x).float(), y)
            loss.backward()
            if rank != 0:
                # push already in flight from the backward hooks
                assert eng._cur is not None
            l, m = opt.step(loss=loss)
            losses.append(float(l.detach()))
            if rank != 0:
                assert m["staleness"] <= 4 + 2
        opt.finish(barrier=False)
        lt = torch.tensor(losses)
        assert torch.isfinite(lt).all(), losses
        assert losses[-1] < losses[0], losses
        assert torch.isfinite(opt.flat.flat_param.float()).all()
        if rank == 0:
            served = sum(eng.staleness_hist.values())
            assert served == steps, f"PS served {served}/{steps}"
    if rank == 0 or (dedicated and rank == 1):
        with open(out_file, "w") as f:
            f.write("ok")
    torch.cuda.synchronize()
    os._exit(0)  # skip NCCL destroy teardown (same-device collectives barred)


def _spawn(codec, dedicated, tmp_path):
    out = str(tmp_path / "ok.txt")
    mp.spawn(_worker, args=(_free_port(), codec, dedicated, out), nprocs=2,
             join=True)
    assert os.path.exists(out)


def test_gpu_async_world2_identity(tmp_path):
    _spawn(None, False, tmp_path)


def test_gpu_async_world2_quant8(tmp_path):
    _spawn("quant8", False, tmp_path)


def test_gpu_async_world2_topk(tmp_path):
    _spawn("topk:0.25", False, tmp_path)


def test_gpu_async_world2_dedicated(tmp_path):
    _spawn(None, True, tmp_path)
