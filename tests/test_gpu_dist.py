"""World-2 RCCL async-PS tests (`pytest -m gpu`, needs >= 2 GPUs).

Measured limitation (tools/nccl_probe.py, RCCL 2.26.6 / torch 2.10): RCCL
REFUSES two ranks on one device — `Duplicate GPU detected` — for collectives
AND pair-group p2p alike, so world>1 RCCL cannot be exercised on a 1-GPU
box at all; these tests self-skip there.  On a >= 2-GPU box they run the
flagship AsySG-InCon path end to end over real RCCL: pair process groups,
p2p initial param sync, deferred recv-ring posting, content-tag arrival
detection, bucket-pipelined pushes from backward hooks, sharded replies,
staleness accounting, clean stop-marker shutdown — for identity, quant8 and
top-k codecs, colocated and dedicated PS.  The same protocol logic runs
world 2/4 under gloo in tests/test_dist_cpu*.py on every CPU run.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]


def _need_two_gpus():
    if torch.cuda.device_count() < 2:
        pytest.skip("RCCL rejects two ranks on one device (Duplicate GPU "
                    "detected — see tools/nccl_probe.py finding); these "
                    "world-2 NCCL tests need >= 2 GPUs")


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

    dev_idx = rank % torch.cuda.device_count()
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE="2",
                      LOCAL_RANK=str(dev_idx))
    torch.cuda.set_device(dev_idx)
    dist.init_process_group("nccl", rank=rank, world_size=2)
    assert ops.HAVE_EXT, "HIP extension must be loaded on GPU"
    dev = torch.device("cuda", dev_idx)
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
    _need_two_gpus()
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
