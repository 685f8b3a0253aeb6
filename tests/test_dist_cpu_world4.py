"""World-size-4 async-PS tests (gloo): one PS serving 3 peers, and a
randomized soak of the recv-ring/window state machine."""

import os
import random
import time

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.timeout(600)

WORLD = 4


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _setup(rank, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["LOCAL_RANK"] = str(rank)
    from pytorch_ps_mpi_amd import init_distributed
    return init_distributed(backend="gloo")


def _async4_worker(rank, port, window, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, port)
    torch.manual_seed(0)
    model = models.build_model("mlp")
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              bucket_mb=0.05, window=window, max_stale=5)
    x, y = models.synthetic_batch("mlp", 8, seed=rank + 1)
    rng = random.Random(1234 + rank)
    losses = []
    for step in range(20):
        if rng.random() < 0.3:
            time.sleep(rng.random() * 0.02)  # random speed skew
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, x, y)
        loss.backward()
        l, m = opt.step(loss=loss)
        losses.append(float(l.detach()))
        if rank != 0:
            assert m["staleness"] <= 5 + 2
    opt.finish()
    assert all(torch.isfinite(torch.tensor(losses)))
    if rank == 0:
        served = sum(opt.engine.staleness_hist.values())
        assert served == 3 * 20, f"PS served {served}, expected 60"
        with open(out_file, "w") as f:
            f.write("ok")


@pytest.mark.parametrize("window", [1, 2, 4])
def test_async_world4_soak(tmp_path, window):
    out = str(tmp_path / "ok.txt")
    port = _free_port()
    mp.spawn(_async4_worker, args=(port, window, out), nprocs=WORLD,
             join=True)
    assert os.path.exists(out)


def _dedicated4_worker(rank, port, window, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, port)
    torch.manual_seed(0)
    model = models.build_model("mlp")
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              bucket_mb=0.05, window=window, max_stale=4, dedicated_ps=True)
    if rank == 0:
        opt.serve()
        opt.finish()
        served = sum(opt.engine.staleness_hist.values())
        assert served == 3 * 12, served
        with open(out_file, "w") as f:
            f.write("ok")
    else:
        x, y = models.synthetic_batch("mlp", 8, seed=rank)
        for _ in range(12):
            opt.zero_grad()
            loss = models.loss_fn("mlp", model, x, y)
            loss.backward()
            opt.step(loss=loss)
        opt.finish()


def test_async_world4_dedicated(tmp_path):
    out = str(tmp_path / "ok.txt")
    port = _free_port()
    mp.spawn(_dedicated4_worker, args=(port, 2, out), nprocs=WORLD, join=True)
    assert os.path.exists(out)
