"""World-size-8 async-PS rehearsal (gloo, CPU): the full 1-PS + 7-worker
topology of the 8xMI355X node — 7 peer rings, bucket-pipelined pushes,
sharded replies, stop-marker shutdown — exercised end to end."""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.timeout(600)

WORLD = 8


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, port, out_file):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(WORLD),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK=str(rank))
    from pytorch_ps_mpi_amd import SGD, init_distributed, models
    init_distributed(backend="gloo")
    torch.manual_seed(0)
    model = models.build_model("mlp")
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              bucket_mb=0.05, window=2, max_stale=5)
    x, y = models.synthetic_batch("mlp", 8, seed=rank + 1)
    steps = 8
    losses = []
    for _ in range(steps):
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
        assert served == 7 * steps, f"PS served {served}, expected {7*steps}"
        assert opt.engine.peers_dropped == 0
        with open(out_file, "w") as f:
            f.write("ok")


def test_async_world8(tmp_path):
    out = str(tmp_path / "ok.txt")
    mp.spawn(_worker, args=(_free_port(), out), nprocs=WORLD, join=True)
    assert os.path.exists(out)
