"""Multi-process CPU tests (gloo, world_size=2) — BASELINE config 1.

Asserts the SURVEY §4 gap-closing properties:
  (a) codec round-trip / cross-rank grad-sum equals single-process
      full-batch gradient (grad_scale='mean', equal shards),
  (b) replicated params stay bitwise identical across ranks after K steps,
  (c) sync-PS broadcast keeps ranks identical,
  (d) async AsySG-InCon trains with bounded staleness and clean shutdown.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.timeout(300)

WORLD = 2


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _setup(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["LOCAL_RANK"] = str(rank)
    from pytorch_ps_mpi_amd import init_distributed
    return init_distributed(backend="gloo")


def _mlp_and_data(rank, batch=16):
    from pytorch_ps_mpi_amd import models
    torch.manual_seed(0)
    model = models.build_model("mlp")
    # full deterministic batch; each rank takes its shard
    x, y = models.synthetic_batch("mlp", batch * WORLD, seed=42)
    xs = x[rank * batch:(rank + 1) * batch]
    ys = y[rank * batch:(rank + 1) * batch]
    return model, (x, y), (xs, ys)


def _single_process_reference(steps, lr, momentum, batch=16):
    """torch.optim.SGD on the FULL batch — the grad-sum oracle."""
    from pytorch_ps_mpi_amd import models
    torch.manual_seed(0)
    model = models.build_model("mlp")
    x, y = models.synthetic_batch("mlp", batch * WORLD, seed=42)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=momentum)
    for _ in range(steps):
        opt.zero_grad()
        models.loss_fn("mlp", model, x, y).backward()
        opt.step()
    return torch.cat([p.detach().flatten() for p in model.parameters()])


def _checksums_equal(opt):
    import torch.distributed as dist
    cs = [None] * WORLD
    dist.all_gather_object(cs, opt.flat.param_checksum())
    return len(set(cs)) == 1


def _replicated_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9,
              mode="replicated", code=codec, grad_scale="mean",
              bucket_mb=0.05, debug_consistency=2)
    for step in range(5):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        opt.step(loss=loss)
        assert _checksums_equal(opt), f"rank divergence at step {step}"
    if rank == 0:
        if codec is None:
            ref = _single_process_reference(5, lr=0.05, momentum=0.9)
            got = torch.cat([p.detach().flatten() for p in model.parameters()])
            err = (got - ref).abs().max().item()
            assert err < 1e-5, f"grad-sum mismatch vs single-process: {err}"
        with open(out_file, "w") as f:
            f.write("ok")
    opt.finish()


def _sync_ps_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9, mode="ps",
              code=codec, grad_scale="mean", bucket_mb=0.05)
    losses = []
    for step in range(6):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, m = opt.step(loss=loss)
        losses.append(float(l.detach()))
        assert _checksums_equal(opt), f"rank divergence at step {step}"
    assert losses[-1] < losses[0]
    if rank == 0:
        with open(out_file, "w") as f:
            f.write("ok")
    opt.finish()


def _async_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              code=codec, bucket_mb=0.05, window=2, max_stale=4)
    losses = []
    for step in range(10):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, m = opt.step(loss=loss)
        losses.append(float(l.detach()))
        if rank != 0:
            assert m.get("staleness", 0) <= 4 + 2
    opt.finish()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0] * 1.5  # training, not diverging
    if rank == 0:
        hist = opt.engine.staleness_hist
        assert sum(hist.values()) >= 5  # PS actually served pushes
        with open(out_file, "w") as f:
            f.write("ok")


def _async_dedicated_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              code=codec, bucket_mb=0.05, window=2, max_stale=4,
              dedicated_ps=True)
    if rank == 0:
        opt.serve()
        opt.finish()
        hist = opt.engine.staleness_hist
        assert sum(hist.values()) >= 8
        with open(out_file, "w") as f:
            f.write("ok")
    else:
        for step in range(8):
            opt.zero_grad()
            loss = models.loss_fn("mlp", model, xs, ys)
            loss.backward()
            opt.step(loss=loss)
        opt.finish()


def _spawn(fn, codec, tmp_path):
    out = str(tmp_path / "ok.txt")
    last = None
    for attempt in range(3):  # _free_port() can race with other suites
        port = _free_port()
        try:
            mp.spawn(fn, args=(port, codec, out), nprocs=WORLD, join=True)
            break
        except Exception as e:  # noqa: BLE001 - retry on rendezvous races
            last = e
            if "EADDRINUSE" not in str(e) and "Address already in use" \
                    not in str(e) and attempt == 2:
                raise
    else:
        raise last
    assert os.path.exists(out)


def test_replicated_identity(tmp_path):
    _spawn(_replicated_worker, None, tmp_path)


def test_replicated_quant8(tmp_path):
    _spawn(_replicated_worker, "quant8", tmp_path)


def test_replicated_topk(tmp_path):
    _spawn(_replicated_worker, "topk:0.25", tmp_path)


def test_sync_ps_identity(tmp_path):
    _spawn(_sync_ps_worker, None, tmp_path)


def test_sync_ps_quant8(tmp_path):
    _spawn(_sync_ps_worker, "quant8", tmp_path)


def test_async_colocated(tmp_path):
    _spawn(_async_worker, None, tmp_path)


def test_async_topk(tmp_path):
    _spawn(_async_worker, "topk:0.25", tmp_path)


def test_async_dedicated_ps(tmp_path):
    _spawn(_async_dedicated_worker, None, tmp_path)


def _async_quorum_worker(rank, port, codec, out_file):
    """quorum=2: the PS batches two pushes per optimizer update."""
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              code=codec, bucket_mb=0.05, window=2, max_stale=6, quorum=2)
    for step in range(8):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        opt.step(loss=loss)
    opt.finish()
    if rank == 0:
        served = sum(opt.engine.staleness_hist.values())
        assert served >= 6
        # with quorum=2 the version counter advances at most every 2 pushes
        assert opt.engine.ps_version <= (served + 8) // 2 + 1
        with open(out_file, "w") as f:
            f.write("ok")


def test_async_quorum(tmp_path):
    _spawn(_async_quorum_worker, None, tmp_path)


def _async_adam_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import Adam, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = Adam(model.named_parameters(), lr=1e-3, mode="async",
               bucket_mb=0.05, window=2, max_stale=4)
    losses = []
    for step in range(8):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, _ = opt.step(loss=loss)
        losses.append(float(l.detach()))
    opt.finish()
    assert all(torch.isfinite(torch.tensor(losses)))
    if rank == 0:
        with open(out_file, "w") as f:
            f.write("ok")


def test_async_adam(tmp_path):
    _spawn(_async_adam_worker, None, tmp_path)


def _accum_worker(rank, port, codec, out_file):
    """Gradient accumulation: overlap=False works; overlap=True raises."""
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9,
              mode="replicated", grad_scale="mean", bucket_mb=0.05,
              overlap=False)
    for step in range(3):
        opt.zero_grad()
        for _ in range(2):  # 2-step accumulation
            loss = models.loss_fn("mlp", model, xs, ys)
            loss.backward()
        opt.step(loss=loss)
        assert _checksums_equal(opt)
    # with overlap=True a second backward must raise
    model2, _f2, (xs2, ys2) = _mlp_and_data(rank)
    opt2 = SGD(model2.named_parameters(), lr=0.05, mode="replicated",
               bucket_mb=0.05, overlap=True)
    opt2.zero_grad()
    models.loss_fn("mlp", model2, xs2, ys2).backward()
    raised = False
    try:
        models.loss_fn("mlp", model2, xs2, ys2).backward()
    except RuntimeError as e:
        raised = "accumulation" in str(e)
    assert raised, "expected accumulation guard to fire"
    # drain the collectives the first backward launched, all ranks together
    opt2.step()
    if rank == 0:
        with open(out_file, "w") as f:
            f.write("ok")
    opt.finish()


def test_grad_accumulation(tmp_path):
    _spawn(_accum_worker, None, tmp_path)


def _async_skew_worker(rank, port, codec, out_file):
    """Speed-skewed ranks stress the recv ring / window logic:
    pass codec == 'slow_ps' (PS sleeps) or 'slow_worker'."""
    import time
    from pytorch_ps_mpi_amd import SGD, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              bucket_mb=0.05, window=2, max_stale=3)
    slow_me = (codec == "slow_ps" and rank == 0) or \
              (codec == "slow_worker" and rank != 0)
    for step in range(8):
        if slow_me:
            time.sleep(0.05)
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, m = opt.step(loss=loss)
        assert torch.isfinite(l.detach())
        if rank != 0:
            assert m["staleness"] <= 3 + 2, m["staleness"]
    opt.finish()
    if rank == 0:
        assert sum(opt.engine.staleness_hist.values()) >= 6
        with open(out_file, "w") as f:
            f.write("ok")


def test_async_slow_ps(tmp_path):
    _spawn(_async_skew_worker, "slow_ps", tmp_path)


def test_async_slow_worker(tmp_path):
    _spawn(_async_skew_worker, "slow_worker", tmp_path)


def _ckpt_worker(rank, port, codec, out_file):
    import tempfile
    import torch.distributed as dist
    from pytorch_ps_mpi_amd import SGD, models
    from pytorch_ps_mpi_amd.utils import checkpoint
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9,
              mode="replicated", grad_scale="mean", bucket_mb=0.05)
    for _ in range(3):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        opt.step(loss=loss)
    path = os.path.join(tempfile.gettempdir(),
                        f"ps_ckpt_test_{port}.pt")
    checkpoint.save(path, opt, extra={"step": 3})  # rank 0 only writes
    dist.barrier()
    # fresh model/opt on every rank; rank-0 file restores + re-broadcasts
    torch.manual_seed(99 + rank)  # deliberately divergent init
    model2 = models.build_model("mlp")
    opt2 = SGD(model2.named_parameters(), lr=0.05, momentum=0.9,
               mode="replicated", grad_scale="mean", bucket_mb=0.05)
    extra = checkpoint.load(path, opt2)
    assert extra["step"] == 3
    assert _checksums_equal(opt2), "ranks inconsistent after restore"
    assert opt2.flat.param_checksum() == opt.flat.param_checksum()
    # training continues identically on both optimizers
    for o, m in ((opt, model), (opt2, model2)):
        o.zero_grad()
        models.loss_fn("mlp", m, xs, ys).backward()
        o.step()
    assert opt2.flat.param_checksum() == opt.flat.param_checksum()
    if rank == 0:
        os.unlink(path)
        with open(out_file, "w") as f:
            f.write("ok")


def test_checkpoint_distributed(tmp_path):
    _spawn(_ckpt_worker, None, tmp_path)


class _PartialUseModel(torch.nn.Module):
    """A parameter whose hook never fires (unused in forward)."""

    def __init__(self):
        super().__init__()
        self.used = torch.nn.Linear(8, 4)
        self.unused = torch.nn.Linear(8, 4)

    def forward(self, x):
        return self.used(x)


def _unused_param_worker(rank, port, codec, out_file):
    import torch.nn.functional as F
    from pytorch_ps_mpi_amd import SGD
    _setup(rank, WORLD, port)
    torch.manual_seed(0)
    model = _PartialUseModel()
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9,
              mode="replicated", grad_scale="mean", bucket_mb=0.05)
    x = torch.randn(8, 8)
    y = torch.randint(0, 4, (8,))
    for _ in range(3):
        opt.zero_grad()
        F.cross_entropy(model(x), y).backward()
        opt.step()  # unused param's bucket launches at step() time
        assert _checksums_equal(opt)
    if rank == 0:
        with open(out_file, "w") as f:
            f.write("ok")


def test_unused_param_bucket(tmp_path):
    _spawn(_unused_param_worker, None, tmp_path)


class _RefStyleTopK:
    """A reference-style `codings` plugin: encode -> picklable dict,
    decode -> ndarray (SURVEY §2.2 contract)."""

    def __init__(self, k=64):
        self.k = k
        self.codes = None

    def encode(self, grad, **kw):
        import numpy as np
        flat = np.asarray(grad).reshape(-1)
        idx = np.argsort(np.abs(flat))[-self.k:]
        return {"n": flat.size, "idx": idx.astype("int32"),
                "val": flat[idx].astype("float32")}

    def decode(self, obj, **kw):
        import numpy as np
        out = np.zeros(obj["n"], dtype="float32")
        out[obj["idx"]] = obj["val"]
        return out


def _host_codec_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, HostCodec, models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    user_code = _RefStyleTopK(k=512)
    opt = SGD(model.named_parameters(), lr=0.05, momentum=0.9,
              mode=codec,  # "replicated" or "ps"
              code=HostCodec(user_code), grad_scale="mean", bucket_mb=0.05)
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, m = opt.step(loss=loss)
        losses.append(float(l.detach()))
        assert _checksums_equal(opt)
    assert losses[-1] < losses[0]
    # the reference stashes raw codes on the object before decoding
    if codec == "replicated" or rank == 0:
        assert user_code.codes is not None and len(user_code.codes) == WORLD
    if rank == 0:
        with open(out_file, "w") as f:
            f.write("ok")
    opt.finish()


def test_host_codec_replicated(tmp_path):
    _spawn(_host_codec_worker, "replicated", tmp_path)


def test_host_codec_sync_ps(tmp_path):
    _spawn(_host_codec_worker, "ps", tmp_path)


def _host_codec_async_worker(rank, port, codec, out_file):
    from pytorch_ps_mpi_amd import SGD, HostCodec
    from pytorch_ps_mpi_amd import models
    _setup(rank, WORLD, port)
    model, _full, (xs, ys) = _mlp_and_data(rank)
    opt = SGD(model.named_parameters(), lr=0.02, momentum=0.9, mode="async",
              code=HostCodec(_RefStyleTopK(k=512)), bucket_mb=0.05,
              window=2, max_stale=4)
    for _ in range(6):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, xs, ys)
        loss.backward()
        l, _ = opt.step(loss=loss)
        assert torch.isfinite(l.detach())
    opt.finish()
    if rank == 0:
        assert sum(opt.engine.staleness_hist.values()) >= 4
        with open(out_file, "w") as f:
            f.write("ok")


def test_host_codec_async(tmp_path):
    _spawn(_host_codec_async_worker, None, tmp_path)
