"""Single-process optimizer semantics: parity with torch.optim, checkpointing,
lr-scheduler compatibility, reference API shape (step -> (loss, metrics))."""

import torch

from pytorch_ps_mpi_amd import SGD, Adam, models
from pytorch_ps_mpi_amd.utils import checkpoint


def _train(model, opt, steps=5, seed=1):
    x, y = models.synthetic_batch("mlp", 32, seed=seed)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, x, y)
        loss.backward()
        l, metrics = opt.step(loss=loss)
        losses.append(float(l.detach()))
    return losses, metrics


def _torch_train(model, opt, steps=5, seed=1):
    x, y = models.synthetic_batch("mlp", 32, seed=seed)
    for _ in range(steps):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, x, y)
        loss.backward()
        opt.step()


def _params(m):
    return torch.cat([p.detach().flatten() for p in m.parameters()])


def test_sgd_parity_with_torch():
    torch.manual_seed(0)
    m1 = models.build_model("mlp")
    torch.manual_seed(0)
    m2 = models.build_model("mlp")
    opt1 = SGD(m1.named_parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9,
                           weight_decay=1e-4)
    _train(m1, opt1)
    _torch_train(m2, opt2)
    assert torch.allclose(_params(m1), _params(m2), atol=1e-6)


def test_adam_parity_with_torch():
    torch.manual_seed(0)
    m1 = models.build_model("mlp")
    torch.manual_seed(0)
    m2 = models.build_model("mlp")
    opt1 = Adam(m1.named_parameters(), lr=1e-3, betas=(0.9, 0.999),
                weight_decay=1e-2, amsgrad=True)
    opt2 = torch.optim.Adam(m2.parameters(), lr=1e-3, betas=(0.9, 0.999),
                            weight_decay=1e-2, amsgrad=True)
    _train(m1, opt1)
    _torch_train(m2, opt2)
    assert torch.allclose(_params(m1), _params(m2), atol=1e-5)


def test_step_returns_loss_and_metrics():
    torch.manual_seed(0)
    m = models.build_model("mlp")
    opt = SGD(m.named_parameters(), lr=0.1)
    losses, metrics = _train(m, opt, steps=3)
    assert losses[-1] < losses[0]
    assert "optim_step_time" in metrics and "step" in metrics


def test_closure_api():
    torch.manual_seed(0)
    m = models.build_model("mlp")
    opt = SGD(m.named_parameters(), lr=0.1)
    x, y = models.synthetic_batch("mlp", 8, seed=2)

    def closure():
        opt.zero_grad()
        loss = models.loss_fn("mlp", m, x, y)
        loss.backward()
        return loss

    loss, metrics = opt.step(closure)
    assert loss is not None


def test_lr_scheduler_compat():
    torch.manual_seed(0)
    m = models.build_model("mlp")
    opt = SGD(m.named_parameters(), lr=0.1, momentum=0.9)
    sched = torch.optim.lr_scheduler.StepLR(opt, step_size=1, gamma=0.5)
    _train(m, opt, steps=1)
    sched.step()
    assert abs(opt.param_groups[0]["lr"] - 0.05) < 1e-9


def test_checkpoint_resume(tmp_path):
    torch.manual_seed(0)
    m1 = models.build_model("mlp")
    opt1 = SGD(m1.named_parameters(), lr=0.1, momentum=0.9)
    _train(m1, opt1, steps=3)
    path = tmp_path / "ck.pt"
    checkpoint.save(str(path), opt1, extra={"epoch": 3})

    torch.manual_seed(123)  # different init; must be overwritten by load
    m2 = models.build_model("mlp")
    opt2 = SGD(m2.named_parameters(), lr=0.1, momentum=0.9)
    extra = checkpoint.load(str(path), opt2)
    assert extra["epoch"] == 3
    assert torch.allclose(_params(m1), _params(m2))
    # continued training must match exactly
    _train(m1, opt1, steps=2, seed=9)
    _train(m2, opt2, steps=2, seed=9)
    assert torch.allclose(_params(m1), _params(m2))


def test_models_forward_backward_small():
    for name, kw, batch in [
        ("resnet18", {"num_classes": 10}, 2),
        ("vit_b16", {"num_classes": 10}, 1),
    ]:
        m = models.build_model(name, **kw)
        x, y = models.synthetic_batch(name, batch, seed=0)
        y = y % 10
        loss = torch.nn.functional.cross_entropy(m(x), y)
        loss.backward()
        assert torch.isfinite(loss)


def test_gpt2_tiny_forward_backward():
    from pytorch_ps_mpi_amd.models.gpt2 import GPT2
    m = GPT2(vocab=128, ctx=32, dim=64, depth=2, heads=2)
    x = torch.randint(0, 128, (2, 17))
    loss = m.loss(x[:, :-1], x[:, 1:])
    loss.backward()
    assert torch.isfinite(loss)


def test_print_summary(capsys):
    from pytorch_ps_mpi_amd.utils.metrics import print_summary
    torch.manual_seed(0)
    m = models.build_model("mlp")
    opt = SGD(m.named_parameters(), lr=0.1)
    ms = []
    x, y = models.synthetic_batch("mlp", 8, seed=0)
    for _ in range(3):
        opt.zero_grad()
        loss = models.loss_fn("mlp", m, x, y)
        loss.backward()
        _, metrics = opt.step(loss=loss)
        ms.append(metrics)
    print_summary(ms)
    outp = capsys.readouterr().out
    assert "optim_step_time" in outp and "mean" in outp


def test_mpi_ps_factory():
    from pytorch_ps_mpi_amd import MPI_PS
    torch.manual_seed(0)
    m = models.build_model("mlp")
    opt = MPI_PS(m.named_parameters(), optim="adam", lr=1e-3)
    assert isinstance(opt, Adam)
    m2 = models.build_model("mlp")
    opt2 = MPI_PS(m2.named_parameters(), optim="sgd", lr=0.1, momentum=0.9)
    assert isinstance(opt2, SGD)
    import pytest as _pytest
    with _pytest.raises(ValueError):
        MPI_PS(m.named_parameters(), optim="rmsprop")


def test_checkpoint_preserves_fp32_master(tmp_path):
    """Resume must not round the fp32 master through bf16 (advisor r1):
    masters whose low mantissa bits differ from their bf16 rounding must
    survive a save/load cycle bitwise."""
    import torch.nn as nn
    from pytorch_ps_mpi_amd import SGD
    from pytorch_ps_mpi_amd.utils import checkpoint
    torch.manual_seed(0)
    model = nn.Linear(32, 32)
    opt = SGD(model.named_parameters(), lr=0.1, momentum=0.9)
    # perturb the master below bf16 resolution
    with torch.no_grad():
        opt.flat.master.add_(torch.randn_like(opt.flat.master) * 1e-6)
    ref = opt.flat.master.clone()
    path = str(tmp_path / "ck.pt")
    checkpoint.save(path, opt)

    model2 = nn.Linear(32, 32)
    opt2 = SGD(model2.named_parameters(), lr=0.1, momentum=0.9)
    checkpoint.load(path, opt2)
    assert torch.equal(opt2.flat.master, ref), "fp32 master not bit-exact"


def test_topkt_multi_message_sum():
    from pytorch_ps_mpi_amd import codecs
    c = codecs.TopKThreshold(alpha=0.3, max_density=0.5)
    n = 2048
    torch.manual_seed(4)
    a = torch.randn(n) * 0.01
    b = torch.randn(n) * 0.01
    a[:5] = 3.0
    b[2:8] = -2.0  # different k_used per message
    wn = c.wire_numel(n, torch.float32)
    wa, wb = (torch.zeros(wn, dtype=torch.uint8) for _ in range(2))
    c.encode(a, wa)
    c.encode(b, wb)
    ka = int(c._views(wa, n, torch.float32)[1][0])
    kb = int(c._views(wb, n, torch.float32)[1][0])
    assert ka != kb, (ka, kb)
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wa, wb], gscale=0.5, src_dtype=torch.float32)
    # indices 2..4 are spikes in BOTH messages: both contributions land
    assert torch.allclose(dst[2:5], 0.5 * (a[2:5] + b[2:5]), atol=1e-6)
    # 0..1 only in a, 5..7 only in b
    assert torch.allclose(dst[:2], 0.5 * a[:2], atol=1e-6)
    assert torch.allclose(dst[5:8], 0.5 * b[5:8], atol=1e-6)
