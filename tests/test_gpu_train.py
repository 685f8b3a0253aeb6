"""End-to-end GPU training smoke (@gpu, single device)."""

import pytest
import torch

from pytorch_ps_mpi_amd import SGD, Adam, models, ops

pytestmark = pytest.mark.gpu


def _train(name, opt_cls, steps=3, batch=4, dtype=torch.bfloat16, **opt_kw):
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = models.build_model(name, device=device, dtype=dtype)
    opt = opt_cls(model.named_parameters(), **opt_kw)
    x, y = models.synthetic_batch(name, batch, device=device, dtype=dtype,
                                  seed=0)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = models.loss_fn(name, model, x, y)
        loss.backward()
        l, m = opt.step(loss=loss)
        losses.append(float(l.detach()))
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses))), losses
    return losses, m


def test_ext_is_the_path():
    assert ops.HAVE_EXT


def test_resnet18_bf16_sgd():
    losses, m = _train("resnet18", SGD, lr=0.05, momentum=0.9)
    assert losses[-1] < losses[0]


def test_resnet50_bf16_sgd():
    losses, _ = _train("resnet50", SGD, steps=2, lr=0.05, momentum=0.9)


def test_resnet18_topk():
    losses, m = _train("resnet18", SGD, lr=0.05, momentum=0.9,
                       code="topk:0.01")
    assert m["wire_codec"] == "topk"


def test_resnet18_quant8():
    losses, m = _train("resnet18", SGD, lr=0.05, momentum=0.9, code="quant8")
    assert m["wire_codec"] == "quant8"


def test_vit_adam_bf16():
    losses, _ = _train("vit_b16", Adam, steps=2, lr=1e-4)


def test_gpt2_adam_bf16():
    device = torch.device("cuda:0")
    from pytorch_ps_mpi_amd.models.gpt2 import GPT2
    torch.manual_seed(0)
    model = GPT2(vocab=1024, ctx=256, dim=256, depth=4, heads=4).to(
        device, torch.bfloat16)
    opt = Adam(model.named_parameters(), lr=1e-4)
    x = torch.randint(0, 1024, (2, 129), device=device)
    for _ in range(2):
        opt.zero_grad()
        loss = model.loss(x[:, :-1], x[:, 1:])
        loss.backward()
        opt.step(loss=loss)
    torch.cuda.synchronize()
    assert torch.isfinite(loss.detach())


def test_bf16_master_consistency():
    """flat_param (bf16) must track master (fp32) after updates."""
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = models.build_model("mlp", device=device, dtype=torch.bfloat16)
    opt = SGD(model.named_parameters(), lr=0.1, momentum=0.9)
    x, y = models.synthetic_batch("mlp", 8, device=device,
                                  dtype=torch.bfloat16, seed=1)
    for _ in range(3):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, x, y)
        loss.backward()
        opt.step(loss=loss)
    torch.cuda.synchronize()
    assert torch.equal(opt.flat.flat_param,
                       opt.flat.master.to(torch.bfloat16))


def test_checkpoint_gpu(tmp_path):
    from pytorch_ps_mpi_amd.utils import checkpoint
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = models.build_model("mlp", device=device, dtype=torch.bfloat16)
    opt = SGD(model.named_parameters(), lr=0.1, momentum=0.9)
    x, y = models.synthetic_batch("mlp", 8, device=device,
                                  dtype=torch.bfloat16, seed=1)
    for _ in range(2):
        opt.zero_grad()
        loss = models.loss_fn("mlp", model, x, y)
        loss.backward()
        opt.step(loss=loss)
    checkpoint.save(str(tmp_path / "ck.pt"), opt)
    torch.manual_seed(7)
    model2 = models.build_model("mlp", device=device, dtype=torch.bfloat16)
    opt2 = SGD(model2.named_parameters(), lr=0.1, momentum=0.9)
    checkpoint.load(str(tmp_path / "ck.pt"), opt2)
    assert torch.equal(opt.flat.flat_param, opt2.flat.flat_param)
    assert torch.equal(opt._mom, opt2._mom)


def test_profile_gpu_metrics():
    """profile_gpu=True adds HIP-event *_gpu_ms spans to the metrics dict."""
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = models.build_model("mlp", device=device, dtype=torch.bfloat16)
    opt = SGD(model.named_parameters(), lr=0.1, momentum=0.9,
              profile_gpu=True)
    x, y = models.synthetic_batch("mlp", 8, device=device,
                                  dtype=torch.bfloat16, seed=0)
    opt.zero_grad()
    loss = models.loss_fn("mlp", model, x, y)
    loss.backward()
    _, m = opt.step(loss=loss)
    gpu_keys = [k for k in m if k.endswith("_gpu_ms")]
    assert gpu_keys, m.keys()
    assert all(v >= 0.0 for k, v in m.items() if k.endswith("_gpu_ms"))
