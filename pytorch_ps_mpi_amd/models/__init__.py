"""Model registry + synthetic data for the BASELINE configs.

There is no dataset/network access in this environment: all benchmarks run on
synthetic inputs of the right shape with random-init weights.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .gpt2 import gpt2_small
from .mlp import mlp
from .resnet import resnet18, resnet50
from .vit import vit_b16

_REGISTRY = {
    "mlp": mlp,
    "resnet18": resnet18,
    "resnet50": resnet50,
    "vit_b16": vit_b16,
    "gpt2_small": gpt2_small,
}

# (input kind, default shape info)
_KIND = {
    "mlp": ("image_flat", (784,), 10),
    "resnet18": ("image", (3, 224, 224), 1000),
    "resnet50": ("image", (3, 224, 224), 1000),
    "vit_b16": ("image", (3, 224, 224), 1000),
    "gpt2_small": ("tokens", (512,), 50257),  # targets stay in the real vocab
}


def model_names():
    return sorted(_REGISTRY)


def build_model(name, device="cpu", dtype=torch.float32, **kw):
    if name not in _REGISTRY:
        raise ValueError(f"unknown model {name!r}; have {model_names()}")
    m = _REGISTRY[name](**kw)
    return m.to(device=device, dtype=dtype)


def synthetic_batch(name, batch, device="cpu", dtype=torch.float32, seed=None,
                    seq_len=None):
    """Random inputs+targets of the model's training shape."""
    gen = None
    if seed is not None:
        gen = torch.Generator(device="cpu").manual_seed(seed)
    kind, shape, ncls = _KIND[name]
    if kind == "image_flat":
        x = torch.randn(batch, *shape, generator=gen).to(device, dtype)
        y = torch.randint(0, ncls, (batch,), generator=gen).to(device)
    elif kind == "image":
        x = torch.randn(batch, *shape, generator=gen).to(device, dtype)
        y = torch.randint(0, ncls, (batch,), generator=gen).to(device)
    else:  # tokens
        T = seq_len or shape[0]
        x = torch.randint(0, ncls, (batch, T + 1), generator=gen).to(device)
        y = x[:, 1:].contiguous()
        x = x[:, :-1].contiguous()
    return x, y


def loss_fn(name, model, x, y):
    if name == "gpt2_small":
        return model.loss(x, y)
    logits = model(x)
    return F.cross_entropy(logits.float(), y)
