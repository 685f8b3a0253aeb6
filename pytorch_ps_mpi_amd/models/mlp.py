"""2-layer MLP on synthetic MNIST — BASELINE config 1 (CPU plumbing)."""

from __future__ import annotations

import torch.nn as nn


class MLP(nn.Module):
    def __init__(self, din=784, hidden=100, num_classes=10):
        super().__init__()
        self.fc1 = nn.Linear(din, hidden)
        self.act = nn.ReLU()
        self.fc2 = nn.Linear(hidden, num_classes)

    def forward(self, x):
        return self.fc2(self.act(self.fc1(x.flatten(1))))


def mlp(num_classes=10):
    return MLP(num_classes=num_classes)
