"""MNIST MLP — BASELINE config 1 (CPU plumbing proof for PyTorchJob).

Mirrors the role of the reference's E2E smoke workloads
(/root/reference/testing/katib_studyjob_test.py drives a tiny training job):
small, fast, runs on CPU with gloo world_size=1..N.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F


class MnistMLP(nn.Module):
    def __init__(self, in_dim: int = 784, hidden: int = 256,
                 n_classes: int = 10):
        super().__init__()
        self.fc1 = nn.Linear(in_dim, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, n_classes)

    def forward(self, x, targets=None):
        x = F.relu(self.fc1(x.flatten(1)))
        x = F.relu(self.fc2(x))
        logits = self.fc3(x)
        if targets is None:
            return logits
        return F.cross_entropy(logits, targets)
