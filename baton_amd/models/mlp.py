"""Demo models + synthetic data — the L3 user layer.

Capability parity with the reference demo (/root/reference/demo.py): a
linear regression model on synthetic data with KNOWN true weights, usable as
a convergence oracle in integration tests, plus a small MLP for the
CPU/gloo plumbing config (BASELINE.json config 1).

The reference's ``Model.train`` (demo.py:29-49) is here the generic
LocalTrainer (runtime/local.py); the model keeps a ``train_round`` wrapper
so the worker's reference-shaped contract still works.
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

from baton_amd.runtime.local import LocalTrainer
from baton_amd.utils.config import TrainConfig

# The reference's fixed true parameter vector (demo.py:54):
# p = (11, 5, 3, 2, 5, 6, 2, 7, 8, 1); y = (p * X).sum(1)
TRUE_WEIGHTS = torch.tensor([11.0, 5, 3, 2, 5, 6, 2, 7, 8, 1])


def make_synthetic_regression(
    n_samples: int, seed: int = 0, noise: float = 0.0
) -> Tuple[torch.Tensor, torch.Tensor]:
    """X ~ N(0,1) [n,10]; y = X @ TRUE_WEIGHTS (+ noise) [n,1]."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n_samples, 10, generator=g)
    y = (x * TRUE_WEIGHTS).sum(dim=1, keepdim=True)
    if noise > 0:
        y = y + noise * torch.randn(n_samples, 1, generator=g)
    return x, y


class LinearRegressionModel(nn.Module):
    """Linear(10,1) regression — the reference's demo model shape
    (demo.py:18-24), with an explicit name (defect D8)."""

    name = "linreg10"

    def __init__(self, train_config: TrainConfig | None = None):
        super().__init__()
        self.fc1 = nn.Linear(10, 1)
        self._trainer = LocalTrainer(train_config or TrainConfig())

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc1(x)

    def train_round(self, *data: torch.Tensor, n_epoch: int = 1) -> List[float]:
        return self._trainer(self, data, n_epoch)


class TinyMLP(nn.Module):
    """Two-layer MLP for the CPU plumbing config (BASELINE.json config 1)."""

    name = "tinymlp"

    def __init__(self, d_in: int = 10, d_hidden: int = 32, d_out: int = 1,
                 train_config: TrainConfig | None = None):
        super().__init__()
        self.fc1 = nn.Linear(d_in, d_hidden)
        self.fc2 = nn.Linear(d_hidden, d_out)
        self._trainer = LocalTrainer(train_config or TrainConfig())

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc2(torch.relu(self.fc1(x)))

    def train_round(self, *data: torch.Tensor, n_epoch: int = 1) -> List[float]:
        return self._trainer(self, data, n_epoch)
