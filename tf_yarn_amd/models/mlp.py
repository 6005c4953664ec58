"""Small MLP (the KerasExperiment 2-layer-MLP plumbing config,
BASELINE.json config 1)."""

from typing import Tuple

import torch
from torch import nn


class MLP(nn.Module):
    def __init__(self, in_dim: int = 784,
                 hidden: Tuple[int, ...] = (128, 64),
                 n_classes: int = 10):
        super().__init__()
        layers = []
        d = in_dim
        for h in hidden:
            layers += [nn.Linear(d, h), nn.ReLU()]
            d = h
        layers.append(nn.Linear(d, n_classes))
        self.net = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x.flatten(1))
