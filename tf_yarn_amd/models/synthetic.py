"""Synthetic datasets shaped like the benchmark workloads.

No network in this environment: benches run on synthetic
Criteo-shaped tabular data, synthetic MNIST and synthetic ImageNet with
random-init weights.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from tf_yarn_amd.models.wide_deep import CRITEO_DENSE, DEFAULT_TABLE_SIZES


def synthetic_criteo_batch(batch_size: int,
                           table_sizes: Optional[List[int]] = None,
                           device: str = "cpu",
                           seed: Optional[int] = None
                           ) -> Tuple[torch.Tensor, torch.Tensor,
                                      torch.Tensor]:
    """One Criteo-shaped batch: (dense [B,13] fp32, ids [B,26] int64,
    labels [B] fp32)."""
    table_sizes = table_sizes or DEFAULT_TABLE_SIZES
    gen = torch.Generator(device="cpu")
    if seed is not None:
        gen.manual_seed(seed)
    dense = torch.rand(batch_size, CRITEO_DENSE, generator=gen)
    # log-normal-ish integer features, already log-transformed
    dense = (dense * 8).log1p()
    ids = torch.stack(
        [torch.randint(0, n, (batch_size,), generator=gen)
         for n in table_sizes], dim=1)
    labels = (torch.rand(batch_size, generator=gen) < 0.26).float()
    if device != "cpu":
        dense = dense.to(device, non_blocking=True)
        ids = ids.to(device, non_blocking=True)
        labels = labels.to(device, non_blocking=True)
    return dense, ids, labels


class SyntheticCriteoDataset(torch.utils.data.Dataset):
    """Map-style synthetic Criteo dataset (deterministic per index)."""

    def __init__(self, n_samples: int,
                 table_sizes: Optional[List[int]] = None,
                 batch_size: int = 1):
        self.n_samples = n_samples
        self.table_sizes = table_sizes or DEFAULT_TABLE_SIZES
        self.batch_size = batch_size

    def __len__(self) -> int:
        return self.n_samples // self.batch_size

    def __getitem__(self, idx: int):
        dense, ids, labels = synthetic_criteo_batch(
            self.batch_size, self.table_sizes, seed=idx)
        if self.batch_size == 1:
            return dense[0], ids[0], labels[0]
        return dense, ids, labels


class SyntheticImageDataset(torch.utils.data.Dataset):
    """ImageNet-shaped (or MNIST-shaped) synthetic classification data."""

    def __init__(self, n_samples: int, shape=(3, 224, 224),
                 n_classes: int = 1000):
        self.n_samples = n_samples
        self.shape = shape
        self.n_classes = n_classes

    def __len__(self) -> int:
        return self.n_samples

    def __getitem__(self, idx: int):
        gen = torch.Generator().manual_seed(idx)
        x = torch.randn(*self.shape, generator=gen)
        y = int(torch.randint(0, self.n_classes, (1,), generator=gen))
        return x, y
