"""Synthetic tabular dataset helper (the winequality.py analog of the
reference examples: a small labeled tabular problem every example shares)."""

import torch

N_FEATURES = 11  # winequality-like shape
N_CLASSES = 2


def get_dataset(n: int = 2048, seed: int = 0):
    """Returns (features [n, 11] fp32, labels [n] int64)."""
    gen = torch.Generator().manual_seed(seed)
    x = torch.randn(n, N_FEATURES, generator=gen)
    w = torch.randn(N_FEATURES, generator=gen)
    y = ((x @ w) + 0.3 * torch.randn(n, generator=gen) > 0).long()
    return x, y


def input_fn(batch_size: int = 64, n: int = 2048, seed: int = 0):
    """Estimator-style input_fn factory."""
    x, y = get_dataset(n, seed)

    def gen():
        for i in range(0, n - batch_size + 1, batch_size):
            yield x[i:i + batch_size], y[i:i + batch_size]

    return gen
