"""Executable spec for the round-2 binned-scatter kernel
(docs/Kernels.md "Round-2 kernel designs").

The HIP kernel will run two passes:
  A) bin each (id, grad-row) pair by table region (id >> REGION_BITS),
     reserving output slots per bin with atomic counters;
  B) apply scatter+SGD bin by bin so concurrent atomics land in one
     region (L2/DRAM-row locality).

This file validates the ALGORITHM (permutation + bin ranges + final
table state) in pure torch so the HIP port only has to match it."""

import torch

REGION_BITS = 14  # 16K-row regions


def binned_permutation(ids: torch.Tensor, n_rows: int):
    """Pass A reference: returns (order, bin_starts) where order is a
    permutation of update indices grouped by region, built with the
    counter-reservation scheme the kernel will use (no stable sort)."""
    n_bins = (n_rows + (1 << REGION_BITS) - 1) >> REGION_BITS
    bins = (ids >> REGION_BITS).long()
    counts = torch.bincount(bins, minlength=n_bins)
    starts = torch.zeros(n_bins + 1, dtype=torch.long)
    torch.cumsum(counts, 0, out=starts[1:])
    cursor = starts[:-1].clone()
    order = torch.empty_like(ids, dtype=torch.long)
    for i in range(ids.numel()):  # kernel: atomicAdd on cursor[bin]
        b = bins[i]
        order[cursor[b]] = i
        cursor[b] += 1
    return order, starts


def binned_scatter_sgd(table, ids, grads, lr, scale):
    """Pass B reference: apply updates bin by bin via the permutation."""
    order, starts = binned_permutation(ids, table.shape[0])
    for b in range(starts.numel() - 1):
        sl = order[starts[b]:starts[b + 1]]
        if sl.numel():
            table.index_add_(0, ids[sl], grads[sl],
                             alpha=-lr * scale)


def test_permutation_is_complete_and_region_grouped():
    torch.manual_seed(1)
    n_rows, n_upd = 100_000, 4096
    ids = torch.randint(0, n_rows, (n_upd,))
    order, starts = binned_permutation(ids, n_rows)
    assert sorted(order.tolist()) == list(range(n_upd))  # permutation
    # within each bin range, all ids belong to that region
    for b in range(starts.numel() - 1):
        sl = order[starts[b]:starts[b + 1]]
        if sl.numel():
            assert ((ids[sl] >> REGION_BITS) == b).all()


def test_binned_scatter_matches_direct_index_add():
    torch.manual_seed(2)
    n_rows, n_upd, dim = 50_000, 8192, 16
    table = torch.randn(n_rows, dim)
    ref = table.clone()
    ids = torch.randint(0, n_rows, (n_upd,))
    grads = torch.randn(n_upd, dim)
    binned_scatter_sgd(table, ids, grads, lr=0.1, scale=0.5)
    ref.index_add_(0, ids, grads, alpha=-0.05)
    assert torch.allclose(table, ref, atol=1e-5)


def test_single_bin_and_empty_bins_edge_cases():
    table = torch.zeros(4, 2)  # n_rows < region size: one bin
    ids = torch.tensor([3, 0, 3])
    grads = torch.ones(3, 2)
    binned_scatter_sgd(table, ids, grads, lr=1.0, scale=1.0)
    assert torch.allclose(table[3], torch.tensor([-2.0, -2.0]))
    assert torch.allclose(table[0], torch.tensor([-1.0, -1.0]))
    assert torch.allclose(table[1], torch.zeros(2))
