#!/usr/bin/env python3
"""A/B micro-benchmark: atomic vs binned sparse scatter at the bench
shape (b=65536, 26 features, 1M rows/feature, dim 16 deep + dim 1 wide).

Round-1 floor: emb_bwd_sgd 340 us + emb_scatter_sum 90 us (805 GB/s of
random fp32 atomics).  Round-2 target: combined < 300 us;
the binned LDS-dedup kernel aims much lower."""

import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tf_yarn_amd import ops  # noqa: E402

B = 65536
F = 26
ROWS_PER = 1_000_000
DIM = 16


def timeit(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    n_rows = ROWS_PER * F
    table = torch.zeros(n_rows, DIM, device="cuda")
    wide = torch.zeros(n_rows, 1, device="cuda")
    offs = (torch.arange(F, device="cuda") * ROWS_PER).unsqueeze(0)
    ids = (torch.randint(0, ROWS_PER, (B, F), device="cuda")
           + offs).reshape(-1).contiguous()
    grad = torch.randn(B * F, DIM, device="cuda").to(torch.bfloat16)
    gw = torch.randn(B, device="cuda").to(torch.bfloat16)
    ids2d = ids.reshape(B, F)

    t_atomic_deep = timeit(lambda: ops.emb_bwd_sgd(
        table, ids, grad, lr=0.01, scale=1.0))
    t_atomic_wide = timeit(lambda: ops.emb_scatter_sum(
        wide, ids2d, gw, alpha=-0.01))

    rb = ops.pick_region_bits(n_rows, ids.numel())
    t_perm = timeit(lambda: ops.binned_permutation(ids, n_rows, rb))
    perm = ops.binned_permutation(ids, n_rows, rb)
    t_binned_deep = timeit(lambda: ops.emb_bwd_sgd_binned(
        table, ids, grad, lr=0.01, scale=1.0, perm=perm))
    t_binned_wide = timeit(lambda: ops.emb_scatter_sum_binned(
        wide, ids, gw, alpha=-0.01, perm=perm))

    atomic = t_atomic_deep + t_atomic_wide
    binned = t_perm + t_binned_deep + t_binned_wide
    print(f"region_bits={rb} n_bins={n_rows >> rb}")
    print(f"atomic : deep {t_atomic_deep:7.1f}us  wide "
          f"{t_atomic_wide:6.1f}us  total {atomic:7.1f}us")
    print(f"binned : perm {t_perm:6.1f}us  deep {t_binned_deep:7.1f}us  "
          f"wide {t_binned_wide:6.1f}us  total {binned:7.1f}us")
    print(f"speedup: {atomic / binned:.2f}x "
          f"({'PASS' if binned < 300 else 'MISS'} <300us gate)")

    # sweep region_bits around the heuristic
    for bits in range(max(7, rb - 3), min(15, rb + 3)):
        p = ops.binned_permutation(ids, n_rows, bits)
        tp = timeit(lambda: ops.binned_permutation(ids, n_rows, bits))
        td = timeit(lambda: ops.emb_bwd_sgd_binned(
            table, ids, grad, lr=0.01, scale=1.0, perm=p))
        tw = timeit(lambda: ops.emb_scatter_sum_binned(
            wide, ids, gw, alpha=-0.01, perm=p))
        print(f"  bits={bits:2d}: perm {tp:6.1f} deep {td:7.1f} "
              f"wide {tw:6.1f} total {tp + td + tw:7.1f}us")


if __name__ == "__main__":
    main()
