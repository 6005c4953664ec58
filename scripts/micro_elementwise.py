#!/usr/bin/env python3
"""Microbenchmarks for the elementwise HIP kernels vs torch equivalents.

Within-process interleaved A/B (guide §5.4 rule 24): median over rounds.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import tf_yarn_amd.ops as ops
import tf_yarn_amd.ops._C as C


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    times = []
    for _ in range(5):
        t0 = time.perf_counter()
        for _ in range(iters // 5):
            fn()
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) / (iters // 5))
    times.sort()
    return times[len(times) // 2] * 1e6  # median us


def main():
    assert torch.cuda.is_available()
    for rows, cols in [(65536, 1024), (65536, 512), (65536, 256),
                       (16384, 1024)]:
        dy = torch.randn(rows, cols, device="cuda").to(torch.bfloat16)
        y = torch.relu(torch.randn(rows, cols, device="cuda")
                       ).to(torch.bfloat16)

        t_sep_dx = timeit(lambda: C.bias_relu_bwd(dy, y))
        t_torch_db = timeit(lambda: dy.sum(dim=0))
        t_fused = timeit(lambda: C.bias_relu_bwd_db(dy, y))
        t_torch_both = timeit(
            lambda: (dy * (y > 0), (dy * (y > 0)).sum(dim=0)))
        gb = rows * cols * 2 * 3 / 1e9
        print(f"[{rows}x{cols}] dx-only {t_sep_dx:7.1f}us  "
              f"torch-db {t_torch_db:7.1f}us  "
              f"fused dx+db {t_fused:7.1f}us ({gb/t_fused*1e6:.0f} GB/s) "
              f" torch-both {t_torch_both:7.1f}us")

    # gather/scatter microbench
    for rows_t, n in [(26_000_000, 65536 * 26)]:
        table = torch.randn(rows_t, 16, device="cuda")
        ids = torch.randint(0, rows_t, (n,), device="cuda")
        grad = torch.randn(n, 16, device="cuda").to(torch.bfloat16)
        t_fwd = timeit(lambda: C.emb_fwd(table, ids, True), 20)
        t_bwd = timeit(lambda: C.emb_bwd_sgd(table, ids, grad, 0.01, 1.0),
                       20)
        fwd_gb = n * 16 * (4 + 2) / 1e9
        bwd_gb = n * 16 * (2 + 8) / 1e9  # grad read + atomic rmw
        print(f"[emb {n} rows x16] fwd {t_fwd:7.1f}us "
              f"({fwd_gb/t_fwd*1e6:.0f} GB/s)  "
              f"bwd_sgd {t_bwd:7.1f}us ({bwd_gb/t_bwd*1e6:.0f} GB/s)")


if __name__ == "__main__":
    main()
