#!/usr/bin/env python3
"""Minimal per-kernel workload for rocprofv3: N iterations of pass A +
both binned applies + the atomic baselines at the bench shape, so the
kernel-stats table attributes the time per kernel."""

import sys

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tf_yarn_amd import ops  # noqa: E402

B, F, ROWS_PER, DIM = 65536, 26, 1_000_000, 16
ITERS = 20


def main():
    bits = int(sys.argv[1]) if len(sys.argv) > 1 else 12
    torch.manual_seed(0)
    n_rows = ROWS_PER * F
    table = torch.zeros(n_rows, DIM, device="cuda")
    wide = torch.zeros(n_rows, 1, device="cuda")
    offs = (torch.arange(F, device="cuda") * ROWS_PER).unsqueeze(0)
    ids = (torch.randint(0, ROWS_PER, (B, F), device="cuda")
           + offs).reshape(-1).contiguous()
    grad = torch.randn(B * F, DIM, device="cuda").to(torch.bfloat16)
    gw = torch.randn(B, device="cuda").to(torch.bfloat16)
    # warmup
    perm = ops.binned_permutation(ids, n_rows, bits)
    ops.emb_bwd_sgd_binned(table, ids, grad, lr=0.01, scale=1.0, perm=perm)
    ops.emb_scatter_sum_binned(wide, ids, gw, alpha=-0.01, perm=perm)
    ops.emb_bwd_sgd(table, ids, grad, lr=0.01, scale=1.0)
    ops.emb_scatter_sum(wide, ids.reshape(B, F), gw, alpha=-0.01)
    torch.cuda.synchronize()
    for _ in range(ITERS):
        p = ops.binned_permutation(ids, n_rows, bits)
        ops.emb_bwd_sgd_binned(table, ids, grad, lr=0.01, scale=1.0,
                               perm=p)
        ops.emb_scatter_sum_binned(wide, ids, gw, alpha=-0.01, perm=p)
    for _ in range(ITERS):
        ops.emb_bwd_sgd(table, ids, grad, lr=0.01, scale=1.0)
        ops.emb_scatter_sum(wide, ids.reshape(B, F), gw, alpha=-0.01)
    torch.cuda.synchronize()
    print(f"done bits={bits} iters={ITERS}")


if __name__ == "__main__":
    main()
