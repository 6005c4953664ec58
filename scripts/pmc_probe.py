#!/usr/bin/env python3
"""Tiny fixed-kernel workload for PMC capture (a few dispatches only)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import tf_yarn_amd.ops._C as C

torch.manual_seed(0)
table = torch.randn(1_000_000 * 26, 16, device="cuda")
ids = torch.randint(0, table.shape[0], (65536 * 26,), device="cuda")
grad = torch.randn(ids.numel(), 16, device="cuda").to(torch.bfloat16)
dy = (torch.randn(65536, 512, device="cuda") / 8).to(torch.bfloat16)
x = (torch.randn(65536, 1024, device="cuda") / 8).to(torch.bfloat16)
for _ in range(3):
    C.emb_fwd(table, ids, True)
    C.emb_bwd_sgd(table, ids, grad, 0.01, 1.0)
    C.wgrad_nt128(dy, x, 16)
torch.cuda.synchronize()
print("pmc probe done")
