#!/usr/bin/env python3
"""Per-phase timing of the wide-and-deep training step on one GPU.

Times forward / backward / optimizer / sparse-apply separately with CUDA
events, and sweeps batch size, so optimization targets the real bottleneck
(guide §7: measure, don't guess)."""

import argparse
import os
import sys
import time

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_TUNED = os.path.join(_REPO, "tuned", "tunableop_wide_deep.csv")
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_FILENAME" not in os.environ:
    import shutil
    import tempfile
    _tdir = tempfile.mkdtemp(prefix="miyarn_tuned_")
    shutil.copy(_TUNED, os.path.join(_tdir, "probe_tuned0.csv"))
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(
        _tdir, "probe_tuned.csv")

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd.models.synthetic import synthetic_criteo_batch
from tf_yarn_amd.models.wide_deep import WideAndDeep
from tf_yarn_amd.ops.optim import FusedSGD


def probe(batch: int, table_rows: int, steps: int = 20,
          dtype=torch.bfloat16) -> None:
    device = "cuda:0"
    tables = [table_rows] * 26
    torch.manual_seed(0)
    model = WideAndDeep(table_sizes=tables, embedding_dim=16,
                        hidden=(1024, 512, 256),
                        compute_dtype=dtype, sharded=True).to(device)
    opt = FusedSGD([p for p in model.parameters()
                    if not getattr(p, "_miyarn_sparse", False)], lr=0.02)
    loss_fn = torch.nn.BCEWithLogitsLoss()
    batches = [synthetic_criteo_batch(batch, tables, device=device, seed=i)
               for i in range(4)]

    phases = ["forward", "loss", "backward", "opt", "sparse"]
    evs = {p: [(torch.cuda.Event(enable_timing=True),
                torch.cuda.Event(enable_timing=True))
               for _ in range(steps)] for p in phases}

    def run(i, record):
        dense, ids, labels = batches[i % 4]
        opt.zero_grad(set_to_none=False)
        if record:
            evs["forward"][i][0].record()
        logits = model(dense, ids)
        if record:
            evs["forward"][i][1].record()
            evs["loss"][i][0].record()
        loss = loss_fn(logits.float(), labels)
        if record:
            evs["loss"][i][1].record()
            evs["backward"][i][0].record()
        loss.backward()
        if record:
            evs["backward"][i][1].record()
            evs["opt"][i][0].record()
        opt.step()
        if record:
            evs["opt"][i][1].record()
            evs["sparse"][i][0].record()
        model.apply_sparse_updates(0.02)
        if record:
            evs["sparse"][i][1].record()

    for i in range(5):
        run(i, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        run(i, True)
    torch.cuda.synchronize()
    wall = (time.perf_counter() - t0) / steps * 1000

    print(f"batch={batch} rows/table={table_rows} dtype={dtype} "
          f"wall={wall:.3f} ms/step "
          f"({batch / wall * 1000:.0f} ex/s)")
    for p in phases:
        times = [a.elapsed_time(b) for a, b in evs[p]]
        times.sort()
        med = times[len(times) // 2]
        print(f"  {p:<9} median {med:7.3f} ms  min {times[0]:7.3f}  "
              f"max {times[-1]:7.3f}")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--batches", default="16384,65536")
    ap.add_argument("--table-rows", type=int, default=1_000_000)
    ap.add_argument("--steps", type=int, default=20)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    for b in [int(x) for x in args.batches.split(",")]:
        probe(b, args.table_rows, args.steps)
