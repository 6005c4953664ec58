#!/usr/bin/env python3
"""Flagship benchmark: Criteo wide-and-deep examples/sec on MI355X.

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``.
For N>1 the driver launches it under ``torch.distributed.run`` with one
rank per GPU (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* in env), RCCL over xGMI.
W untimed warmup steps, then exactly K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; MAX time over ranks;
rank 0 prints ONE JSON line.

Metric (BASELINE.json): examples/sec (whole node), Criteo wide-and-deep,
synthetic Criteo-shaped data, random-init weights, bf16 compute, weak
scaling (per-GPU batch fixed).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

# hipBLASLt algorithm tuning for the MLP GEMM shapes (must be set before
# the first GEMM).  A pre-tuned table for the default config ships in
# tuned/ (captured on MI355X); each rank copies it to its own result file
# so tuning only runs for shapes not already in the table.
_REPO = os.path.dirname(os.path.abspath(__file__))
_TUNED = os.path.join(_REPO, "tuned", "tunableop_wide_deep.csv")
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "0")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_FILENAME" not in os.environ:
    import shutil
    import tempfile
    _tdir = tempfile.mkdtemp(prefix="miyarn_tuned_")
    _dev = os.environ.get("LOCAL_RANK", "0")
    shutil.copy(_TUNED, os.path.join(_tdir, f"bench_tuned{_dev}.csv"))
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(
        _tdir, "bench_tuned.csv")

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from tf_yarn_amd.models.synthetic import synthetic_criteo_batch  # noqa: E402
from tf_yarn_amd.models.wide_deep import WideAndDeep  # noqa: E402
from tf_yarn_amd.ops.optim import FusedSGD  # noqa: E402
from tf_yarn_amd.parallel.ddp import BucketedDataParallel  # noqa: E402

PER_GPU_BATCH = 65536
TABLE_ROWS_PER_FEATURE = 1_000_000
EMBEDDING_DIM = 16
HIDDEN = (1024, 512, 256)
N_DATA_BATCHES = 8  # distinct synthetic batches cycled through


def log(msg: str) -> None:
    print(f"[bench] {msg}", file=sys.stderr, flush=True)


def _ensure_master_port(rank: int) -> None:
    """Pick MASTER_PORT when the launcher did not set one (torchrun always
    does).  A fixed fallback (29500) turns a stale process on the box into
    a rendezvous hang; instead rank 0 probes for a free port starting from
    a base and publishes it through a file all local ranks share (the
    bench contract is single-node)."""
    if "MASTER_PORT" in os.environ:
        return
    import socket
    import tempfile
    rdv = os.path.join(tempfile.gettempdir(),
                       f"miyarn_bench_port_{os.environ.get('MIYARN_RDV_TAG', 'default')}")
    if rank == 0:
        port = None
        for cand in range(29500, 29600):
            with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
                try:
                    s.bind(("127.0.0.1", cand))
                    port = cand
                    break
                except OSError:
                    continue
        if port is None:
            raise RuntimeError("no free port in 29500-29599")
        with open(rdv + ".tmp", "w") as f:
            f.write(str(port))
        os.replace(rdv + ".tmp", rdv)
        os.environ["MASTER_PORT"] = str(port)
        log(f"rank 0 chose MASTER_PORT={port}")
    else:
        start = time.time()
        deadline = start + 120
        while time.time() < deadline:
            try:
                # ignore stale files from earlier runs: rank 0 writes its
                # file AFTER every rank has started
                if os.path.getmtime(rdv) >= start - 5:
                    with open(rdv) as f:
                        os.environ["MASTER_PORT"] = f.read().strip()
                    return
            except (FileNotFoundError, ValueError, OSError):
                pass
            time.sleep(0.2)
        raise RuntimeError("timed out waiting for rank 0's MASTER_PORT file")


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--batch", type=int, default=PER_GPU_BATCH)
    parser.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    parser.add_argument("--table-rows", type=int,
                        default=TABLE_ROWS_PER_FEATURE,
                        help="rows per categorical table (tests use small)")
    parser.add_argument("--probe-steps", type=int, default=3,
                        help="instrumented steps AFTER the timed region "
                             "for the comm_breakdown JSON field (0=off)")
    parser.add_argument("--graphs", action="store_true",
                        help="capture the step in a hipGraph (measured "
                             "2.39 vs 2.10 ms/step eager on MI355X — "
                             "replay sustains higher power and downclocks"
                             ", so off by default)")
    args = parser.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    use_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if use_gpu else "cpu"
    if use_gpu:
        torch.cuda.set_device(local_rank)
    backend = "nccl" if use_gpu else "gloo"

    if world_size > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        _ensure_master_port(rank)
        dist.init_process_group(backend, rank=rank, world_size=world_size)

    compute_dtype = (torch.bfloat16 if args.dtype == "bf16" and use_gpu
                     else torch.float32)
    table_sizes = [args.table_rows] * 26

    torch.manual_seed(1234 + rank)  # sharded tables are per-rank
    # Feature-sharded embeddings + all-to-all: the xGMI-native scaling
    # path (replicated tables would allgather ~54 MB/rank/step).  Dense
    # params still random-init identically via the reducer's broadcast.
    model = WideAndDeep(table_sizes=table_sizes,
                        embedding_dim=EMBEDDING_DIM, hidden=HIDDEN,
                        compute_dtype=compute_dtype,
                        sharded=True).to(device)
    # no BN-style buffers in this model: skip the per-forward broadcast
    ddp = BucketedDataParallel(model, broadcast_buffers=False) \
        if world_size > 1 else model
    module = ddp.module if world_size > 1 else model

    lr = 0.02
    opt = FusedSGD([p for p in model.parameters()
                    if not getattr(p, "_miyarn_sparse", False)], lr=lr)

    # Pre-generate distinct synthetic batches on-device (data=synthetic;
    # varying ids each step so the gather/scatter path is exercised).
    batches = []
    for i in range(N_DATA_BATCHES):
        dense, ids, labels = synthetic_criteo_batch(
            args.batch, table_sizes, device=device, seed=1000 * rank + i)
        batches.append((dense, ids, labels))

    # world_size==1 has no bucket views: set_to_none avoids the fill+
    # accumulate kernels entirely (grads are assigned fresh each step)
    set_to_none = world_size == 1

    # NOTE: running the sparse scatters on a side stream concurrent with
    # the dense optimizer step measured SLOWER (1.81 vs 1.70 ms/step on
    # MI355X): the overlap window (the ~30 us fused-SGD step) is smaller
    # than the per-step cross-stream event cost.  The stream hook stays
    # on apply_sparse_updates for callers with wider windows.
    def eager_step(i: int) -> torch.Tensor:
        dense, ids, labels = batches[i % N_DATA_BATCHES]
        opt.zero_grad(set_to_none=set_to_none)
        # labels passed through: the 3-part logit sum is fused into the
        # BCE loss kernel (ops.bce_head_loss)
        loss = ddp(dense, ids, labels=labels)
        loss.backward()
        # sparse exchange overlaps with the dense optimizer step
        module.start_sparse_sync()
        opt.step()
        module.finish_sparse_sync(lr)
        return loss

    step = eager_step
    if use_gpu and args.graphs:
        # hipGraph capture: the step is launch-dense (~60 kernels); one
        # graph per pre-generated batch replays it with zero host gaps.
        # Capture happens after warmup so optimizer state and tuned GEMM
        # algos are steady; falls back to eager on capture failure.
        for i in range(args.warmup):
            eager_step(i)
        torch.cuda.synchronize()
        try:
            graphs = []
            losses = []
            for i in range(N_DATA_BATCHES):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    losses.append(eager_step(i))
                graphs.append(g)
            torch.cuda.synchronize()

            def step(i: int) -> torch.Tensor:  # noqa: F811
                j = i % N_DATA_BATCHES
                graphs[j].replay()
                return losses[j]

            log("hipGraph capture OK: replaying captured steps")
        except Exception as e:  # noqa: BLE001
            log(f"hipGraph capture failed ({e}); running eager")
            step = eager_step

    log(f"rank {rank}/{world_size} device={device} "
        f"dtype={compute_dtype} batch/gpu={args.batch}")
    for i in range(args.warmup):
        step(i)
    if use_gpu:
        torch.cuda.synchronize()
    if world_size > 1:
        dist.barrier()
    t0 = time.perf_counter()
    last_loss = None
    for i in range(args.steps):
        last_loss = step(args.warmup + i)
    if use_gpu:
        torch.cuda.synchronize()
    if world_size > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks + spread (a straggling rank shows up here)
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if backend == "nccl" else "cpu")
    tmin = t.clone()
    if world_size > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dist.all_reduce(tmin, op=dist.ReduceOp.MIN)
    rank_spread_ms = (float(t.item()) - float(tmin.item())) \
        / args.steps * 1000.0
    elapsed = float(t.item())

    # -- per-phase comm/compute breakdown (extra instrumented steps AFTER
    # the timed region; headline number is unperturbed).  Makes a poor
    # scaling curve attributable: alltoall ids/vec/wide/grad, wide
    # allgather, dense-reducer drain, plus fwd/bwd/opt phase times.
    breakdown = None
    if args.probe_steps > 0:
        from tf_yarn_amd.utils import commprobe
        commprobe.reset()
        commprobe.enable()
        phases = {"forward": 0.0, "backward": 0.0, "opt_sparse": 0.0}
        for i in range(args.probe_steps):
            dense, ids, labels = batches[i % N_DATA_BATCHES]
            opt.zero_grad(set_to_none=set_to_none)
            with commprobe.span("phase_forward"):
                loss = ddp(dense, ids, labels=labels)
            with commprobe.span("phase_backward"):
                loss.backward()
            with commprobe.span("phase_opt_sparse"):
                module.start_sparse_sync()
                opt.step()
                module.finish_sparse_sync(lr)
        del phases
        summ = commprobe.summary()
        commprobe.disable()
        n = float(args.probe_steps)
        breakdown = {tag: {"ms_per_step": v["ms"] / n,
                           "calls_per_step": v["count"] / n}
                     for tag, v in sorted(summ.items())}
        # cross-rank MAX of total a2a time: a slow link shows up here
        a2a_ms = sum(v["ms"] for tag, v in summ.items()
                     if tag.startswith(("a2a_", "ag_"))) / n
        tt = torch.tensor([a2a_ms], dtype=torch.float64,
                          device=device if backend == "nccl" else "cpu")
        if world_size > 1:
            dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        breakdown["sparse_comm_max_over_ranks"] = {
            "ms_per_step": float(tt.item()), "calls_per_step": 1.0}

    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000.0
        examples_per_sec = args.batch * world_size * args.steps / elapsed
        result = {
            "metric": "examples/sec (whole node), Criteo wide-and-deep",
            "value": examples_per_sec,
            "unit": "examples/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": args.dtype if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "wide_and_deep_criteo",
                "global_batch": args.batch * world_size,
                "per_gpu_batch": args.batch,
                "dense_features": 13,
                "sparse_features": 26,
                "embedding_dim": EMBEDDING_DIM,
                "table_rows_per_feature": args.table_rows,
                "hidden": list(HIDDEN),
                "parallelism": f"dp{world_size}",
                "final_loss": float(last_loss.item())
                if last_loss is not None else None,
                "alltoall_mode": __import__(
                    "tf_yarn_amd.models.sharded_embedding",
                    fromlist=["negotiated_alltoall_mode"]
                ).negotiated_alltoall_mode(),
                "rank_spread_ms_per_step": rank_spread_ms,
            },
            "comm_breakdown": breakdown,
        }
        print(json.dumps(result), flush=True)

    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
