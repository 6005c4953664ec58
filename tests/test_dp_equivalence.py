"""dpN == dp1 loss-trajectory equivalence for the FULL bench model.

The bench's multi-GPU path (WideAndDeep sharded embeddings +
BucketedDataParallel + FusedSGD + overlapped sparse sync) has to produce
the same optimization trajectory as a single process on the same global
batch — the gloo dry-run criterion for the 8-GPU scale run (VERDICT r1
next-round #1; reference data plane it re-implements:
``/root/reference/tf_yarn/pytorch/tasks/worker.py:94-121``).

Identical init is forced by overwriting dense params and copying slices
of one deterministic global table into each rank's shard (construction
seeds differ per world size, so post-construction overwrite is the only
portable way to pin them).
"""

import pytest
import torch
import torch.multiprocessing as mp

from tf_yarn_amd.kv import KVClient, KVServer

F, D, ROWS = 8, 8, 100
HIDDEN = (32, 16)
GLOBAL_B = 48
STEPS = 4
LR = 0.05
TABLES = [ROWS] * F


def _global_tables():
    g = torch.Generator().manual_seed(321)
    deep = torch.randn(ROWS * F, D, generator=g) * 0.2
    wide = torch.randn(ROWS * F, 1, generator=g) * 0.01
    return deep, wide


def _global_data():
    g = torch.Generator().manual_seed(99)
    dense = torch.randn(STEPS, GLOBAL_B, 13, generator=g)
    ids = torch.randint(0, ROWS, (STEPS, GLOBAL_B, F), generator=g)
    labels = (torch.rand(STEPS, GLOBAL_B, generator=g) < 0.3).float()
    return dense, ids, labels


def _pin_dense_params(model):
    """Deterministic dense init independent of construction RNG state."""
    for i, (n, p) in enumerate(sorted(model.named_parameters())):
        if getattr(p, "_miyarn_sparse", False):
            continue
        g = torch.Generator().manual_seed(5000 + i)
        with torch.no_grad():
            p.copy_(torch.randn(p.shape, generator=g) * 0.05)


def _shard_rows(table, rank, world):
    parts = [table[f * ROWS:(f + 1) * ROWS] for f in range(F)
             if f % world == rank]
    return torch.cat(parts)


def _make_model(rank: int, world: int):
    from tf_yarn_amd.models.wide_deep import WideAndDeep, _DeepInput
    torch.manual_seed(0)
    model = WideAndDeep(table_sizes=TABLES, embedding_dim=D,
                        hidden=HIDDEN, sharded=True)
    _pin_dense_params(model)
    deep, wide = _global_tables()
    emb = model.embeddings
    with torch.no_grad():
        emb.weight.copy_(_shard_rows(deep, rank, world))
        emb.wide_weight.copy_(_shard_rows(wide, rank, world))
        if world > 1:
            # The sharded path assembles embedding features in
            # owner-permuted order; permute the first MLP layer's
            # embedding columns so the network computes the same
            # function as dp1's natural order.
            perm = [f for s in range(world) for f in range(F)
                    if f % world == s]
            pad = _DeepInput.DENSE_PAD
            w1 = model.mlp[0].weight
            new = w1.clone()
            for j, f in enumerate(perm):
                new[:, pad + j * D:pad + (j + 1) * D] = \
                    w1[:, pad + f * D:pad + (f + 1) * D]
            w1.copy_(new)
    return model


def _run_steps(model, ddp, opt, rank: int, world: int):
    """Returns the per-step GLOBAL mean loss trajectory."""
    import torch.distributed as dist
    dense, ids, labels = _global_data()
    b = GLOBAL_B // world
    module = model
    losses = []
    for s in range(STEPS):
        d = dense[s, rank * b:(rank + 1) * b]
        i = ids[s, rank * b:(rank + 1) * b]
        y = labels[s, rank * b:(rank + 1) * b]
        opt.zero_grad(set_to_none=(world == 1))
        if world > 1:
            ddp.zero_grad_buffers()
        loss = ddp(d, i, labels=y)
        loss.backward()
        module.start_sparse_sync()
        opt.step()
        module.finish_sparse_sync(LR)
        gl = loss.detach().clone()
        if world > 1:
            dist.all_reduce(gl)
            gl /= world
        losses.append(float(gl))
    return losses


def _dp1_trajectory():
    from tf_yarn_amd.ops.optim import FusedSGD
    model = _make_model(0, 1)
    opt = FusedSGD([p for p in model.parameters()
                    if not getattr(p, "_miyarn_sparse", False)], lr=LR)
    return _run_steps(model, model, opt, 0, 1)


def _worker(rank, world, kv_addr, out_q):
    from tf_yarn_amd.ops.optim import FusedSGD
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world, backend="gloo",
                            kv_client=client)
    try:
        model = _make_model(rank, world)
        ddp = BucketedDataParallel(model, broadcast_buffers=False)
        opt = FusedSGD([p for p in model.parameters()
                        if not getattr(p, "_miyarn_sparse", False)], lr=LR)
        losses = _run_steps(model, ddp, opt, rank, world)
        out_q.put((rank, losses))
    finally:
        comm.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4, 8])
@pytest.mark.timeout(420)
def test_dpN_matches_dp1_loss_trajectory(world):
    ref = _dp1_trajectory()
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker,
                         args=(r, world, server.address, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world):
            rank, losses = out_q.get(timeout=360)
            results[rank] = losses
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    for rank in range(world):
        for s, (got, want) in enumerate(zip(results[rank], ref)):
            assert got == pytest.approx(want, abs=2e-5), (
                f"dp{world} rank {rank} step {s}: loss {got} != dp1 "
                f"{want} — trajectory diverged (full trajectories: "
                f"{results[rank]} vs {ref})")
