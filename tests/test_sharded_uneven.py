"""Sharded embeddings with UNEVEN feature ownership — the actual dp8
bench shape (26 features over 8 ranks = 4,4,3,3,3,3,3,3 per rank).
Every other sharded test splits evenly; the all-to-all split-size
arithmetic for ragged ownership is what the driver's 8-GPU scale run
exercises first, so it gets its own exactness test vs the replicated
single-process reference (gloo emulation on CPU)."""

import pytest
import torch
import torch.multiprocessing as mp

from tf_yarn_amd.kv import KVClient, KVServer

D, ROWS, B = 8, 40, 6
LR = 0.2


def _global_tables(F, seed=55):
    g = torch.Generator().manual_seed(seed)
    deep = torch.randn(ROWS * F, D, generator=g)
    wide = torch.randn(ROWS * F, 1, generator=g) * 0.01
    return deep, wide


def _global_batch(F, W):
    g = torch.Generator().manual_seed(8)
    return torch.randint(0, ROWS, (W * B, F), generator=g)


def _shard_rows(table, rank, world, F):
    parts = [table[f * ROWS:(f + 1) * ROWS] for f in range(F)
             if f % world == rank]
    return torch.cat(parts)


def _reference(F, W):
    deep, wide = _global_tables(F)
    deep = deep.clone().requires_grad_(True)
    wide = wide.clone().requires_grad_(True)
    ids = _global_batch(F, W)
    offs = torch.arange(F) * ROWS
    flat = (ids + offs).reshape(-1)
    out = deep.index_select(0, flat).reshape(W * B, F * D)
    wide_out = wide.reshape(-1).index_select(0, flat).reshape(
        W * B, F).sum(dim=1)
    loss = out.float().pow(2).mean() + wide_out.float().pow(2).mean()
    loss.backward()
    return (out.detach(), wide_out.detach(),
            deep.detach() - LR * deep.grad,
            wide.detach() - LR * wide.grad)


def _worker(rank, world, F, kv_addr, out_q):
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    from tf_yarn_amd.parallel import comm
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world, backend="gloo",
                            kv_client=client)
    try:
        emb = ShardedCriteoEmbeddings([ROWS] * F, D)
        deep, wide = _global_tables(F)
        emb.weight.data.copy_(_shard_rows(deep, rank, world, F))
        emb.wide_weight.data.copy_(_shard_rows(wide, rank, world, F))
        ids = _global_batch(F, world)[rank * B:(rank + 1) * B]
        buf = torch.zeros(B, F * D)
        out_buf, wide_out = emb(ids, buf, 0)
        loss = out_buf.float().pow(2).mean() \
            + wide_out.float().pow(2).mean()
        loss.backward()
        emb.apply_sparse_updates(LR)
        out_q.put((rank,
                   out_buf.detach().numpy().copy(),
                   wide_out.detach().numpy().copy(),
                   emb.weight.detach().numpy().copy(),
                   emb.wide_weight.detach().numpy().copy()))
    finally:
        comm.destroy_process_group()


@pytest.mark.parametrize("world,F", [(4, 6), (3, 26), (8, 26)])
@pytest.mark.timeout(420)
def test_uneven_sharding_matches_replicated(world, F):
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker,
                         args=(r, world, F, server.address, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world):
            r, out, wide, dw, ww = out_q.get(timeout=360)
            results[r] = (torch.from_numpy(out), torch.from_numpy(wide),
                          torch.from_numpy(dw), torch.from_numpy(ww))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()

    ref_out, ref_wide, ref_deep, ref_wide_t = _reference(F, world)
    perm = [f for s in range(world) for f in range(F) if f % world == s]
    for r in range(world):
        out, wide, dw, ww = results[r]
        ref_r = ref_out[r * B:(r + 1) * B].reshape(B, F, D)
        expect = ref_r[:, perm, :].reshape(B, F * D)
        assert torch.allclose(out, expect, atol=1e-5), \
            f"W={world} F={F} rank {r} forward mismatch"
        assert torch.allclose(wide, ref_wide[r * B:(r + 1) * B],
                              atol=1e-5), f"rank {r} wide mismatch"
        assert torch.allclose(dw, _shard_rows(ref_deep, r, world, F),
                              atol=1e-5), f"rank {r} deep update"
        assert torch.allclose(ww, _shard_rows(ref_wide_t, r, world, F),
                              atol=1e-5), f"rank {r} wide update"
