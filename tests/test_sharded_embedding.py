"""Feature-sharded embeddings: exact parity with a single-process
replicated reference, over gloo world_size=2 (the all-to-all runs through
the gloo emulation; on GPU the same code takes RCCL all_to_all_single)."""

import pytest
import torch
import torch.multiprocessing as mp
from torch import nn

from tf_yarn_amd.kv import KVClient, KVServer

F, D, ROWS, B, W = 4, 8, 50, 6, 2
TABLES = [ROWS] * F
LR = 0.25


def _global_tables(seed=123):
    g = torch.Generator().manual_seed(seed)
    deep = torch.randn(ROWS * F, D, generator=g)
    wide = torch.randn(ROWS * F, 1, generator=g) * 0.01
    return deep, wide


def _global_batch():
    g = torch.Generator().manual_seed(7)
    return torch.randint(0, ROWS, (W * B, F), generator=g)


def _reference():
    """Single-process full-table training step; returns (outputs
    [WB, F*D] in NATURAL feature order, wide [WB], updated tables)."""
    deep, wide = _global_tables()
    deep = deep.clone().requires_grad_(True)
    wide = wide.clone().requires_grad_(True)
    ids = _global_batch()
    offs = torch.arange(F) * ROWS
    flat = (ids + offs).reshape(-1)
    out = deep.index_select(0, flat).reshape(W * B, F * D)
    wide_out = wide.reshape(-1).index_select(0, flat).reshape(
        W * B, F).sum(dim=1)
    # global-mean loss over the whole (W*B) batch
    loss = out.float().pow(2).mean() + wide_out.float().pow(2).mean()
    loss.backward()
    new_deep = deep.detach() - LR * deep.grad
    new_wide = wide.detach() - LR * wide.grad
    return out.detach(), wide_out.detach(), new_deep, new_wide


def _shard_rows(table, rank):
    """Rows of the features rank owns (f % W == rank), feature-major."""
    parts = [table[f * ROWS:(f + 1) * ROWS] for f in range(F)
             if f % W == rank]
    return torch.cat(parts)


def _worker(rank, kv_addr, out_q):
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    from tf_yarn_amd.parallel import comm
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=W, backend="gloo",
                            kv_client=client)
    try:
        emb = ShardedCriteoEmbeddings(TABLES, D)
        deep, wide = _global_tables()
        emb.weight.data.copy_(_shard_rows(deep, rank))
        emb.wide_weight.data.copy_(_shard_rows(wide, rank))
        ids = _global_batch()[rank * B:(rank + 1) * B]
        buf = torch.zeros(B, F * D)
        out_buf, wide_out = emb(ids, buf, 0)
        # local-mean loss; global equivalence needs the 1/W apply scale
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


@pytest.mark.timeout(180)
def test_sharded_matches_replicated_reference():
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, server.address, out_q))
             for r in range(W)]
    for p in procs:
        p.start()
    try:
        results = {}
        for _ in range(W):
            r, out, wide, dw, ww = out_q.get(timeout=150)
            results[r] = (torch.from_numpy(out), torch.from_numpy(wide),
                          torch.from_numpy(dw), torch.from_numpy(ww))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()

    ref_out, ref_wide, ref_deep, ref_wide_t = _reference()
    # owner-permuted feature order: [f for s in range(W) for f owned by s]
    perm = [f for s in range(W) for f in range(F) if f % W == s]
    for r in range(W):
        out, wide, dw, ww = results[r]
        ref_r = ref_out[r * B:(r + 1) * B].reshape(B, F, D)
        expect = ref_r[:, perm, :].reshape(B, F * D)
        assert torch.allclose(out, expect, atol=1e-5), \
            f"rank {r} forward mismatch"
        assert torch.allclose(wide, ref_wide[r * B:(r + 1) * B],
                              atol=1e-5), f"rank {r} wide mismatch"
        assert torch.allclose(dw, _shard_rows(ref_deep, r), atol=1e-5), \
            f"rank {r} deep update mismatch"
        assert torch.allclose(ww, _shard_rows(ref_wide_t, r),
                              atol=1e-5), f"rank {r} wide update mismatch"


def test_sharded_local_world1_matches_replicated():
    """W == 1 path equals the plain local computation."""
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    emb = ShardedCriteoEmbeddings(TABLES, D)
    deep, wide = _global_tables()
    emb.weight.data.copy_(deep)
    emb.wide_weight.data.copy_(wide)
    ids = _global_batch()[:B]
    buf = torch.zeros(B, F * D)
    out_buf, wide_out = emb(ids, buf, 0)
    offs = torch.arange(F) * ROWS
    flat = (ids + offs).reshape(-1)
    ref = deep.index_select(0, flat).reshape(B, F * D)
    ref_wide = wide.reshape(-1).index_select(0, flat).reshape(
        B, F).sum(dim=1)
    assert torch.allclose(out_buf, ref, atol=1e-5)
    assert torch.allclose(wide_out, ref_wide, atol=1e-5)
    # backward + apply runs
    (out_buf.pow(2).mean() + wide_out.pow(2).mean()).backward()
    emb.apply_sparse_updates(0.1)


def _worker4(rank, kv_addr, out_q):
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    from tf_yarn_amd.parallel import comm
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=4, backend="gloo",
                            kv_client=client)
    try:
        import torch as t
        emb = ShardedCriteoEmbeddings([20] * 6, 4)  # 6 feats / 4 ranks
        g = t.Generator().manual_seed(11)
        full = t.randn(120, 4, generator=g)
        own = t.cat([full[f * 20:(f + 1) * 20]
                     for f in range(6) if f % 4 == rank])
        emb.weight.data.copy_(own)
        ids = t.randint(0, 20, (3, 6),
                        generator=t.Generator().manual_seed(rank))
        buf = t.zeros(3, 24)
        out_buf, wide = emb(ids, buf, 0)
        # reference: local gather on the full table (perm feature order)
        offs = t.arange(6) * 20
        perm = [f for s in range(4) for f in range(6) if f % 4 == s]
        flat = (ids + offs)[:, perm].reshape(-1)
        ref = full.index_select(0, flat).reshape(3, 24)
        out_q.put((rank, bool(t.allclose(out_buf, ref, atol=1e-5))))
    finally:
        comm.destroy_process_group()


@pytest.mark.timeout(180)
def test_sharded_forward_world4_uneven_features():
    """6 features over 4 ranks (2,2,1,1): uneven alltoall splits."""
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker4, args=(r, server.address, out_q))
             for r in range(4)]
    for p in procs:
        p.start()
    try:
        results = dict(out_q.get(timeout=150) for _ in range(4))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    assert all(results.values()), results
