"""Reducer numerics: BucketedDataParallel vs single-process reference.

Multi-process on CPU via gloo, world_size=2 (the distributed
path must be covered by CPU multi-process tests; GPU runs only confirm).
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import nn

from tf_yarn_amd.kv import KVClient, KVServer


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))


def _worker_grads(rank, world_size, kv_addr, out_q, bucket_cap_mb,
                  set_to_none, steps):
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        model = _make_model()
        ddp = BucketedDataParallel(model, bucket_cap_mb=bucket_cap_mb)
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        for step in range(steps):
            torch.manual_seed(100 + step * world_size + rank)
            x = torch.randn(8, 16)
            y = torch.randn(8, 4)
            opt.zero_grad(set_to_none=set_to_none)
            if set_to_none:
                ddp.zero_grad_buffers()
            loss = nn.functional.mse_loss(ddp(x), y)
            loss.backward()
            opt.step()
        # ship by value (numpy): torch tensors over mp.Queue use fd-sharing
        # and hang when the producer exits before the consumer reads
        grads = [p.grad.numpy().copy() for p in model.parameters()]
        params = [p.detach().numpy().copy() for p in model.parameters()]
        out_q.put((rank, grads, params))
    finally:
        comm.destroy_process_group()


def _reference_grads_params(world_size, steps):
    """Single-process equivalent: average grads over all ranks' batches."""
    model = _make_model()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    for step in range(steps):
        opt.zero_grad()
        losses = []
        for rank in range(world_size):
            torch.manual_seed(100 + step * world_size + rank)
            x = torch.randn(8, 16)
            y = torch.randn(8, 4)
            losses.append(nn.functional.mse_loss(model(x), y))
        # mean over ranks == allreduce-AVG of per-rank grads
        (sum(losses) / world_size).backward()
        opt.step()
    return ([p.grad.clone() for p in model.parameters()],
            [p.detach().clone() for p in model.parameters()])


@pytest.mark.parametrize("bucket_cap_mb,set_to_none", [
    (32, False),
    (0.001, False),   # force many buckets
    (32, True),       # zero_grad(set_to_none=True) re-link path
])
def test_bucketed_ddp_matches_reference(bucket_cap_mb, set_to_none):
    world_size = 2
    steps = 3
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(
        target=_worker_grads,
        args=(r, world_size, server.address, out_q, bucket_cap_mb,
              set_to_none, steps))
        for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, grads, params = out_q.get(timeout=120)
            results[rank] = (grads, params)
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    ref_grads, ref_params = _reference_grads_params(world_size, steps)
    for rank in range(world_size):
        grads, params = results[rank]
        for g, rg in zip(grads, ref_grads):
            assert torch.allclose(torch.from_numpy(g), rg, atol=1e-6), \
                f"rank {rank}: grads diverge from reference"
        for p_, rp in zip(params, ref_params):
            assert torch.allclose(torch.from_numpy(p_), rp, atol=1e-6), \
                f"rank {rank}: params diverge from reference"


def test_single_process_passthrough():
    """world_size==1 (no process group): the wrapper is a no-op."""
    model = _make_model()
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    ddp = BucketedDataParallel(model)
    x = torch.randn(4, 16)
    loss = ddp(x).sum()
    loss.backward()
    assert all(p.grad is not None for p in model.parameters())


def _rendezvous_worker(rank, world_size, kv_addr, out_q):
    from tf_yarn_amd.parallel import comm
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        t = torch.tensor([float(rank + 1)])
        dist.all_reduce(t)
        out_q.put((rank, t.item()))
    finally:
        comm.destroy_process_group()


def test_kv_rendezvous_store_gloo():
    """Process-group bootstrap through the framework's own KV store."""
    world_size = 2
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_rendezvous_worker,
                         args=(r, world_size, server.address, out_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    try:
        results = dict(out_q.get(timeout=120) for _ in range(world_size))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    assert results == {0: 3.0, 1: 3.0}


def _worker_no_sync(rank, world_size, kv_addr, out_q):
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        model = _make_model()
        ddp = BucketedDataParallel(model, bucket_cap_mb=1)
        torch.manual_seed(200 + rank)
        x1, y1 = torch.randn(8, 16), torch.randn(8, 4)
        x2, y2 = torch.randn(8, 16), torch.randn(8, 4)
        with ddp.no_sync():  # accumulation step: NO allreduce
            nn.functional.mse_loss(ddp(x1), y1).backward()
        nn.functional.mse_loss(ddp(x2), y2).backward()  # synced step
        grads = [p.grad.numpy().copy() for p in model.parameters()]
        out_q.put((rank, grads))
    finally:
        comm.destroy_process_group()


def test_no_sync_gradient_accumulation():
    """no_sync parity with torch DDP semantics: grads accumulate locally
    during no_sync and the next synced backward averages the TOTALS."""
    world_size = 2
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_no_sync,
                         args=(r, world_size, server.address, out_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, grads = out_q.get(timeout=120)
            results[rank] = grads
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    # reference: mean over ranks of (g_batch1 + g_batch2)
    ref_model = _make_model()
    accum = [torch.zeros_like(p) for p in ref_model.parameters()]
    for rank in range(world_size):
        m = _make_model()
        torch.manual_seed(200 + rank)
        x1, y1 = torch.randn(8, 16), torch.randn(8, 4)
        x2, y2 = torch.randn(8, 16), torch.randn(8, 4)
        nn.functional.mse_loss(m(x1), y1).backward()
        nn.functional.mse_loss(m(x2), y2).backward()
        for a, p in zip(accum, m.parameters()):
            a += p.grad / world_size
    for rank in range(world_size):
        for g, ref in zip(results[rank], accum):
            assert torch.allclose(torch.from_numpy(g), ref, atol=1e-6), \
                f"rank {rank}: no_sync accumulation diverges"


class _BranchyModel(nn.Module):
    """`side` is unused in forward when use_side=False."""

    def __init__(self):
        super().__init__()
        torch.manual_seed(7)
        self.main = nn.Linear(16, 4)
        self.side = nn.Linear(16, 4)

    def forward(self, x, use_side=False):
        out = self.main(x)
        if use_side:
            out = out + self.side(x)
        return out


def _worker_unused(rank, world_size, kv_addr, out_q):
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        model = _BranchyModel()
        ddp = BucketedDataParallel(model, bucket_cap_mb=1,
                                   find_unused_parameters=True)
        torch.manual_seed(300 + rank)
        x, y = torch.randn(8, 16), torch.randn(8, 4)
        ddp.zero_grad_buffers()
        nn.functional.mse_loss(ddp(x, use_side=False), y).backward()
        grads = {n: p.grad.numpy().copy()
                 for n, p in model.named_parameters()}
        out_q.put((rank, grads))
    finally:
        comm.destroy_process_group()


def test_find_unused_parameters_syncs_partial_graph():
    """A branch unused in forward must not deadlock or error with
    find_unused_parameters=True; used grads average across ranks and
    unused grads stay zero (torch DDP semantics)."""
    world_size = 2
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_unused,
                         args=(r, world_size, server.address, out_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, grads = out_q.get(timeout=120)
            results[rank] = grads
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    # reference for the used branch
    ref = {n: torch.zeros_like(p)
           for n, p in _BranchyModel().named_parameters()}
    for rank in range(world_size):
        m = _BranchyModel()
        torch.manual_seed(300 + rank)
        x, y = torch.randn(8, 16), torch.randn(8, 4)
        nn.functional.mse_loss(m(x, use_side=False), y).backward()
        for n, p in m.named_parameters():
            if p.grad is not None:
                ref[n] += p.grad / world_size
    for rank in range(world_size):
        for n, g in results[rank].items():
            assert torch.allclose(torch.from_numpy(g), ref[n],
                                  atol=1e-6), f"rank {rank} {n}"


def _worker_buffers(rank, world_size, kv_addr, out_q):
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        torch.manual_seed(50)  # same params everywhere
        model = nn.Sequential(nn.Linear(8, 8), nn.BatchNorm1d(8))
        # per-rank different buffer state before wrapping
        model[1].running_mean.fill_(float(rank + 1))
        ddp = BucketedDataParallel(model, broadcast_buffers=True)
        ddp(torch.randn(4, 8))  # forward triggers rank-0 buffer broadcast
        out_q.put((rank, model[1].running_mean.numpy().copy()))
    finally:
        comm.destroy_process_group()


def test_broadcast_buffers_syncs_from_rank0():
    world_size = 2
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_buffers,
                         args=(r, world_size, server.address, out_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, buf = out_q.get(timeout=120)
            results[rank] = buf
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    # both ranks hold rank 0's PRE-forward buffer value in the broadcast
    # position... the broadcast happens before forward, so rank 1 sees
    # rank 0's fill value (1.0) folded into its own post-forward update.
    assert torch.allclose(torch.from_numpy(results[0]),
                          torch.from_numpy(results[1]), atol=1e-6), \
        "buffers diverged across ranks after broadcast_buffers forward"


def _worker_partial_strict(rank, world_size, kv_addr, out_q):
    from tf_yarn_amd.parallel import comm
    from tf_yarn_amd.parallel.ddp import BucketedDataParallel
    client = KVClient(kv_addr)
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="gloo", kv_client=client)
    try:
        model = _BranchyModel()
        # One big bucket: the unused `side` params make it PARTIAL, the
        # regression case (silent stale allreduce) from round 1.
        ddp = BucketedDataParallel(model, bucket_cap_mb=64,
                                   find_unused_parameters=False)
        torch.manual_seed(300 + rank)
        x, y = torch.randn(8, 16), torch.randn(8, 4)
        ddp.zero_grad_buffers()
        err = ""
        try:
            nn.functional.mse_loss(ddp(x, use_side=False), y).backward()
        except RuntimeError as e:
            err = str(e)
        # after the loud error the reducer state must be clean: a full
        # step (use_side=True) must sync correctly on both ranks
        ddp.zero_grad_buffers()
        torch.manual_seed(400 + rank)
        x2, y2 = torch.randn(8, 16), torch.randn(8, 4)
        nn.functional.mse_loss(ddp(x2, use_side=True), y2).backward()
        grads = {n: p.grad.numpy().copy()
                 for n, p in model.named_parameters()}
        out_q.put((rank, err, grads))
    finally:
        comm.destroy_process_group()


def test_partial_bucket_errors_loudly_and_recovers():
    """find_unused_parameters=False + a partially-filled bucket must raise
    (naming the unused params) instead of silently allreducing stale view
    contents, and the reducer must stay usable afterwards."""
    world_size = 2
    server = KVServer()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_partial_strict,
                         args=(r, world_size, server.address, out_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, err, grads = out_q.get(timeout=120)
            results[rank] = (err, grads)
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        server.stop()
    # reference grads for the recovery step (full graph, averaged)
    ref = {n: torch.zeros_like(p)
           for n, p in _BranchyModel().named_parameters()}
    for rank in range(world_size):
        m = _BranchyModel()
        torch.manual_seed(400 + rank)
        x2, y2 = torch.randn(8, 16), torch.randn(8, 4)
        nn.functional.mse_loss(m(x2, use_side=True), y2).backward()
        for n, p in m.named_parameters():
            ref[n] += p.grad / world_size
    for rank in range(world_size):
        err, grads = results[rank]
        assert "no gradient" in err and "side" in err, \
            f"rank {rank}: expected loud unused-param error, got: {err!r}"
        for n, g in grads.items():
            assert torch.allclose(torch.from_numpy(g), ref[n],
                                  atol=1e-6), f"rank {rank} {n}"
