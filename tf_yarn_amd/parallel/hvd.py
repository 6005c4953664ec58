"""Horovod-style optimizer-level allreduce on RCCL over xGMI.

Re-implements the surface the reference wires through Horovod-gloo
(``hvd.DistributedOptimizer``,
``BroadcastGlobalVariablesHook/Callback(0)``, ``gloo_allred_task.py``):
the ``DistributedOptimizer`` wrapper averages gradients across ranks
before ``step()``, and rank-0 weight broadcast mirrors
``BroadcastGlobalVariables``.

Round-2 redesign: gradients live as views into
PREALLOCATED flat buckets, and per-parameter post-accumulate hooks
launch each bucket's async allreduce as soon as its last grad lands —
overlapped with the rest of backward, exactly the reducer technique
proven in ``parallel/ddp.py`` — instead of the round-1 synchronous
``torch.cat`` + copy-back inside ``step()`` (two extra full-gradient
memory passes per step, zero overlap; Horovod's own background fusion
thread is the reference behavior being matched).  ``step()`` just
drains the in-flight works.
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch
import torch.distributed as dist
from torch.optim import Optimizer

logger = logging.getLogger(__name__)

DEFAULT_BUCKET_BYTES = 32 * 1024 * 1024


def _world(process_group) -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(process_group)
    return 1


def _bucketize(tensors: List[torch.Tensor],
               cap_bytes: int) -> List[List[torch.Tensor]]:
    groups: List[List[torch.Tensor]] = []
    cur: List[torch.Tensor] = []
    cur_bytes = 0
    key = None
    for t in tensors:
        k = (t.device, t.dtype)
        nb = t.numel() * t.element_size()
        if cur and (k != key or cur_bytes + nb > cap_bytes):
            groups.append(cur)
            cur, cur_bytes = [], 0
        cur.append(t)
        cur_bytes += nb
        key = k
    if cur:
        groups.append(cur)
    return groups


@torch.no_grad()
def allreduce_tensors(tensors: List[torch.Tensor], average: bool = True,
                      process_group=None,
                      bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """Fused-bucket allreduce of a tensor list (in place).  One-shot
    API (no preallocated state) for callers outside the optimizer."""
    world = _world(process_group)
    if world <= 1 or not tensors:
        return
    for group in _bucketize(tensors, bucket_bytes):
        flat = torch.cat([t.reshape(-1) for t in group])
        dist.all_reduce(flat, group=process_group)
        if average:
            flat.div_(world)
        offset = 0
        for t in group:
            t.copy_(flat[offset:offset + t.numel()].view_as(t))
            offset += t.numel()


@torch.no_grad()
def broadcast_parameters(module_or_params, root_rank: int = 0,
                         process_group=None,
                         bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """Rank-0 variable broadcast (the BroadcastGlobalVariables hook/callback
    of the reference examples, ``collective_all_reduce_example.py:69``)."""
    if isinstance(module_or_params, torch.nn.Module):
        tensors = [p.detach() for p in module_or_params.parameters()]
        tensors += [b.detach() for b in module_or_params.buffers()]
    else:
        tensors = [p.detach() for p in module_or_params]
    if _world(process_group) <= 1:
        return
    for group in _bucketize(tensors, bucket_bytes):
        flat = torch.cat([t.reshape(-1) for t in group])
        dist.broadcast(flat, src=root_rank, group=process_group)
        offset = 0
        for t in group:
            t.copy_(flat[offset:offset + t.numel()].view_as(t))
            offset += t.numel()


class _Bucket:
    __slots__ = ("index", "params", "buffer", "views", "pending", "seen",
                 "work")

    def __init__(self, index: int, params: List[torch.nn.Parameter]):
        self.index = index
        self.params = params
        p0 = params[0]
        numel = sum(p.numel() for p in params)
        self.buffer = torch.zeros(numel, dtype=p0.dtype, device=p0.device)
        self.views: List[torch.Tensor] = []
        offset = 0
        for p in params:
            self.views.append(
                self.buffer[offset:offset + p.numel()].view_as(p))
            offset += p.numel()
        self.pending = len(params)
        self.seen = [False] * len(params)
        self.work: Optional[dist.Work] = None


class DistributedOptimizer(Optimizer):
    """Wrap any torch optimizer with bucketed gradient allreduce
    overlapped with backward (the ``hvd.DistributedOptimizer``
    surface).  Gradients become views into preallocated flat buckets;
    each bucket's async allreduce launches (in strict bucket order, the
    RCCL launch-order contract) when its last grad arrives, and
    ``step()`` waits for the in-flight works before applying."""

    def __init__(self, optimizer: Optimizer, process_group=None,
                 average: bool = True,
                 bucket_bytes: int = DEFAULT_BUCKET_BYTES):
        self.optimizer = optimizer
        self.process_group = process_group
        self.average = average
        self.bucket_bytes = bucket_bytes
        # Present the wrapped optimizer's state transparently.
        self.param_groups = optimizer.param_groups
        self.state = optimizer.state
        self.defaults = optimizer.defaults
        self._world_size = _world(process_group)
        self._buckets: List[_Bucket] = []
        self._param_to_bucket = {}
        self._ready: List[bool] = []
        self._next_launch = 0
        self._hook_handles = []
        if self._world_size > 1:
            self._build_buckets()
            self._register_hooks()

    # -- construction ------------------------------------------------------

    def _dense_params(self):
        return [p for g in self.optimizer.param_groups for p in g["params"]
                if p.requires_grad
                and not getattr(p, "_miyarn_sparse", False)]

    def _build_buckets(self) -> None:
        params = list(reversed(self._dense_params()))
        for i, group in enumerate(_bucketize(params, self.bucket_bytes)):
            bucket = _Bucket(i, group)
            self._buckets.append(bucket)
            for slot, p in enumerate(group):
                self._param_to_bucket[p] = (bucket, slot)
                p.grad = bucket.views[slot]
        self._ready = [False] * len(self._buckets)
        logger.info("hvd reducer: %d params in %d buckets",
                    len(params), len(self._buckets))

    def _register_hooks(self) -> None:
        for p in self._param_to_bucket:
            self._hook_handles.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready))

    # -- backward machinery ------------------------------------------------

    def _on_grad_ready(self, param) -> None:
        bucket, slot = self._param_to_bucket[param]
        if param.grad is None:
            return
        if param.grad.data_ptr() != bucket.views[slot].data_ptr():
            bucket.views[slot].copy_(param.grad)
            param.grad = bucket.views[slot]
        if not bucket.seen[slot]:
            bucket.seen[slot] = True
            bucket.pending -= 1
        if bucket.pending == 0:
            self._ready[bucket.index] = True
            while (self._next_launch < len(self._buckets)
                   and self._ready[self._next_launch]):
                self._launch(self._buckets[self._next_launch])
                self._next_launch += 1

    def _launch(self, bucket: _Bucket) -> None:
        op = dist.ReduceOp.AVG if (
            self.average
            and dist.get_backend(self.process_group) == "nccl"
        ) else dist.ReduceOp.SUM
        bucket.work = dist.all_reduce(bucket.buffer, op=op, async_op=True,
                                      group=self.process_group)

    @torch.no_grad()
    def synchronize(self) -> None:
        """Drain: launch any not-yet-launched buckets (zero-filling
        slots whose grads never arrived, matching hvd's skip-None
        semantics) and wait for all in-flight works."""
        if self._world_size <= 1:
            return
        for bucket in self._buckets[self._next_launch:]:
            for slot, p in enumerate(bucket.params):
                if bucket.seen[slot]:
                    continue
                if p.grad is not None and \
                        p.grad.data_ptr() != bucket.views[slot].data_ptr():
                    bucket.views[slot].copy_(p.grad)
                    p.grad = bucket.views[slot]
                elif p.grad is None:
                    bucket.views[slot].zero_()
                    p.grad = bucket.views[slot]
        while self._next_launch < len(self._buckets):
            self._launch(self._buckets[self._next_launch])
            self._next_launch += 1
        scale_needed = (self.average and
                        dist.get_backend(self.process_group) != "nccl")
        for bucket in self._buckets:
            if bucket.work is not None:
                bucket.work.wait()
                bucket.work = None
            if scale_needed:
                bucket.buffer.div_(self._world_size)
            bucket.pending = len(bucket.params)
            bucket.seen = [False] * len(bucket.params)
        self._ready = [False] * len(self._buckets)
        self._next_launch = 0

    def step(self, closure=None):
        self.synchronize()
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = True):
        return self.optimizer.zero_grad(set_to_none=set_to_none)

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        return self.optimizer.load_state_dict(sd)
