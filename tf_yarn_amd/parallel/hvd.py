"""Horovod-style optimizer-level allreduce on RCCL over xGMI.

Re-implements the surface the reference wires through Horovod-gloo
(``hvd.DistributedOptimizer``,
``BroadcastGlobalVariablesHook/Callback(0)``, ``gloo_allred_task.py``):
gradients are fused into flat buckets and averaged with RCCL ring
allreduce at ``step()`` time (Horovod-gloo reduced per-tensor, unfused —
bucketing is the xGMI-native upgrade), and rank-0 weight broadcast uses the
same bucket buffers.
"""

from __future__ import annotations

import logging
from typing import List

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
    """Fused-bucket allreduce of a tensor list (in place)."""
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


class DistributedOptimizer(Optimizer):
    """Wrap any torch optimizer with a pre-step fused-bucket gradient
    allreduce (the ``hvd.DistributedOptimizer`` surface)."""

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

    def synchronize(self) -> None:
        grads = [p.grad for g in self.optimizer.param_groups
                 for p in g["params"]
                 if p.grad is not None
                 and not getattr(p, "_miyarn_sparse", False)]
        allreduce_tensors(grads, average=self.average,
                          process_group=self.process_group,
                          bucket_bytes=self.bucket_bytes)

    def step(self, closure=None):
        self.synchronize()
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = True):
        return self.optimizer.zero_grad(set_to_none=set_to_none)

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        return self.optimizer.load_state_dict(sd)
