"""torch.optim-compatible optimizers over the fused MI355X kernels.

Mixed-precision contract: bf16 parameters keep an fp32 master copy in the
optimizer state; the fused kernel updates the master and writes the bf16
compute copy in the same pass (one HBM round trip).  Params flagged
``_miyarn_sparse`` (embedding tables) are skipped — they sync and update
through the sparse allgather+scatter path.
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch
from torch.optim import Optimizer

from tf_yarn_amd import ops


def _split_param(p: torch.Tensor, state: dict):
    """Return (master_fp32, bf16_copy_or_None) for a param."""
    if p.dtype == torch.bfloat16:
        if "master" not in state:
            state["master"] = p.detach().float().clone()
        return state["master"], p.data
    return p.data, None


def _flat(t: torch.Tensor) -> torch.Tensor:
    """Flat view in STORAGE order (handles channels_last params, whose
    logical view(-1) is not a view of storage)."""
    if t.is_contiguous():
        return t.view(-1)
    return torch.as_strided(t, (t.numel(),), (1,),
                            storage_offset=t.storage_offset())


def _grad_of(p: torch.Tensor) -> Optional[torch.Tensor]:
    """Gradient with the SAME memory layout as the param (elementwise
    kernels walk storage order, so layouts must match)."""
    g = p.grad
    if g is None:
        return None
    if g.stride() == p.data.stride():
        return g
    aligned = torch.empty_like(p.data, dtype=g.dtype)
    aligned.copy_(g)
    return aligned


class FusedSGD(Optimizer):
    def __init__(self, params: Iterable, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False, grad_scale: float = 1.0):
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov,
                        grad_scale=grad_scale)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            # Bucket params by (grad dtype, momentum?, bf16 copy?, first?)
            # so each bucket is one multi-tensor kernel launch.
            buckets = {}
            for p in group["params"]:
                if getattr(p, "_miyarn_sparse", False):
                    continue
                g = _grad_of(p)
                if g is None:
                    continue
                state = self.state[p]
                master, bf16 = _split_param(p, state)
                first = "step" not in state
                state["step"] = state.get("step", 0) + 1
                mom_buf = None
                if group["momentum"] != 0:
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(master)
                    mom_buf = state["momentum_buffer"]
                key = (g.dtype, mom_buf is not None, bf16 is not None,
                       first)
                b = buckets.setdefault(key, ([], [], [], []))
                b[0].append(_flat(master))
                b[1].append(_flat(g))
                if mom_buf is not None:
                    b[2].append(_flat(mom_buf))
                if bf16 is not None:
                    b[3].append(_flat(bf16))
                if bf16 is None and p.dtype != torch.float32:
                    p.data.copy_(master.to(p.dtype))
            for (gdtype, has_mom, has_bf16, first), \
                    (masters, grads, moms, bf16s) in buckets.items():
                ops.fused_sgd_mt(
                    masters, grads,
                    moms if has_mom else None,
                    bf16s if has_bf16 else None,
                    lr=group["lr"], momentum=group["momentum"],
                    dampening=group["dampening"],
                    weight_decay=group["weight_decay"],
                    nesterov=group["nesterov"], first_step=first,
                    grad_scale=group["grad_scale"])
        return loss


class FusedAdam(Optimizer):
    def __init__(self, params: Iterable, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, adamw: bool = False,
                 grad_scale: float = 1.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw,
                        grad_scale=grad_scale)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if getattr(p, "_miyarn_sparse", False):
                    continue
                g = _grad_of(p)
                if g is None:
                    continue
                state = self.state[p]
                master, bf16 = _split_param(p, state)
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(master)
                    state["exp_avg_sq"] = torch.zeros_like(master)
                state["step"] = state.get("step", 0) + 1
                ops.fused_adam(
                    _flat(master), _flat(g),
                    _flat(state["exp_avg"]),
                    _flat(state["exp_avg_sq"]),
                    _flat(bf16) if bf16 is not None else None,
                    lr=group["lr"], beta1=beta1, beta2=beta2,
                    eps=group["eps"], weight_decay=group["weight_decay"],
                    adamw=group["adamw"], step=state["step"],
                    grad_scale=group["grad_scale"])
        return loss


class FusedAdagrad(Optimizer):
    def __init__(self, params: Iterable, lr: float = 1e-2,
                 eps: float = 1e-10, weight_decay: float = 0.0,
                 grad_scale: float = 1.0):
        defaults = dict(lr=lr, eps=eps, weight_decay=weight_decay,
                        grad_scale=grad_scale)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group["params"]:
                if getattr(p, "_miyarn_sparse", False):
                    continue
                g = _grad_of(p)
                if g is None:
                    continue
                state = self.state[p]
                master, bf16 = _split_param(p, state)
                if "sum" not in state:
                    state["sum"] = torch.zeros_like(master)
                ops.fused_adagrad(
                    _flat(master), _flat(g), _flat(state["sum"]),
                    lr=group["lr"], eps=group["eps"],
                    weight_decay=group["weight_decay"],
                    grad_scale=group["grad_scale"])
                if bf16 is not None:
                    bf16.copy_(master.to(torch.bfloat16))
        return loss


class FusedAdadelta(Optimizer):
    """Adadelta — the README's Keras example optimizer
    (reference README.md:106)."""

    def __init__(self, params: Iterable, lr: float = 1.0, rho: float = 0.9,
                 eps: float = 1e-6, weight_decay: float = 0.0,
                 grad_scale: float = 1.0):
        defaults = dict(lr=lr, rho=rho, eps=eps,
                        weight_decay=weight_decay, grad_scale=grad_scale)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group["params"]:
                if getattr(p, "_miyarn_sparse", False):
                    continue
                g = _grad_of(p)
                if g is None:
                    continue
                state = self.state[p]
                master, bf16 = _split_param(p, state)
                if "square_avg" not in state:
                    state["square_avg"] = torch.zeros_like(master)
                    state["acc_delta"] = torch.zeros_like(master)
                ops.fused_adadelta(
                    _flat(master), _flat(g),
                    _flat(state["square_avg"]),
                    _flat(state["acc_delta"]),
                    lr=group["lr"], rho=group["rho"], eps=group["eps"],
                    weight_decay=group["weight_decay"],
                    grad_scale=group["grad_scale"])
                if bf16 is not None:
                    bf16.copy_(master.to(torch.bfloat16))
        return loss
