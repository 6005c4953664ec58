"""MI355X HIP kernel library: Python wrappers + CPU references.

Policy (required for the GPU-native check): on a machine with a GPU the
HIP extension MUST be present — ops raise immediately if it failed to
import, instead of silently falling back to eager torch.  On CPU-only
machines (the CI container) the same functions run reference torch
implementations so every numerics test has a CPU baseline.
"""

from __future__ import annotations

import logging
from typing import Optional

import torch

logger = logging.getLogger(__name__)

try:
    from tf_yarn_amd.ops import _C  # built by setup.py build_ext --inplace
    HAVE_EXT = True
except ImportError as _e:  # pragma: no cover - exercised only sans build
    _C = None
    HAVE_EXT = False
    _IMPORT_ERROR = _e


def _require_ext() -> None:
    if not HAVE_EXT:
        raise RuntimeError(
            "tf_yarn_amd.ops._C is not built but a GPU tensor was passed. "
            "Run `python setup.py build_ext --inplace` "
            f"(import error: {_IMPORT_ERROR})")


def _on_gpu(*tensors: torch.Tensor) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---- fused optimizers ------------------------------------------------------

def fused_sgd(param: torch.Tensor, grad: torch.Tensor,
              momentum_buf: Optional[torch.Tensor] = None,
              param_bf16: Optional[torch.Tensor] = None,
              *, lr: float, momentum: float = 0.0, dampening: float = 0.0,
              weight_decay: float = 0.0, nesterov: bool = False,
              first_step: bool = False, grad_scale: float = 1.0) -> None:
    if _on_gpu(param, grad):
        _require_ext()
        _C.fused_sgd(param, grad, momentum_buf, param_bf16, lr, momentum,
                     dampening, weight_decay, nesterov, first_step,
                     grad_scale)
        return
    g = grad.float() * grad_scale
    g = g.add(param, alpha=weight_decay)
    if momentum_buf is not None:
        if first_step:
            momentum_buf.copy_(g)
        else:
            momentum_buf.mul_(momentum).add_(g, alpha=1.0 - dampening)
        g = g.add(momentum_buf, alpha=momentum) if nesterov \
            else momentum_buf.clone()
    param.add_(g, alpha=-lr)
    if param_bf16 is not None:
        param_bf16.copy_(param.to(torch.bfloat16))


def fused_sgd_mt(params, grads, momentum_bufs=None, params_bf16=None,
                 *, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False, first_step: bool = False,
                 grad_scale: float = 1.0) -> None:
    """Multi-tensor SGD: one kernel launch per <=24 tensors."""
    if params and params[0].is_cuda:
        _require_ext()
        _C.fused_sgd_mt(list(params), list(grads),
                        list(momentum_bufs) if momentum_bufs else [],
                        list(params_bf16) if params_bf16 else [],
                        lr, momentum, dampening, weight_decay, nesterov,
                        first_step, grad_scale)
        return
    for i, (p, g) in enumerate(zip(params, grads)):
        fused_sgd(p, g,
                  momentum_bufs[i] if momentum_bufs else None,
                  params_bf16[i] if params_bf16 else None,
                  lr=lr, momentum=momentum, dampening=dampening,
                  weight_decay=weight_decay, nesterov=nesterov,
                  first_step=first_step, grad_scale=grad_scale)


def fused_adam(param: torch.Tensor, grad: torch.Tensor,
               exp_avg: torch.Tensor, exp_avg_sq: torch.Tensor,
               param_bf16: Optional[torch.Tensor] = None,
               *, lr: float, beta1: float = 0.9, beta2: float = 0.999,
               eps: float = 1e-8, weight_decay: float = 0.0,
               adamw: bool = False, step: int = 1,
               grad_scale: float = 1.0) -> None:
    if _on_gpu(param, grad):
        _require_ext()
        _C.fused_adam(param, grad, exp_avg, exp_avg_sq, param_bf16, lr,
                      beta1, beta2, eps, weight_decay, adamw, step,
                      grad_scale)
        return
    g = grad.float() * grad_scale
    if adamw:
        param.mul_(1.0 - lr * weight_decay)
    else:
        g = g.add(param, alpha=weight_decay)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param.addcdiv_(exp_avg / bc1, denom, value=-lr)
    if param_bf16 is not None:
        param_bf16.copy_(param.to(torch.bfloat16))


def fused_adagrad(param: torch.Tensor, grad: torch.Tensor,
                  state_sum: torch.Tensor, *, lr: float, eps: float = 1e-10,
                  weight_decay: float = 0.0,
                  grad_scale: float = 1.0) -> None:
    if _on_gpu(param, grad):
        _require_ext()
        _C.fused_adagrad(param, grad, state_sum, lr, eps, weight_decay,
                         grad_scale)
        return
    g = grad.float() * grad_scale
    g = g.add(param, alpha=weight_decay)
    state_sum.addcmul_(g, g, value=1.0)
    param.addcdiv_(g, state_sum.sqrt().add_(eps), value=-lr)


def fused_adadelta(param: torch.Tensor, grad: torch.Tensor,
                   square_avg: torch.Tensor, acc_delta: torch.Tensor,
                   *, lr: float = 1.0, rho: float = 0.9, eps: float = 1e-6,
                   weight_decay: float = 0.0,
                   grad_scale: float = 1.0) -> None:
    if _on_gpu(param, grad):
        _require_ext()
        _C.fused_adadelta(param, grad, square_avg, acc_delta, lr, rho, eps,
                          weight_decay, grad_scale)
        return
    g = grad.float() * grad_scale
    g = g.add(param, alpha=weight_decay)
    square_avg.mul_(rho).addcmul_(g, g, value=1 - rho)
    dx = (acc_delta + eps).sqrt_().div_((square_avg + eps).sqrt()).mul_(g)
    acc_delta.mul_(rho).addcmul_(dx, dx, value=1 - rho)
    param.add_(dx, alpha=-lr)


# ---- embedding -------------------------------------------------------------

def emb_fwd(table: torch.Tensor, ids: torch.Tensor,
            out_bf16: bool = False) -> torch.Tensor:
    """Row gather: out[i, :] = table[ids[i], :] (ids pre-offset, flat)."""
    if _on_gpu(table, ids):
        _require_ext()
        return _C.emb_fwd(table, ids, out_bf16)
    out = table.index_select(0, ids.reshape(-1))
    return out.to(torch.bfloat16) if out_bf16 else out


def emb_bwd_sgd(table: torch.Tensor, ids: torch.Tensor, grad: torch.Tensor,
                *, lr: float, scale: float = 1.0) -> None:
    """table[ids[i], :] -= lr * scale * grad[i, :] (fused sparse update).

    The atomic scatter is the default: the atomic-free sorted kernel
    (`_C.emb_bwd_sgd_sorted`) is 4.6x faster per-kernel, but paying
    torch.sort + a 54 MB gather every step measured NET-slower at the
    bench shape (bench 2.30 vs 1.86 ms/step) — use it only when ids
    arrive pre-sorted."""
    if _on_gpu(table, ids, grad):
        _require_ext()
        _C.emb_bwd_sgd(table, ids.reshape(-1).contiguous(),
                       grad.contiguous(), lr, scale)
        return
    table.index_add_(0, ids.reshape(-1),
                     grad.reshape(ids.numel(), -1).float(),
                     alpha=-lr * scale)


def pick_region_bits(n_rows: int, n_updates: int) -> int:
    """Region size heuristic for the binned scatter: aim for an expected
    per-bin update count around half the LDS hash capacity (1024 for deep
    tables) so dedup almost never overflows into the atomic fallback,
    while keeping enough bins (>2048) to fill all 256 CUs."""
    import math
    if n_updates <= 0:
        return 11
    target = max(1.0, 512.0 * n_rows / n_updates)
    bits = int(math.log2(target))
    return max(7, min(14, bits))


def binned_permutation(ids: torch.Tensor, n_rows: int,
                       region_bits: int):
    """Pass A of the binned scatter (GPU): permutation of update indices
    grouped by 2^region_bits-row table regions + bin start offsets.
    Algorithm spec: tests/test_binned_scatter_spec.py."""
    _require_ext()
    return _C.binned_permutation(ids.reshape(-1).contiguous(), n_rows,
                                 region_bits)


def emb_bwd_sgd_binned(table: torch.Tensor, ids: torch.Tensor,
                       grad: torch.Tensor, *, lr: float,
                       scale: float = 1.0,
                       perm=None, region_bits: int = None) -> None:
    """Binned replacement for :func:`emb_bwd_sgd` (dim-16 tables): LDS
    hash dedup per region + exclusive-owner writeback, no global atomics
    on the hot path.  ``perm`` = (order, starts) from
    :func:`binned_permutation` can be shared between the deep and wide
    tables (same flat ids).  CPU reference: plain index_add_."""
    if not _on_gpu(table, ids, grad):
        table.index_add_(0, ids.reshape(-1),
                         grad.reshape(ids.numel(), -1).float(),
                         alpha=-lr * scale)
        return
    _require_ext()
    ids = ids.reshape(-1).contiguous()
    if perm is None:
        if region_bits is None:
            region_bits = pick_region_bits(table.shape[0], ids.numel())
        perm = _C.binned_permutation(ids, table.shape[0], region_bits)
    order, starts = perm
    _C.emb_bwd_sgd_binned(table, ids, grad.contiguous(), lr, scale,
                          order, starts)


def emb_scatter_sum_binned(table: torch.Tensor, ids: torch.Tensor,
                           grad: torch.Tensor, alpha: float,
                           perm=None, region_bits: int = None) -> None:
    """Binned replacement for :func:`emb_scatter_sum` (scalar wide
    tables): grad for flat update j is ``grad[j // F]``."""
    n = ids.numel()
    g_div = n // grad.numel()
    if not _on_gpu(table, ids, grad):
        expanded = grad.float().reshape(-1, 1).expand(-1, g_div) \
            .reshape(-1)
        table.reshape(-1).index_add_(0, ids.reshape(-1), expanded,
                                     alpha=alpha)
        return
    _require_ext()
    ids = ids.reshape(-1).contiguous()
    if perm is None:
        if region_bits is None:
            region_bits = pick_region_bits(table.shape[0], n)
        perm = _C.binned_permutation(ids, table.shape[0], region_bits)
    order, starts = perm
    _C.emb_scatter_sum_binned(table, ids, grad.contiguous(), g_div,
                              alpha, order, starts)


def emb_bwd_sgd_fused_wide(table: torch.Tensor, wide_table: torch.Tensor,
                           ids: torch.Tensor, grad: torch.Tensor,
                           gw: torch.Tensor, *, lr: float,
                           scale: float = 1.0) -> None:
    """Fused sparse update of BOTH CTR tables from one pass over the
    shared flat ids:  ``table[ids[i]] -= lr*scale*grad[i]`` and
    ``wide_table[ids[i]] -= lr*scale*gw[i // (n//gw.numel())]``."""
    if _on_gpu(table, ids, grad) and grad.dtype == gw.dtype:
        _require_ext()
        _C.emb_bwd_sgd_fused_wide(table, wide_table,
                                  ids.reshape(-1).contiguous(),
                                  grad.contiguous(), gw.contiguous(),
                                  lr, scale)
        return
    n = ids.numel()
    g_div = n // gw.numel()
    table.index_add_(0, ids.reshape(-1),
                     grad.reshape(n, -1).float(), alpha=-lr * scale)
    expanded = gw.float().reshape(-1, 1).expand(-1, g_div).reshape(-1)
    wide_table.reshape(-1).index_add_(0, ids.reshape(-1), expanded,
                                      alpha=-lr * scale)


def emb_bwd_dense(grad_table: torch.Tensor, ids: torch.Tensor,
                  grad: torch.Tensor, scale: float = 1.0) -> None:
    if _on_gpu(grad_table, ids, grad):
        _require_ext()
        _C.emb_bwd_dense(grad_table, ids, grad.contiguous(), scale)
        return
    grad_table.index_add_(0, ids.reshape(-1),
                          grad.reshape(ids.numel(), -1).float(),
                          alpha=scale)


# ---- elementwise -----------------------------------------------------------

def bias_relu_fwd(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    if _on_gpu(x, bias):
        _require_ext()
        return _C.bias_relu_fwd(x.contiguous(), bias.contiguous())
    return torch.relu(x + bias)


def bias_relu_bwd(dy: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    if _on_gpu(dy, y):
        _require_ext()
        return _C.bias_relu_bwd(dy.contiguous(), y.contiguous())
    return dy * (y > 0).to(dy.dtype)


def convert_scaled(src: torch.Tensor, dst: torch.Tensor,
                   scale: float = 1.0) -> None:
    if _on_gpu(src, dst):
        _require_ext()
        _C.convert_scaled(src, dst, scale)
        return
    dst.copy_((src.float() * scale).to(dst.dtype))


def emb_fwd_into(table: torch.Tensor, ids: torch.Tensor,
                 out: torch.Tensor, col_offset: int) -> None:
    """Gather row (b,f) into out[b, col_offset + f*dim : +dim]."""
    if _on_gpu(table, ids, out):
        _require_ext()
        _C.emb_fwd_into(table, ids.reshape(-1).contiguous(), out,
                        col_offset)
        return
    batch = out.shape[0]
    f = ids.numel() // batch
    dim = table.shape[1]
    gathered = table.index_select(0, ids.reshape(-1)).reshape(
        batch, f * dim)
    out[:, col_offset:col_offset + f * dim] = gathered.to(out.dtype)


def emb_gather_sum(table: torch.Tensor, ids: torch.Tensor,
                   out_bf16: bool = False) -> torch.Tensor:
    """out[b] = sum_f table[ids[b, f], 0] for a scalar (dim-1) table."""
    batch = ids.shape[0]
    if _on_gpu(table, ids):
        _require_ext()
        return _C.emb_gather_sum(table, ids.reshape(-1).contiguous(),
                                 batch, out_bf16)
    out = table.reshape(-1).index_select(
        0, ids.reshape(-1)).reshape(batch, -1).sum(dim=1)
    return out.to(torch.bfloat16) if out_bf16 else out


def emb_scatter_sum(table: torch.Tensor, ids: torch.Tensor,
                    grad: torch.Tensor, alpha: float) -> None:
    """table[ids[b, f], 0] += alpha * grad[b] (gather-sum backward)."""
    if _on_gpu(table, ids, grad):
        _require_ext()
        _C.emb_scatter_sum(table, ids.reshape(-1).contiguous(),
                           grad.contiguous(), alpha)
        return
    f = ids.reshape(grad.numel(), -1).shape[1]
    expanded = grad.float().reshape(-1, 1).expand(-1, f).reshape(-1)
    table.reshape(-1).index_add_(0, ids.reshape(-1), expanded,
                                 alpha=alpha)


def col_reduce_dot(x: torch.Tensor, dy: torch.Tensor) -> torch.Tensor:
    """dw[m] = sum_b dy[b] * x[b, m] (fp32 result)."""
    if _on_gpu(x, dy):
        _require_ext()
        return _C.col_reduce_dot(x.contiguous(), dy.contiguous())
    return (x.float() * dy.float().unsqueeze(1)).sum(dim=0)


def row_dot(x: torch.Tensor, w: torch.Tensor,
            bias: "torch.Tensor | None" = None) -> torch.Tensor:
    """y[b] = x[b] . w (+ bias): streaming GEMV for the single-logit
    head forward (hipBLASLt's M=1 bias-GEMV ran at ~175 GB/s)."""
    if _on_gpu(x, w) and x.size(1) % 4 == 0 and x.size(1) <= 512:
        _require_ext()
        return _C.row_dot(x.contiguous(), w.contiguous(), bias)
    y = x @ w
    return y if bias is None else y + bias


def gemm_bt(a: torch.Tensor, b: torch.Tensor,
            bias: "torch.Tensor | None" = None,
            relu: bool = False) -> torch.Tensor:
    """C[M, N] = A[M, K] @ B[N, K]^T (nn.Linear forward form), optional
    fused bias+ReLU epilogue. Both operands K-major bf16 on GPU."""
    if _on_gpu(a, b):
        _require_ext()
        return _C.gemm_bt(a.contiguous(), b.contiguous(), bias, relu)
    out = a.float() @ b.float().t()
    if bias is not None:
        out = out + bias.float()
    if relu:
        out = torch.relu(out)
    return out.to(a.dtype)


class ScalarHeadFn(torch.autograd.Function):
    """y[b] = x[b].w + bias for a single-logit head; the wgrad runs as a
    streaming column reduction instead of hipBLASLt's M=1 GEMM
    (187 us -> ~15 us measured at 65536x256)."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        if _on_gpu(x, weight) and x.size(1) % 4 == 0 \
                and x.size(1) <= 512 and bias.numel() == 1:
            _require_ext()
            return _C.row_dot(x.contiguous(), weight.contiguous(),
                              bias.reshape(1))
        return x @ weight + bias

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:  # input data (wide head) needs no dx
            dx = dy.unsqueeze(1) * weight.unsqueeze(0)
        if ctx.needs_input_grad[1]:
            dw = col_reduce_dot(x, dy).to(weight.dtype)
        if ctx.needs_input_grad[2]:
            db = dy.sum().reshape(1).to(weight.dtype)
        return dx, dw, db


class BCEHeadFn(torch.autograd.Function):
    """Per-example BCEWithLogits over logits = deep + wide + dhead,
    fused into one kernel pair (saves the two adds, the fp32 cast, and
    torch's separate loss fwd/bwd elementwise kernels)."""

    @staticmethod
    def forward(ctx, deep, wide, dhead, labels):
        if deep.is_cuda and HAVE_EXT and deep.dtype == torch.bfloat16 \
                and wide.dtype == torch.bfloat16 \
                and dhead.dtype == torch.bfloat16:
            loss, sig = _C.bce_head_fwd(
                deep.contiguous(), wide.contiguous(), dhead.contiguous(),
                labels.contiguous().float())
        else:
            z = (deep.float() + wide.float() + dhead.float())
            labels = labels.float()
            loss = torch.nn.functional.binary_cross_entropy_with_logits(
                z, labels, reduction="none")
            sig = torch.sigmoid(z).to(deep.dtype)
        ctx.save_for_backward(sig, labels)
        ctx.dtype = deep.dtype
        return loss

    @staticmethod
    def backward(ctx, g):
        sig, labels = ctx.saved_tensors
        if sig.is_cuda and HAVE_EXT and sig.dtype == torch.bfloat16:
            dl = _C.bce_head_bwd(sig, labels.contiguous().float(),
                                 g.contiguous().float())
        else:
            dl = ((sig.float() - labels.float()) * g.float()).to(ctx.dtype)
        return dl, dl, dl, None


def bce_head_loss(deep: torch.Tensor, wide: torch.Tensor,
                  dhead: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean BCEWithLogits over the fused 3-part head."""
    return BCEHeadFn.apply(deep, wide, dhead, labels).mean()


class BiasReLU(torch.autograd.Function):
    """Autograd wrapper for the fused bias+ReLU epilogue (backward fuses
    the dbias reduction into the dx kernel on GPU)."""

    @staticmethod
    def forward(ctx, x, bias):
        y = bias_relu_fwd(x, bias)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if dy.is_cuda and HAVE_EXT and y.size(-1) % 4 == 0 \
                and y.size(-1) <= 8192:
            # v3 fused kernel: atomic-free per-block column partials +
            # one tiny reduce (earlier atomic/LDS variants measured
            # slower than the separate path; see micro_elementwise.py).
            dx, dbias = _C.bias_relu_bwd_db(
                dy.contiguous(), y.contiguous(),
                dy.dtype == torch.bfloat16)
            return dx, dbias.to(dy.dtype)
        dx = bias_relu_bwd(dy.contiguous(), y)
        dims = tuple(range(dx.dim() - 1))
        dbias = dx.sum(dim=dims)
        return dx, dbias


def bias_relu(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    return BiasReLU.apply(x, bias)


def _custom_wgrad_kernel(dz: torch.Tensor, x: torch.Tensor):
    """Pick the split-K MFMA wgrad variant for this shape, or None to use
    hipBLASLt.  Measured (micro_gemm.py, B=65536): 256^2 tiles reach
    312/359 TF on the 1024x432 / 512x1024 layers (hipBLASLt in-context ran
    them at ~207 us each); 128^2 tiles win the small 256x512 layer
    (94 us vs ~142-190)."""
    import os
    if os.environ.get("MIYARN_WGRAD") == "lib":
        return None  # A/B escape hatch: force hipBLASLt
    if not (dz.is_cuda and HAVE_EXT):
        return None
    if dz.dtype != torch.bfloat16 or x.dtype != torch.bfloat16:
        return None
    n, m = dz.size(1), x.size(1)
    if m % 8 != 0 or dz.size(0) % 64 != 0:
        return None
    if n % 256 == 0 and n * m > 256 * 512:
        return _C.wgrad_nt256
    if n % 128 == 0 and n * m <= 256 * 512:
        return _C.wgrad_nt128
    return None


def _custom_fwd_ok(x: torch.Tensor, weight: torch.Tensor,
                   bias: torch.Tensor) -> bool:
    """True when gemm_bt should fuse the whole forward (GEMM + bias +
    ReLU in one kernel).  OPT-IN (MIYARN_FWD=custom): measured on MI355X
    the fused kernel runs at 361-613 TF on the MLP forward shapes while
    hipBLASLt + the separate bias_relu kernel reach 444-831 TF combined —
    with only 7-16 K-tile iterations the software pipeline never hits the
    steady state the wgrad kernels reach over B=65536 (scripts/
    micro_gemm.py --fwd)."""
    import os
    if os.environ.get("MIYARN_FWD") != "custom":
        return False
    if not (x.is_cuda and HAVE_EXT):
        return False
    if x.dtype != torch.bfloat16 or weight.dtype != torch.bfloat16 \
            or bias.dtype != torch.bfloat16:
        return False
    if x.dim() != 2 or not x.is_contiguous() or not weight.is_contiguous():
        return False
    m, k = x.shape
    return m % 256 == 0 and k % 16 == 0 and weight.size(0) % 16 == 0


def _lt_fwd_ok(x, weight, bias) -> bool:
    """Fuse bias+ReLU into the hipBLASLt GEMM epilogue (lt_linear): one
    library call, no z round trip (the separate epilogue kernel cost
    81 us/step + 470 MB of z traffic at the bench shape).
    MIYARN_LT_FWD=0 restores the torch-matmul + epilogue-kernel path."""
    import os
    if os.environ.get("MIYARN_LT_FWD", "1") in ("", "0"):
        return False
    return (x.is_cuda and HAVE_EXT and x.dim() == 2
            and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and bias.dtype == torch.bfloat16
            and x.is_contiguous() and weight.is_contiguous())


class LinearBiasReLU(torch.autograd.Function):
    """y = relu(x @ w.T + bias) with a fully-controlled backward:
    fused dx+dbias kernel, dgrad via hipBLASLt, wgrad via the custom
    split-K MFMA kernel on shapes where it wins."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        if _custom_fwd_ok(x, weight, bias):
            y = _C.gemm_bt(x, weight, bias, True)
        elif _lt_fwd_ok(x, weight, bias):
            y = _C.lt_linear(x, weight, bias.contiguous(), True)
        else:
            z = x.matmul(weight.t())
            y = bias_relu_fwd(z, bias)
        ctx.save_for_backward(x, weight, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, y = ctx.saved_tensors
        dy = dy.contiguous()
        if dy.is_cuda and HAVE_EXT and y.size(-1) % 4 == 0 \
                and y.size(-1) <= 8192:
            dz, dbias = _C.bias_relu_bwd_db(
                dy, y.contiguous(), dy.dtype == torch.bfloat16)
            if dbias.dtype != dy.dtype:
                dbias = dbias.to(dy.dtype)
        else:
            dz = bias_relu_bwd(dy, y)
            dbias = dz.sum(dim=tuple(range(dz.dim() - 1)))
        dx = dz.matmul(weight)
        kernel = _custom_wgrad_kernel(dz, x)
        if kernel is not None:
            # split-K partial sum + cast fused into one reduce kernel
            dw = kernel(dz, x, 0, weight.dtype == torch.bfloat16)
            if dw.dtype != weight.dtype:
                dw = dw.to(weight.dtype)
        else:
            dw = dz.t().matmul(x)
        return dx, dw, dbias


def linear_bias_relu(x: torch.Tensor, weight: torch.Tensor,
                     bias: torch.Tensor) -> torch.Tensor:
    return LinearBiasReLU.apply(x, weight, bias)
