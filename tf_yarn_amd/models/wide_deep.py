"""Criteo wide-and-deep model, MI355X-first.

The reference's target workload family (CTR models trained through the
Keras/Estimator paths; SURVEY §5 "target workloads are CTR/tabular").
Design decisions for MI355X:

* All 26 categorical tables are fused into ONE [total_rows, D] fp32 buffer;
  Python pre-adds per-feature row offsets so the gather/scatter kernels see
  flat ids (:mod:`tf_yarn_amd.ops` ``emb_fwd`` / ``emb_bwd_sgd``).
* The embedding gradient never materializes as a dense table: backward
  stashes (ids, grad_rows); at step time the rows are allgathered across
  data-parallel ranks (equal batch sizes => equal counts, so
  ``all_gather_into_tensor`` over RCCL) and applied with the fused
  atomic scatter+SGD kernel scaled by 1/world.  This replaces the dense
  allreduce that would move the whole table every step.
* Dense compute runs bf16 (``emb_fwd`` fuses the fp32->bf16 cast into the
  gather); dense params keep fp32 master weights.
* The MLP epilogue uses the fused bias+ReLU kernel.
"""

from __future__ import annotations

import logging
import math
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist
from torch import nn

from tf_yarn_amd import ops

logger = logging.getLogger(__name__)

# Criteo display-advertising schema: 13 integer (dense) + 26 categorical.
CRITEO_DENSE = 13
CRITEO_SPARSE = 26
# Per-feature hash sizes for the synthetic Criteo-1TB-shaped benchmark
# (hashed to 1M rows max per feature, a common Criteo preprocessing).
DEFAULT_TABLE_SIZES = [1_000_000] * CRITEO_SPARSE


class _FusedLookup(torch.autograd.Function):
    """Gather via the HIP kernel; backward stashes sparse (ids, grad)."""

    @staticmethod
    def forward(ctx, table, flat_ids, out_bf16, sink):
        out = ops.emb_fwd(table, flat_ids, out_bf16)
        ctx.sink = sink
        ctx.save_for_backward(flat_ids)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (flat_ids,) = ctx.saved_tensors
        ctx.sink.append((flat_ids, grad_out.contiguous()))
        return None, None, None, None


class SparseEmbedding(nn.Module):
    """Fused multi-table embedding with sparse-allgather DP sync.

    ``forward(ids [B, F])`` -> ``[B, F*dim]``.
    After ``loss.backward()`` call :meth:`apply_sparse_updates` (the
    training loop or the framework optimizer wrapper does this).
    """

    def __init__(self, table_sizes: List[int], dim: int,
                 out_bf16: bool = False, init_std: Optional[float] = None):
        super().__init__()
        self.table_sizes = list(table_sizes)
        self.dim = dim
        self.out_bf16 = out_bf16
        total = sum(table_sizes)
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(table_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("offsets", offsets, persistent=True)
        std = init_std if init_std is not None else 1.0 / math.sqrt(dim)
        weight = torch.randn(total, dim) * std
        self.weight = nn.Parameter(weight)
        self.weight._miyarn_sparse = True  # dense reducer must skip it
        self._sink: List[Tuple[torch.Tensor, torch.Tensor]] = []

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        b, f = ids.shape
        assert f == len(self.table_sizes), "feature count mismatch"
        flat = (ids + self.offsets.unsqueeze(0)).reshape(-1)
        out = _FusedLookup.apply(self.weight, flat, self.out_bf16,
                                 self._sink)
        return out.reshape(b, f * self.dim)

    def pending_grads(self) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        return self._sink

    @torch.no_grad()
    def start_sparse_sync(self, process_group=None) -> None:
        """Launch the cross-rank allgather of (ids, grad rows)
        asynchronously (overlaps with the dense optimizer step)."""
        world = (dist.get_world_size(process_group)
                 if dist.is_available() and dist.is_initialized() else 1)
        self._pending_world = world
        self._gathered: List[Tuple[torch.Tensor, torch.Tensor, list]] = []
        for flat_ids, grad in self._sink:
            grad2d = grad.reshape(flat_ids.numel(), self.dim)
            if world > 1:
                n = flat_ids.numel()
                all_ids = torch.empty(n * world, dtype=flat_ids.dtype,
                                      device=flat_ids.device)
                all_grads = torch.empty((n * world, self.dim),
                                        dtype=grad2d.dtype,
                                        device=grad2d.device)
                works = [
                    dist.all_gather_into_tensor(all_ids, flat_ids,
                                                group=process_group,
                                                async_op=True),
                    dist.all_gather_into_tensor(all_grads, grad2d,
                                                group=process_group,
                                                async_op=True),
                ]
                self._gathered.append((all_ids, all_grads, works))
            else:
                self._gathered.append((flat_ids, grad2d, []))
        self._sink.clear()

    @torch.no_grad()
    def finish_sparse_sync(self, lr: float) -> None:
        """Wait for the allgathers and apply the fused scatter+SGD update
        scaled by 1/world_size."""
        world = getattr(self, "_pending_world", 1)
        for all_ids, all_grads, works in getattr(self, "_gathered", []):
            for w in works:
                w.wait()
            ops.emb_bwd_sgd(self.weight.data, all_ids, all_grads,
                            lr=lr, scale=1.0 / world)
        self._gathered = []

    @torch.no_grad()
    def apply_sparse_updates(self, lr: float,
                             process_group=None) -> None:
        """Allgather sparse grads across DP ranks and apply the fused
        scatter+SGD update, scaled by 1/world_size."""
        if not self._sink:
            return
        self.start_sparse_sync(process_group)
        self.finish_sparse_sync(lr)

    def clear_pending(self) -> None:
        self._sink.clear()
        self._gathered = []


class _GatherSum(torch.autograd.Function):
    """Wide-part fused gather-sum: out[b] = sum_f table[ids[b,f]]."""

    @staticmethod
    def forward(ctx, table, flat_ids, batch, out_bf16, sink):
        out = ops.emb_gather_sum(
            table, flat_ids.reshape(batch, -1), out_bf16)
        ctx.sink = sink
        ctx.save_for_backward(flat_ids)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (flat_ids,) = ctx.saved_tensors
        ctx.sink.append((flat_ids, grad_out.contiguous()))
        return None, None, None, None, None


class WideScalarEmbedding(nn.Module):
    """Per-id scalar weights with fused gather-sum forward and
    scatter-add sparse update (the wide part of wide-and-deep)."""

    def __init__(self, table_sizes: List[int], out_bf16: bool = False,
                 init_std: float = 0.01):
        super().__init__()
        self.table_sizes = list(table_sizes)
        self.out_bf16 = out_bf16
        total = sum(table_sizes)
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(table_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("offsets", offsets, persistent=True)
        self.weight = nn.Parameter(torch.randn(total, 1) * init_std)
        self.weight._miyarn_sparse = True
        self._sink: List[Tuple[torch.Tensor, torch.Tensor]] = []

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        b = ids.shape[0]
        flat = (ids + self.offsets.unsqueeze(0)).reshape(-1)
        return _GatherSum.apply(self.weight, flat, b, self.out_bf16,
                                self._sink)

    def pending_grads(self):
        return self._sink

    @torch.no_grad()
    def start_sparse_sync(self, process_group=None) -> None:
        world = (dist.get_world_size(process_group)
                 if dist.is_available() and dist.is_initialized() else 1)
        self._pending_world = world
        self._gathered = []
        for flat_ids, grad in self._sink:
            if world > 1:
                n = flat_ids.numel()
                b = grad.numel()
                all_ids = torch.empty(n * world, dtype=flat_ids.dtype,
                                      device=flat_ids.device)
                all_grads = torch.empty(b * world, dtype=grad.dtype,
                                        device=grad.device)
                works = [
                    dist.all_gather_into_tensor(all_ids, flat_ids,
                                                group=process_group,
                                                async_op=True),
                    dist.all_gather_into_tensor(all_grads,
                                                grad.reshape(-1),
                                                group=process_group,
                                                async_op=True),
                ]
                self._gathered.append((all_ids, all_grads, works))
            else:
                self._gathered.append((flat_ids, grad.reshape(-1), []))
        self._sink.clear()

    @torch.no_grad()
    def finish_sparse_sync(self, lr: float) -> None:
        world = getattr(self, "_pending_world", 1)
        for all_ids, all_grads, works in getattr(self, "_gathered", []):
            for w in works:
                w.wait()
            ops.emb_scatter_sum(self.weight.data,
                                all_ids.reshape(all_grads.numel(), -1),
                                all_grads, alpha=-lr / world)
        self._gathered = []

    @torch.no_grad()
    def apply_sparse_updates(self, lr: float, process_group=None) -> None:
        if not self._sink:
            return
        self.start_sparse_sync(process_group)
        self.finish_sparse_sync(lr)

    def clear_pending(self) -> None:
        self._sink.clear()
        self._gathered = []


class _DeepInput(torch.autograd.Function):
    """Build the MLP input [B, dense_pad + F*dim] in ONE buffer: dense
    features go to columns [0:13] (padded to 16 for quad alignment) and the
    embedding gather writes straight into the rest (no concat kernel).
    Backward stashes the embedding slice of the grad as the sparse sink."""

    DENSE_PAD = 16

    @staticmethod
    def forward(ctx, dense, table, flat_ids, dim, out_bf16, sink):
        b = dense.shape[0]
        f = flat_ids.numel() // b
        dtype = torch.bfloat16 if out_bf16 else torch.float32
        out = torch.empty(b, _DeepInput.DENSE_PAD + f * dim,
                          dtype=dtype, device=dense.device)
        out[:, :dense.shape[1]] = dense.to(dtype)
        out[:, dense.shape[1]:_DeepInput.DENSE_PAD] = 0
        ops.emb_fwd_into(table, flat_ids, out, _DeepInput.DENSE_PAD)
        ctx.sink = sink
        ctx.save_for_backward(flat_ids)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (flat_ids,) = ctx.saved_tensors
        emb_grad = grad_out[:, _DeepInput.DENSE_PAD:].contiguous()
        ctx.sink.append((flat_ids, emb_grad))
        return None, None, None, None, None, None


class ScalarHead(nn.Module):
    """Single-logit head as a GEMV: ``logits[b] = x[b] . w + bias``.

    An nn.Linear(in, 1) makes autograd emit an M=1, K=batch wgrad GEMM
    that hipBLASLt runs terribly (140 us measured for 33 MFLOP); the mv
    form lowers to GEMV/broadcast kernels instead."""

    def __init__(self, in_features: int, dtype=torch.float32):
        super().__init__()
        bound = 1.0 / math.sqrt(in_features)
        self.weight = nn.Parameter(
            (torch.rand(in_features, dtype=torch.float32) * 2 - 1) * bound)
        self.bias = nn.Parameter(torch.zeros(1, dtype=torch.float32))
        if dtype != torch.float32:
            self.weight.data = self.weight.data.to(dtype)
            self.bias.data = self.bias.data.to(dtype)

    def forward(self, x):
        if x.size(-1) % 4 == 0:
            return ops.ScalarHeadFn.apply(x, self.weight, self.bias)
        if x.is_cuda:
            # Ragged width (e.g. the 13 raw dense features): hipBLASLt
            # runs this bias-GEMV at ~9 GB/s (192 us measured at
            # 65536x13); zero-pad to a quad boundary and take the
            # streaming row_dot path instead (~2 us pad + ~5 us dot).
            pad = (-x.size(-1)) % 4
            xp = nn.functional.pad(x, (0, pad))
            wp = nn.functional.pad(self.weight, (0, pad))
            return ops.ScalarHeadFn.apply(xp, wp, self.bias)
        return x @ self.weight + self.bias


class FusedLinearReLU(nn.Module):
    """Linear (library GEMM) + fused bias+ReLU epilogue kernel."""

    def __init__(self, in_features: int, out_features: int,
                 dtype=torch.float32):
        super().__init__()
        self.weight = nn.Parameter(
            torch.empty(out_features, in_features, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype))
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    def forward(self, x):
        return ops.linear_bias_relu(x, self.weight, self.bias)


class WideAndDeep(nn.Module):
    """Wide (per-id scalar weights) + deep (embeddings -> MLP) CTR model."""

    def __init__(self,
                 dense_dim: int = CRITEO_DENSE,
                 table_sizes: Optional[List[int]] = None,
                 embedding_dim: int = 16,
                 hidden: Tuple[int, ...] = (1024, 512, 256),
                 compute_dtype: torch.dtype = torch.float32,
                 sharded: bool = False,
                 process_group=None):
        super().__init__()
        table_sizes = table_sizes or DEFAULT_TABLE_SIZES
        self.compute_dtype = compute_dtype
        self.dense_dim = dense_dim
        bf16 = compute_dtype == torch.bfloat16
        self.sharded = sharded
        if sharded:
            from tf_yarn_amd.models.sharded_embedding import \
                ShardedCriteoEmbeddings
            self.embeddings = ShardedCriteoEmbeddings(
                table_sizes, embedding_dim, out_bf16=bf16,
                process_group=process_group)
            self.deep_embedding = None
            self.wide_embedding = None
        else:
            self.embeddings = None
            self.deep_embedding = SparseEmbedding(
                table_sizes, embedding_dim, out_bf16=bf16)
            # Wide: one scalar weight per categorical id
            self.wide_embedding = WideScalarEmbedding(table_sizes,
                                                      out_bf16=bf16)
        self.wide_dense = ScalarHead(dense_dim, compute_dtype)
        layers: List[nn.Module] = []
        # dense block padded to 16 columns for quad-aligned fused gather
        in_dim = _DeepInput.DENSE_PAD + len(table_sizes) * embedding_dim
        for h in hidden:
            layers.append(FusedLinearReLU(in_dim, h))
            in_dim = h
        self.mlp = nn.Sequential(*layers)
        self.head = ScalarHead(in_dim, compute_dtype)
        if bf16:
            # dense compute in bf16; masters are managed by the optimizer
            self.mlp = self.mlp.to(torch.bfloat16)

    def forward(self, dense: torch.Tensor, sparse_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Returns logits [B], or — when ``labels`` is given — the mean
        BCEWithLogits loss with the 3-part logit sum fused into the loss
        kernel (ops.bce_head_loss)."""
        dense = dense.to(self.compute_dtype)
        if self.sharded:
            b = dense.shape[0]
            emb = self.embeddings
            deep_in = torch.empty(
                b, _DeepInput.DENSE_PAD + emb.F * emb.dim,
                dtype=self.compute_dtype, device=dense.device)
            deep_in[:, :self.dense_dim] = dense
            deep_in[:, self.dense_dim:_DeepInput.DENSE_PAD] = 0
            deep_in, wide_sum = emb(sparse_ids, deep_in,
                                    _DeepInput.DENSE_PAD)
            deep_out = self.head(self.mlp(deep_in))
            wide_sum = wide_sum.to(deep_out.dtype)
            wd = self.wide_dense(dense)
            if labels is not None:
                return ops.bce_head_loss(deep_out, wide_sum, wd, labels)
            return deep_out + wide_sum + wd
        emb = self.deep_embedding
        flat = (sparse_ids + emb.offsets.unsqueeze(0)).reshape(-1)
        deep_in = _DeepInput.apply(
            dense, emb.weight, flat, emb.dim,
            self.compute_dtype == torch.bfloat16, emb._sink)
        deep_out = self.head(self.mlp(deep_in))
        wide_emb = self.wide_embedding(sparse_ids).to(deep_out.dtype)
        wd = self.wide_dense(dense)
        if labels is not None:
            return ops.bce_head_loss(deep_out, wide_emb, wd, labels)
        return deep_out + wide_emb + wd

    def start_sparse_sync(self, process_group=None) -> None:
        """Kick off the replicated-mode allgathers (called right after
        ``loss.backward()`` so communication overlaps with the dense
        optimizer step).  Sharded mode already exchanged grads during
        backward (all-to-all), so there is nothing to start."""
        if self.sharded:
            return
        self.deep_embedding.start_sparse_sync(process_group)
        self.wide_embedding.start_sparse_sync(process_group)

    def finish_sparse_sync(self, lr: float, stream=None) -> None:
        if self.sharded:
            self.embeddings.apply_sparse_updates(lr, stream=stream)
            return
        self.deep_embedding.finish_sparse_sync(lr)
        self.wide_embedding.finish_sparse_sync(lr)

    def apply_sparse_updates(self, lr: float, process_group=None) -> None:
        self.start_sparse_sync(process_group)
        self.finish_sparse_sync(lr)

    def clear_pending(self) -> None:
        if self.sharded:
            self.embeddings.clear_pending()
            return
        self.deep_embedding.clear_pending()
        self.wide_embedding.clear_pending()


class FlatInputWideAndDeep(nn.Module):
    """WideAndDeep behind a single-tensor input — the shape the Keras
    ``fit(x, y)`` / Estimator ``input_fn`` surfaces expect:
    ``x = [dense fp32 columns | categorical id columns]``.  Importable
    (not example-local) so whole-model Keras checkpoints pickle."""

    def __init__(self, dense_dim: int = CRITEO_DENSE,
                 table_sizes: Optional[List[int]] = None,
                 embedding_dim: int = 16,
                 hidden: Tuple[int, ...] = (1024, 512, 256),
                 compute_dtype: torch.dtype = torch.float32,
                 sharded: bool = False, process_group=None):
        super().__init__()
        self.dense_dim = dense_dim
        self.net = WideAndDeep(dense_dim=dense_dim,
                               table_sizes=table_sizes,
                               embedding_dim=embedding_dim, hidden=hidden,
                               compute_dtype=compute_dtype,
                               sharded=sharded,
                               process_group=process_group)

    def forward(self, x: torch.Tensor,
                labels: Optional[torch.Tensor] = None) -> torch.Tensor:
        dense = x[:, :self.dense_dim]
        ids = x[:, self.dense_dim:].long()
        return self.net(dense, ids, labels=labels)

    def start_sparse_sync(self, process_group=None) -> None:
        self.net.start_sparse_sync(process_group)

    def finish_sparse_sync(self, lr: float, stream=None) -> None:
        self.net.finish_sparse_sync(lr, stream=stream)

    def apply_sparse_updates(self, lr: float, process_group=None) -> None:
        self.net.apply_sparse_updates(lr, process_group)

    def clear_pending(self) -> None:
        self.net.clear_pending()
