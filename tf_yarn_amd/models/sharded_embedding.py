"""Feature-sharded embeddings with all-to-all exchange (model-parallel
embeddings, DLRM-style) — the xGMI-native scaling path.

With replicated tables, data-parallel sparse sync needs an allgather of
every rank's (ids, grad rows): ~54 MB/rank/step at batch 65536, which a
ring allgather moves in ~2.5 ms over 8 GPUs — more than the whole 1-GPU
step.  xGMI is 7 point-to-point links per GPU, so ALL-TO-ALL traffic
rides all links concurrently; sharding the 26 categorical tables across
ranks (rank r owns features f with f % world == r) turns the sparse
exchange into two all-to-alls (~48 MB each way, ~0.2 ms) and makes every
table row owner-updated (no cross-rank gradient dup at all).

Flow per step (world W, local batch B, F features, dim D):
  fwd:  ids[:, feats_of(s)] --all-to-all--> owners gather their shard
        (fused HIP kernel, bf16 out) --all-to-all--> [B, F*D] assembled
        straight into the MLP input buffer; the wide part reuses the SAME
        routed ids, owners compute partial gather-sums, tiny all-to-all
        back.
  bwd:  grad slices --all-to-all--> owners stash (ids, grads); the wide
        grad is a small allgather.
  update: owner-local fused scatter+SGD, scale 1/W (no further comm).

W == 1 degenerates to the local fused path with zero communication.
gloo has no all_to_all: a correctness-equivalent allgather+slice
emulation backs the CPU tests (`_all_to_all_single`).
"""

from __future__ import annotations

import logging
import weakref
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist
from torch import nn

from tf_yarn_amd import ops
from tf_yarn_amd.utils import commprobe

logger = logging.getLogger(__name__)


def _world_rank(group) -> Tuple[int, int]:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(group), dist.get_rank(group)
    return 1, 0


# Per-group probe cache, weak-keyed on the ProcessGroup object itself so a
# destroyed+gc'd group can never alias a new group's entry (id() reuse).
_ALLTOALL_MODE: "weakref.WeakKeyDictionary" = weakref.WeakKeyDictionary()


# Binned scatter pays only when ids repeat (the LDS-hash dedup divides
# the table RMW traffic by the duplication factor but costs a ~200 us
# permutation).  From the round-2 A/B at the bench shape: atomic 422 us
# flat; binned perm ~200 + applies that scale with unique/n — break-even
# near duplication factor ~2.
SKEW_THRESHOLD = 2.0


def _binned_mode() -> str:
    """MIYARN_BINNED_SCATTER: '1' force on, '0' force off, unset/'auto'
    = probe the id distribution once and decide (uniform ids keep the
    measured-fastest atomic kernels; power-law CTR traffic flips to the
    dedup path automatically)."""
    import os
    v = os.environ.get("MIYARN_BINNED_SCATTER", "auto")
    if v == "1":
        return "on"
    if v in ("", "0"):
        return "off"
    return "auto"


def negotiated_alltoall_mode(group=None) -> str:
    """The mode the self-check picked for this group: 'native',
    'emulate', 'gloo-emulate', or 'unprobed' (reported in bench.py's
    JSON so a silent fallback is visible in the scale run)."""
    if not (dist.is_available() and dist.is_initialized()):
        return "unprobed"
    if dist.get_backend(group) == "gloo":
        return "gloo-emulate"
    key = (group if group is not None
           else dist.distributed_c10d._get_default_group())
    return _ALLTOALL_MODE.get(key, "unprobed")


def _perm_stream_enabled() -> bool:
    import os
    v = os.environ.get("MIYARN_PERM_STREAM", "1")
    return v != "" and v != "0"


def _alltoall_self_check(group=None) -> str:
    """One-time probe: every rank sends rank*100+dest to dest; verify the
    native all_to_all_single delivers it.  On any mismatch or error the
    group permanently falls back to the allgather emulation (correct on
    every backend) — insurance for the unattended 8-GPU scale run."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    device = (torch.device("cuda", torch.cuda.current_device())
              if dist.get_backend(group) == "nccl" else "cpu")
    send = torch.tensor([rank * 100 + d for d in range(world)],
                        dtype=torch.int64, device=device)
    recv = torch.zeros(world, dtype=torch.int64, device=device)
    try:
        dist.all_to_all_single(recv, send,
                               output_split_sizes=[1] * world,
                               input_split_sizes=[1] * world, group=group)
        expected = torch.tensor([s * 100 + rank for s in range(world)],
                                dtype=torch.int64, device=device)
        ok = bool(torch.equal(recv, expected))
    except Exception:  # noqa: BLE001
        ok = False
    # every rank must agree on the mode
    flag = torch.tensor([1 if ok else 0], dtype=torch.int64,
                        device=device)
    dist.all_reduce(flag, op=dist.ReduceOp.MIN, group=group)
    mode = "native" if int(flag.item()) == 1 else "emulate"
    if mode == "emulate":
        logger.warning("all_to_all_single self-check failed; using the "
                       "allgather emulation for this group")
    return mode


def _all_to_all_single(output: torch.Tensor, input_: torch.Tensor,
                       out_splits: List[int], in_splits: List[int],
                       group=None) -> None:
    """dist.all_to_all_single with a gloo fallback (allgather + slice) so
    the distributed path is testable on CPU (gloo lacks alltoall), plus a
    one-time native self-check on other backends."""
    backend = dist.get_backend(group)
    if backend != "gloo":
        key = (group if group is not None
               else dist.distributed_c10d._get_default_group())
        mode = _ALLTOALL_MODE.get(key)
        if mode is None:
            mode = _alltoall_self_check(group)
            _ALLTOALL_MODE[key] = mode
        if mode == "native":
            dist.all_to_all_single(output, input_,
                                   output_split_sizes=out_splits,
                                   input_split_sizes=in_splits,
                                   group=group)
            return
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    # Emulation: allgather every rank's full input (+ splits), then take
    # the chunk addressed to this rank from each peer.
    my_splits = torch.tensor(in_splits, dtype=torch.int64)
    all_splits = [torch.zeros(world, dtype=torch.int64)
                  for _ in range(world)]
    dist.all_gather(all_splits, my_splits, group=group)
    max_numel = max(int(s.sum()) for s in all_splits)
    padded = torch.zeros(max_numel, dtype=input_.dtype,
                         device=input_.device)
    padded[:input_.numel()] = input_.reshape(-1)
    gathered = [torch.zeros_like(padded) for _ in range(world)]
    dist.all_gather(gathered, padded, group=group)
    offset = 0
    for p in range(world):
        splits_p = all_splits[p].tolist()
        start = sum(splits_p[:rank])
        n = splits_p[rank]
        output.reshape(-1)[offset:offset + n] = \
            gathered[p][start:start + n]
        offset += n


class _ShardedLookup(torch.autograd.Function):
    """Forward/backward all-to-all routing around the owner-local fused
    gather; backward stashes owner-local (ids, grads) in the sink."""

    @staticmethod
    def forward(ctx, deep_table, wide_table, module, ids, deep_out_buf,
                col_offset):
        m: "ShardedCriteoEmbeddings" = module
        W, B = m.world, ids.shape[0]
        D = m.dim
        device = ids.device

        if W == 1:
            flat = (ids + m.full_offsets.unsqueeze(0)).reshape(-1)
            ops.emb_fwd_into(deep_table, flat, deep_out_buf, col_offset)
            wide_out = ops.emb_gather_sum(wide_table, flat.reshape(B, -1),
                                          m.out_bf16)
            m._start_perm_async(flat)
            ctx.module = m
            ctx.save_for_backward(flat)
            ctx.local_only = True
            ctx.mark_dirty(deep_out_buf)
            return deep_out_buf, wide_out

        # ---- route ids to owners ----------------------------------------
        # send chunk for peer s: ids[:, feats_of(s)] row-major [B, F_s]
        ids_send = torch.cat(
            [ids[:, m.feats_of[s]].reshape(-1) for s in range(W)])
        in_splits = [B * len(m.feats_of[s]) for s in range(W)]
        n_own = B * m.f_own
        ids_recv = torch.empty(W * n_own, dtype=torch.int64, device=device)
        with commprobe.span("a2a_ids"):
            _all_to_all_single(ids_recv, ids_send, [n_own] * W, in_splits,
                               m.group)
        # local row offsets: layout [peer][b][j], j = feature slot
        flat_local = ids_recv + m.own_offsets_tiled[:ids_recv.numel()]

        m._start_perm_async(flat_local)

        # ---- owner-local gathers ----------------------------------------
        vec = ops.emb_fwd(deep_table, flat_local, m.out_bf16)  # [W*n_own, D]
        wide_partial = ops.emb_gather_sum(
            wide_table, flat_local.reshape(W * B, m.f_own), m.out_bf16)

        # ---- route vectors back -----------------------------------------
        vec_recv = torch.empty(B * m.F * D, dtype=vec.dtype, device=device)
        with commprobe.span("a2a_vec"):
            _all_to_all_single(
                vec_recv, vec.reshape(-1),
                [B * len(m.feats_of[s]) * D for s in range(W)],
                [n_own * D] * W, m.group)
        # assemble [B, F, D] (perm feature order) into the MLP input slice
        pos = 0
        target = deep_out_buf[:, col_offset:col_offset + m.F * D] \
            .reshape(B, m.F, D)
        off = 0
        for s in range(W):
            f_s = len(m.feats_of[s])
            chunk = vec_recv[off:off + B * f_s * D].reshape(B, f_s, D)
            target[:, pos:pos + f_s, :] = chunk
            pos += f_s
            off += B * f_s * D

        wide_recv = torch.empty(W * B, dtype=wide_partial.dtype,
                                device=device)
        with commprobe.span("a2a_wide"):
            _all_to_all_single(wide_recv, wide_partial.reshape(-1),
                               [B] * W, [B] * W, m.group)
        wide_out = wide_recv.reshape(W, B).sum(dim=0)

        ctx.module = m
        ctx.save_for_backward(flat_local)
        ctx.local_only = False
        ctx.B = B
        ctx.mark_dirty(deep_out_buf)
        return deep_out_buf, wide_out

    @staticmethod
    def backward(ctx, grad_deep_buf, grad_wide):
        m: "ShardedCriteoEmbeddings" = ctx.module
        (flat_ids,) = ctx.saved_tensors
        W, D = m.world, m.dim
        # grad_deep_buf is the WHOLE MLP-input grad; the embedding slice
        # starts at the col_offset recorded by the module
        col_offset = m._last_col_offset
        emb_grad = grad_deep_buf[:, col_offset:]
        if ctx.local_only:
            m._deep_sink.append((flat_ids, emb_grad.contiguous()))
            m._wide_sink.append((flat_ids, grad_wide.contiguous()))
            return None, None, None, None, None, None
        B = ctx.B
        n_own = B * m.f_own
        # route deep grads to owners: send chunk for owner s = the
        # features s owns, [B, F_s, D] row-major
        g3 = emb_grad.reshape(B, m.F, D)
        pos = 0
        sends = []
        for s in range(W):
            f_s = len(m.feats_of[s])
            sends.append(g3[:, pos:pos + f_s, :].reshape(-1))
            pos += f_s
        g_send = torch.cat(sends)
        g_recv = torch.empty(W * n_own * D, dtype=g_send.dtype,
                             device=g3.device)
        with commprobe.span("a2a_grad"):
            _all_to_all_single(
                g_recv, g_send, [n_own * D] * W,
                [B * len(m.feats_of[s]) * D for s in range(W)], m.group)
        m._deep_sink.append((flat_ids, g_recv.reshape(W * n_own, D)))
        # wide: every owner needs every rank's grad_wide [B]
        gw = grad_wide.contiguous().reshape(-1)
        gw_all = torch.empty(W * B, dtype=gw.dtype, device=gw.device)
        with commprobe.span("ag_wide"):
            dist.all_gather_into_tensor(gw_all, gw, group=m.group)
        m._wide_sink.append((flat_ids, gw_all))
        return None, None, None, None, None, None


class ShardedCriteoEmbeddings(nn.Module):
    """Deep (dim-D) + wide (dim-1) categorical embeddings, feature-sharded
    across the process group.  ``forward(ids, deep_out_buf, col_offset)``
    fills ``deep_out_buf[:, col_offset:]`` with [B, F*D] (owner-permuted
    feature order, consistent across steps) and returns the wide sums [B].
    """

    def __init__(self, table_sizes: List[int], dim: int,
                 out_bf16: bool = False, process_group=None,
                 init_std: Optional[float] = None,
                 wide_init_std: float = 0.01):
        super().__init__()
        import math
        self.F = len(table_sizes)
        self.dim = dim
        self.out_bf16 = out_bf16
        self.group = process_group
        self.world, self.rank = _world_rank(process_group)
        W = self.world
        self.feats_of = [[f for f in range(self.F) if f % W == s]
                         for s in range(W)]
        self.own_feats = self.feats_of[self.rank]
        self.f_own = len(self.own_feats)
        own_sizes = [table_sizes[f] for f in self.own_feats]
        total_own = sum(own_sizes)
        std = init_std if init_std is not None else 1.0 / math.sqrt(dim)
        self.weight = nn.Parameter(torch.randn(total_own, dim) * std)
        self.weight._miyarn_sparse = True
        self.weight._miyarn_sharded = True  # never broadcast/replicate
        self.wide_weight = nn.Parameter(
            torch.randn(total_own, 1) * wide_init_std)
        self.wide_weight._miyarn_sparse = True
        self.wide_weight._miyarn_sharded = True
        offs = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(own_sizes), 0)[:-1]),
            dtype=torch.int64)
        # PER-RANK buffers (different size/content on each rank): they must
        # never be broadcast by the reducer — flagged _miyarn_sharded,
        # which BucketedDataParallel skips.
        offs._miyarn_sharded = True
        self.register_buffer("own_offsets", offs, persistent=True)
        # full offsets for the W == 1 local path (perm == identity there)
        full = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(table_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("full_offsets", full, persistent=True)
        tiled0 = torch.empty(0, dtype=torch.int64)
        tiled0._miyarn_sharded = True
        self.register_buffer("own_offsets_tiled", tiled0,
                             persistent=False)
        self._deep_sink: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self._wide_sink: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self._last_col_offset = 0
        self._perm_stream = None  # side stream for async pass A
        self._pending_perm = None  # (ids data_ptr, (order, starts), event)
        self._binned_auto: Optional[bool] = None  # skew probe result

    def _ensure_tiled(self, numel: int, device) -> None:
        if self.own_offsets_tiled.numel() < numel:
            reps = (numel + self.f_own - 1) // self.f_own
            tiled = self.own_offsets.to(device).repeat(reps)[:numel]
            tiled._miyarn_sharded = True  # per-rank: never broadcast
            self.own_offsets_tiled = tiled

    def forward(self, ids: torch.Tensor, deep_out_buf: torch.Tensor,
                col_offset: int
                ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (deep_buf_with_grad_fn, wide_sums [B]).  Downstream
        code MUST use the returned buffer (it carries the autograd
        edge)."""
        self._last_col_offset = col_offset
        if self.world > 1:
            self._ensure_tiled(self.world * ids.shape[0] * self.f_own,
                               ids.device)
        else:
            self._ensure_tiled(ids.numel(), ids.device)
        out_buf, wide = _ShardedLookup.apply(
            self.weight, self.wide_weight, self, ids, deep_out_buf,
            col_offset)
        return out_buf, wide

    @torch.no_grad()
    def apply_sparse_updates(self, lr: float, stream=None) -> None:
        """Owner-local fused scatter+SGD; scale 1/world (grads come from a
        global-mean loss split across ranks).

        ``stream``: optional side HIP stream.  The scatters are atomic-
        bound (~0.8 TB/s, CUs mostly idle), so running them concurrently
        with the dense optimizer step hides most of their latency; the
        CALLER must make the compute stream wait on ``stream`` before the
        next forward's gather reads the tables."""
        if stream is not None:
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for flat_ids, grad in self._deep_sink:
                    flat_ids.record_stream(stream)
                    grad.record_stream(stream)
                for flat_ids, gw in self._wide_sink:
                    flat_ids.record_stream(stream)
                    gw.record_stream(stream)
                self._apply_pending(lr)
            return
        self._apply_pending(lr)

    def _start_perm_async(self, flat_ids: torch.Tensor) -> None:
        """Kick off the binned-scatter permutation (pass A) on a side
        stream DURING FORWARD: it depends only on the ids, and its
        ~200 us of latency/atomic-bound work hides under the MFMA-bound
        MLP GEMMs instead of sitting on the backward critical path."""
        if not (flat_ids.is_cuda and self.dim == 16 and ops.HAVE_EXT
                and torch.is_grad_enabled() and self._binned_on(flat_ids)):
            return
        rb = ops.pick_region_bits(self.weight.shape[0], flat_ids.numel())
        if not _perm_stream_enabled():
            # inline (A/B diagnostic): pass A runs on the main stream
            perm = ops.binned_permutation(flat_ids, self.weight.shape[0],
                                          rb)
            self._pending_perm = (flat_ids.data_ptr(), perm, None)
            return
        if self._perm_stream is None:
            self._perm_stream = torch.cuda.Stream()
        self._perm_stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._perm_stream):
            flat_ids.record_stream(self._perm_stream)
            perm = ops.binned_permutation(flat_ids, self.weight.shape[0],
                                          rb)
        ev = torch.cuda.Event()
        ev.record(self._perm_stream)
        self._pending_perm = (flat_ids.data_ptr(), perm, ev)

    def _binned_on(self, flat_ids: torch.Tensor) -> bool:
        """Resolve the binned-vs-atomic choice (see ``_binned_mode``)."""
        mode = _binned_mode()
        if mode == "on":
            return True
        if mode == "off":
            return False
        if self._binned_auto is None:
            n = flat_ids.numel()
            if n == 0:
                return False
            uniq = int(torch.unique(flat_ids).numel())
            dup = n / max(1, uniq)
            self._binned_auto = dup >= SKEW_THRESHOLD
            logger.info(
                "sparse scatter auto-select: duplication factor %.2f -> "
                "%s path", dup,
                "binned-dedup" if self._binned_auto else "atomic")
        return self._binned_auto

    def _apply_pending(self, lr: float) -> None:
        scale = 1.0 / self.world
        # Binned scatter (round-2 kernel): one region-binned permutation
        # shared by the deep and wide updates (identical flat ids), LDS
        # dedup + exclusive-owner writeback instead of 27M random global
        # atomics.  MIYARN_BINNED_SCATTER=0 restores the atomic kernels
        # for A/B runs.
        first_ids = self._deep_sink[0][0] if self._deep_sink else None
        use_binned = (self.weight.is_cuda and self.dim == 16
                      and ops.HAVE_EXT and first_ids is not None
                      and self._binned_on(first_ids))
        perms = {}
        if use_binned and self._pending_perm is not None:
            key, perm, ev = self._pending_perm
            if ev is not None:
                torch.cuda.current_stream().wait_event(ev)
                for t in perm:
                    t.record_stream(torch.cuda.current_stream())
            perms[key] = perm
            self._pending_perm = None

        def perm_for(flat_ids):
            key = flat_ids.data_ptr()
            if key not in perms:
                rb = ops.pick_region_bits(self.weight.shape[0],
                                          flat_ids.numel())
                perms[key] = ops.binned_permutation(
                    flat_ids, self.weight.shape[0], rb)
            return perms[key]

        # Atomic fast path: deep+wide updates share the same flat ids, so
        # one fused kernel reads the ids once and issues both scatters
        # (saves the wide kernel's launch + its 13.6 MB id re-read).
        wide_by_ids = {gw_ids.data_ptr(): (i, gw)
                       for i, (gw_ids, gw) in enumerate(self._wide_sink)}
        fused_wide_done = set()
        for flat_ids, grad in self._deep_sink:
            grad2d = grad.reshape(flat_ids.numel(), self.dim)
            if use_binned:
                ops.emb_bwd_sgd_binned(
                    self.weight.data, flat_ids, grad2d,
                    lr=lr, scale=scale, perm=perm_for(flat_ids))
                continue
            key = flat_ids.data_ptr()
            mate = wide_by_ids.get(key)
            if (mate is not None and flat_ids.is_cuda and ops.HAVE_EXT
                    and self.dim % 4 == 0
                    and grad2d.dtype == mate[1].dtype
                    and mate[0] not in fused_wide_done):
                ops.emb_bwd_sgd_fused_wide(
                    self.weight.data, self.wide_weight.data, flat_ids,
                    grad2d, mate[1], lr=lr, scale=scale)
                fused_wide_done.add(mate[0])
            else:
                ops.emb_bwd_sgd(self.weight.data, flat_ids, grad2d,
                                lr=lr, scale=scale)
        self._deep_sink.clear()
        for i, (flat_ids, gw) in enumerate(self._wide_sink):
            if i in fused_wide_done:
                continue
            # gw layout [W, B] (or [B] local); ids layout [peer][b][j]
            if use_binned:
                ops.emb_scatter_sum_binned(
                    self.wide_weight.data, flat_ids, gw,
                    alpha=-lr * scale, perm=perm_for(flat_ids))
                continue
            fan = self.F if self.world == 1 else self.f_own
            ids2d = flat_ids.reshape(gw.numel(), fan)
            ops.emb_scatter_sum(self.wide_weight.data, ids2d, gw,
                                alpha=-lr * scale)
        self._wide_sink.clear()

    def clear_pending(self) -> None:
        self._deep_sink.clear()
        self._wide_sink.clear()
