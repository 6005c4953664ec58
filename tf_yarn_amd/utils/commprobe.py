"""Lightweight per-phase communication timing probe.

Makes a poor multi-GPU scaling curve immediately attributable: when
enabled, every tagged collective span (sharded-embedding all-to-alls,
sparse allgathers, dense-reducer drain) records its device time via CUDA
event pairs (host ``perf_counter`` on CPU/gloo), and ``summary()``
reduces them to per-tag totals.  Disabled (the default) the probe is a
zero-cost no-op, so the bench's headline timed region is unperturbed —
bench.py runs a few *extra* instrumented steps after the timed region
and emits the result as ``comm_breakdown`` in its JSON line.

The reference delegates this diagnosis surface entirely to
``NCCL_DEBUG=INFO`` (``pytorch/tasks/worker.py:95``); here the per-phase
numbers come from the framework's own data plane.
"""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Tuple

import torch

_enabled = False
_cuda_spans: List[Tuple[str, "torch.cuda.Event", "torch.cuda.Event"]] = []
_host_totals: Dict[str, float] = defaultdict(float)
_counts: Dict[str, int] = defaultdict(int)


def enabled() -> bool:
    return _enabled


def enable() -> None:
    global _enabled
    _enabled = True


def disable() -> None:
    global _enabled
    _enabled = False


def reset() -> None:
    _cuda_spans.clear()
    _host_totals.clear()
    _counts.clear()


@contextmanager
def span(tag: str):
    """Time a tagged region.  On GPU, uses CUDA events on the current
    stream (device time, correct even for async collectives enqueued
    here); on CPU, host wall time."""
    if not _enabled:
        yield
        return
    if torch.cuda.is_available() and torch.cuda.is_initialized():
        start = torch.cuda.Event(enable_timing=True)
        end = torch.cuda.Event(enable_timing=True)
        start.record()
        try:
            yield
        finally:
            end.record()
            _cuda_spans.append((tag, start, end))
            _counts[tag] += 1
    else:
        t0 = time.perf_counter()
        try:
            yield
        finally:
            _host_totals[tag] += time.perf_counter() - t0
            _counts[tag] += 1


def add_host_time(tag: str, seconds: float) -> None:
    """Record an already-measured host-side duration (e.g. the reducer's
    work.wait() drain, which is host blocking by nature)."""
    if _enabled:
        _host_totals[tag] += seconds
        _counts[tag] += 1


def summary() -> Dict[str, Dict[str, float]]:
    """Reduce all recorded spans to {tag: {ms, count}}.  Synchronizes the
    device so every event pair has completed."""
    if torch.cuda.is_available() and torch.cuda.is_initialized():
        torch.cuda.synchronize()
    totals: Dict[str, float] = defaultdict(float)
    for tag, start, end in _cuda_spans:
        totals[tag] += start.elapsed_time(end)  # ms
    for tag, sec in _host_totals.items():
        totals[tag] += sec * 1000.0
    return {tag: {"ms": totals[tag], "count": _counts[tag]}
            for tag in totals}
