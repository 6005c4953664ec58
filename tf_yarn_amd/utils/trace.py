"""Per-step phase tracing.

The reference's only timing is the coarse ``catchtime`` context manager and
KV timestamp events (SURVEY §5 "Tracing/profiling"); both are kept
(``_task_commons.catchtime``, ``event.broadcast_*_timer``).  This module
adds the MI355X-side instrument: a CUDA-event phase timer for training
loops (the measurement tool behind ``scripts/perf_probe.py``) plus a
chrome-trace exporter so a step breakdown can be opened in
``chrome://tracing`` / perfetto alongside rocprofv3 output.
"""

from __future__ import annotations

import json
import time
from collections import defaultdict
from typing import Dict, List, Optional

import torch


class PhaseTimer:
    """Times named phases of a training step.

    GPU phases use CUDA events (async, resolved at ``summary()``); on CPU
    it falls back to wall clock.  Usage::

        timer = PhaseTimer()
        for step in range(n):
            with timer.phase("forward"):
                out = model(x)
            with timer.phase("backward"):
                loss.backward()
        print(timer.summary())
    """

    def __init__(self, use_cuda: Optional[bool] = None):
        self.use_cuda = (torch.cuda.is_available()
                         if use_cuda is None else use_cuda)
        self._events: Dict[str, List] = defaultdict(list)
        self._trace: List[dict] = []
        self._t0 = time.perf_counter()

    class _Phase:
        def __init__(self, timer: "PhaseTimer", name: str):
            self.timer = timer
            self.name = name

        def __enter__(self):
            t = self.timer
            if t.use_cuda:
                start = torch.cuda.Event(enable_timing=True)
                start.record()
                self._start = start
            else:
                self._start = time.perf_counter()
            return self

        def __exit__(self, *exc):
            t = self.timer
            if t.use_cuda:
                end = torch.cuda.Event(enable_timing=True)
                end.record()
                t._events[self.name].append((self._start, end))
            else:
                t._events[self.name].append(
                    (self._start, time.perf_counter()))

    def phase(self, name: str) -> "PhaseTimer._Phase":
        return PhaseTimer._Phase(self, name)

    def _times_ms(self, name: str) -> List[float]:
        out = []
        for start, end in self._events[name]:
            if self.use_cuda:
                end.synchronize()
                out.append(start.elapsed_time(end))
            else:
                out.append((end - start) * 1000.0)
        return out

    def summary(self) -> Dict[str, Dict[str, float]]:
        """Per-phase {median, min, max, total} in milliseconds."""
        result = {}
        for name in self._events:
            times = sorted(self._times_ms(name))
            if not times:
                continue
            result[name] = {
                "median_ms": times[len(times) // 2],
                "min_ms": times[0],
                "max_ms": times[-1],
                "total_ms": sum(times),
                "count": len(times),
            }
        return result

    def export_chrome_trace(self, path: str) -> None:
        """Write a chrome://tracing-compatible JSON of all phases."""
        events = []
        cursor: Dict[str, float] = defaultdict(float)
        for name in self._events:
            for dur in self._times_ms(name):
                events.append({
                    "name": name, "ph": "X", "pid": 0, "tid": name,
                    "ts": cursor[name] * 1000.0,
                    "dur": dur * 1000.0,
                })
                cursor[name] += dur
        with open(path, "w") as fd:
            json.dump({"traceEvents": events}, fd)
