"""Minimal TensorBoard-style event writing.

The environment has no ``tensorboard`` package; this writer keeps the
reference's per-worker summary-writer contract
(``pytorch/tasks/worker.py:113`` passes a writer to ``main_fn``;
the evaluator derives "already evaluated" steps from eval event files,
``evaluator_task.py:134-140``) with a JSONL event format that the
framework's own tensorboard side-task serves over HTTP.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, List, Optional


class SummaryWriter:
    """API-compatible subset of ``torch.utils.tensorboard.SummaryWriter``:
    ``add_scalar`` / ``add_scalars`` / ``flush`` / ``close``."""

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._path = os.path.join(
            log_dir, f"events.miyarn.{int(time.time() * 1e6)}.jsonl")
        self._fd = open(self._path, "a", buffering=1)

    def add_scalar(self, tag: str, value, step: Optional[int] = None,
                   walltime: Optional[float] = None) -> None:
        self._fd.write(json.dumps({
            "tag": tag,
            "value": float(value),
            "step": int(step) if step is not None else None,
            "time": walltime if walltime is not None else time.time(),
        }) + "\n")

    def add_scalars(self, main_tag: str, tag_scalar_dict: Dict[str, float],
                    step: Optional[int] = None) -> None:
        for k, v in tag_scalar_dict.items():
            self.add_scalar(f"{main_tag}/{k}", v, step)

    def flush(self) -> None:
        self._fd.flush()

    def close(self) -> None:
        try:
            self._fd.flush()
            self._fd.close()
        except ValueError:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


def read_events(log_dir: str) -> List[dict]:
    """Read all events under *log_dir* (recursively)."""
    events = []
    for root, _, files in os.walk(log_dir):
        for name in sorted(files):
            if ".jsonl" not in name:
                continue
            with open(os.path.join(root, name)) as fd:
                for line in fd:
                    line = line.strip()
                    if line:
                        try:
                            events.append(json.loads(line))
                        except json.JSONDecodeError:
                            pass
    return events
