"""Client-side monitoring of evaluator-published stats.

Parity with reference ``tf_yarn/evaluator_metrics.py``: the evaluator task
publishes four stats to the KV store after every eval step
(``evaluator_metrics.py:12-17``); the launcher polls them with per-metric
thresholds, dedups repeats, and forwards to mlflow.
"""

from __future__ import annotations

import logging
from typing import Dict

from tf_yarn_amd import mlflow
from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)

MONITORED_METRICS = {
    "awake_time_ratio": "Awake/idle ratio",
    "eval_step_mean_duration": "Eval step mean duration (in sec)",
    "last_training_step": "Last training step of evaluated checkpoint",
    "nb_eval_steps": "Number of evaluation steps done",
}


class EvaluatorMetricsLogger:
    """Reference ``evaluator_metrics.py:22-70``."""

    def __init__(self, evaluator_list, client: KVClient,
                 log_thresholds: Dict[str, list] = None,
                 n_try: int = 0):
        self.evaluator_list = evaluator_list
        self.client = client
        self.log_thresholds = log_thresholds or {}
        self.n_try = n_try
        self.last_metrics: Dict[str, Dict[str, str]] = {
            e: {m: None for m in MONITORED_METRICS}
            for e in evaluator_list
        }

    def log(self) -> None:
        for evaluator in self.evaluator_list:
            for metric, label in MONITORED_METRICS.items():
                key = f"{evaluator}/{metric}"
                raw = self.client.get(key)
                if raw is None:
                    continue
                value = raw.decode()
                if value == self.last_metrics[evaluator][metric]:
                    continue  # dedup repeats (reference :52-62)
                self.last_metrics[evaluator][metric] = value
                if not self._within_thresholds(metric, value):
                    continue
                logger.info("%s %s: %s", evaluator, label, value)
                mlflow.log_metric(
                    f"{mlflow.format_key(key)}_{self.n_try}", float(value))

    def _within_thresholds(self, metric: str, value: str) -> bool:
        if metric not in self.log_thresholds:
            return True
        try:
            v = float(value)
        except ValueError:
            return True
        lo, hi = self.log_thresholds[metric]
        if lo is not None and v < lo:
            return False
        if hi is not None and v > hi:
            return False
        return True
