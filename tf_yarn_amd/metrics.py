"""Run-result metrics (parity with reference ``tf_yarn/metrics.py``)."""

from __future__ import annotations

import logging
from typing import Dict, List, NamedTuple, Optional, Tuple

from tf_yarn_amd import mlflow
from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)


class Metrics(NamedTuple):
    """Aggregated run result (reference ``metrics.py:19-38``)."""
    total_training_duration: Optional[float]
    total_eval_duration: Optional[float]
    container_duration: Dict[str, Optional[float]]
    train_eval_time_per_node: Dict[str, Optional[float]]

    def log_mlflow(self, n_try: int = 0) -> None:
        if self.total_training_duration is not None:
            mlflow.log_metric(
                f"total_training_duration_{n_try}",
                self.total_training_duration)
        if self.total_eval_duration is not None:
            mlflow.log_metric(
                f"total_eval_duration_{n_try}", self.total_eval_duration)
        for task, duration in self.container_duration.items():
            if duration is not None:
                mlflow.log_metric(
                    f"{mlflow.format_key(task)}_container_duration_{n_try}",
                    duration)
        for task, duration in self.train_eval_time_per_node.items():
            if duration is not None:
                mlflow.log_metric(
                    f"{mlflow.format_key(task)}_train_eval_time_{n_try}",
                    duration)


class OneShotMetricsLogger:
    """Poll KV keys once each and log them as they appear
    (reference ``metrics.py:41-59``); used e.g. for the tensorboard URL."""

    def __init__(self, client: KVClient,
                 events: List[Tuple[str, str]],
                 n_try: int = 0):
        self.client = client
        self.events = events
        self.n_try = n_try

    def log(self) -> None:
        self.events = [e for e in self.events if not self._log(*e)]

    def _log(self, key: str, label: str) -> bool:
        value = self.client.get(key)
        if value is None:
            return False
        logger.info("%s %s", label, value.decode())
        mlflow.set_tag(f"{mlflow.format_key(label)}_{self.n_try}",
                       value.decode())
        return True
