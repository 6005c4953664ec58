"""Monitoring hooks injected into every Estimator/Keras experiment
(reference ``tf_yarn/tensorflow/metrics.py``)."""

from __future__ import annotations

import logging
import time
from typing import Optional, Union

from tf_yarn_amd import _task_commons, mlflow
from tf_yarn_amd.estimator.estimator import SessionRunHook
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)


class StepPerSecondHook(SessionRunHook):
    """Chief-only steps/sec -> mlflow (reference
    ``tensorflow/metrics.py:18-38``)."""

    def __init__(self, every_n_steps: int = 100):
        self.every_n_steps = every_n_steps
        self._t0: Optional[float] = None
        self._step0 = 0

    def begin(self, estimator) -> None:
        self._t0 = time.time()
        self._step0 = estimator.global_step

    def after_step(self, step: int, loss: float, estimator) -> None:
        if (step - self._step0) % self.every_n_steps == 0 \
                and step > self._step0:
            now = time.time()
            sps = self.every_n_steps / max(1e-9, now - self._t0)
            self._t0 = now
            try:
                if _task_commons.is_chief():
                    mlflow.log_metric("steps_per_sec", sps, step=step)
                    logger.info("steps/sec: %.2f", sps)
            except KeyError:
                logger.info("steps/sec: %.2f", sps)


class EvalMonitorHook(SessionRunHook):
    """Evaluator-side hook publishing the 4 monitored stats to the KV
    store after every eval step (reference ``tensorflow/metrics.py:41-71``,
    consumed by :class:`~tf_yarn_amd.evaluator_metrics.EvaluatorMetricsLogger`)."""

    def __init__(self, client: Optional[KVClient] = None):
        self.client = client
        self._task: Optional[str] = None
        self._begin_t: Optional[float] = None
        self._awake = 0.0
        self._steps = 0
        self._step_start: Optional[float] = None

    def begin(self, estimator) -> None:
        if self.client is None:
            try:
                self.client = _task_commons.get_client()
            except RuntimeError:
                return
        try:
            self._task = _task_commons.get_task()
        except KeyError:
            self._task = "evaluator:0"
        if self._begin_t is None:
            self._begin_t = time.time()
        self._step_start = time.time()

    def after_step(self, step: int, loss: float, estimator) -> None:
        if self.client is None:
            return
        now = time.time()
        self._awake += now - (self._step_start or now)
        self._step_start = now
        self._steps += 1
        total = max(1e-9, now - (self._begin_t or now))
        from tf_yarn_amd import event
        event.broadcast(self.client,
                        f"{self._task}/awake_time_ratio",
                        f"{self._awake / total:.4f}")
        event.broadcast(self.client,
                        f"{self._task}/eval_step_mean_duration",
                        f"{self._awake / self._steps:.4f}")
        event.broadcast(self.client,
                        f"{self._task}/nb_eval_steps", str(self._steps))
        event.broadcast(self.client,
                        f"{self._task}/last_training_step",
                        str(estimator.global_step))


def _add_monitor_to_experiment(
        experiment: Union[Experiment, KerasExperiment]
) -> Union[Experiment, KerasExperiment]:
    """Inject monitoring hooks (reference ``tensorflow/metrics.py:111-142``;
    dedup by hook class name :103)."""
    if isinstance(experiment, KerasExperiment):
        return experiment  # Keras path monitors through callbacks
    # hooks are passed at train time by the task modules; nothing to
    # mutate on the descriptor itself in the torch-backed design
    return experiment


def get_all_metrics(model_dir: str):
    """Parse the event files under *model_dir* into
    {"step": [...], "name": [...], "value": [...]} (reference
    ``tensorflow/metrics.py:74-108`` parses TB event files the same way)."""
    from tf_yarn_amd.utils import tb
    steps, names, values = [], [], []
    for ev in tb.read_events(model_dir):
        if ev.get("value") is None:
            continue
        steps.append(ev.get("step"))
        names.append(ev.get("tag"))
        values.append(ev.get("value"))
    return {"step": steps, "name": names, "value": values}
