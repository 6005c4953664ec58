"""Side-car continuous evaluator task.

Parity with reference ``tf_yarn/tensorflow/tasks/evaluator_task.py``: scan
``model_dir`` for ``model.ckpt-N`` checkpoints not yet evaluated (the
done-set is derived from the eval event files, ``:134-140``), run
``estimator.evaluate(checkpoint_path=...)`` per checkpoint with the
monitoring hook, run the eval_spec's exporters, and stop when
``train_spec.max_steps`` is reached or after a 20-minute idle timeout
(``:21-23``)."""

from __future__ import annotations

import logging
import os
import sys
import time
from typing import Optional, Set

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.estimator import estimator as est_mod
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
from tf_yarn_amd.estimator.metrics import EvalMonitorHook
from tf_yarn_amd.estimator.tasks import task_common
from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)

IDLE_TIMEOUT_SECS = 20 * 60  # reference evaluator_task.py:21-23
SLEEP_SECS = 10
KERAS_SLEEP_SECS = 30  # reference :54-74


def get_ckpt_to_eval(model_dir: str,
                     evaluated: Set[int]) -> Optional[str]:
    """Oldest unevaluated checkpoint (reference ``:38-51``)."""
    for path in est_mod.list_checkpoints(model_dir):
        if est_mod.checkpoint_step(path) not in evaluated:
            return path
    return None


def stop_cond_reached(max_steps: Optional[int],
                      evaluated: Set[int]) -> bool:
    """Reference ``:28-35``."""
    return bool(max_steps is not None and evaluated
                and max(evaluated) >= max_steps)


def evaluate(experiment: Experiment,
             client: Optional[KVClient] = None) -> None:
    """Continuous-eval loop (reference ``:76-127``)."""
    estimator = experiment.estimator
    model_dir = estimator.model_dir
    max_steps = experiment.train_spec.max_steps
    eval_spec = experiment.eval_spec
    evaluated: Set[int] = set(
        est_mod.evaluated_steps(estimator.eval_dir(eval_spec.name)))
    hook = EvalMonitorHook(client)
    last_progress = time.time()
    time.sleep(eval_spec.start_delay_secs)
    while True:
        ckpt = get_ckpt_to_eval(model_dir, evaluated)
        if ckpt is not None:
            step = est_mod.checkpoint_step(ckpt)
            logger.info("evaluating %s", ckpt)
            result = estimator.evaluate(
                eval_spec.input_fn, steps=eval_spec.steps,
                checkpoint_path=ckpt, name=eval_spec.name,
                hooks=[hook])
            evaluated.add(step)
            last_progress = time.time()
            for exporter in eval_spec.exporters:
                export_path = os.path.join(model_dir, exporter.name)
                exporter.export(estimator, export_path, ckpt, result)
            if stop_cond_reached(max_steps, evaluated):
                logger.info("max_steps %s evaluated; stopping", max_steps)
                return
            time.sleep(eval_spec.throttle_secs)
        else:
            if stop_cond_reached(max_steps, evaluated):
                return
            if time.time() - last_progress > IDLE_TIMEOUT_SECS:
                logger.info("no new checkpoint for %ds; stopping",
                            IDLE_TIMEOUT_SECS)
                return
            time.sleep(SLEEP_SECS)


def keras_evaluate(experiment: KerasExperiment,
                   client: Optional[KVClient] = None) -> None:
    """Keras variant: reload the whole model per checkpoint-{epoch}
    (reference ``:54-74``)."""
    from tf_yarn_amd.estimator.keras import load_model
    model_dir = experiment.model_dir
    evaluated: Set[str] = set()
    last_progress = time.time()
    epochs = experiment.train_params.get("epochs", 1)
    while True:
        ckpts = sorted(
            p for p in (os.listdir(model_dir)
                        if os.path.isdir(model_dir) else [])
            if p.startswith("checkpoint-"))
        new = [c for c in ckpts if c not in evaluated]
        if new:
            for name in new:
                path = os.path.join(model_dir, name)
                model = load_model(path)
                if experiment.validation_data_fn is not None:
                    vx, vy = experiment.validation_data_fn()
                    if model.loss_fn is None:
                        model.compile(optimizer="sgd", loss="mse")
                    loss = model.evaluate(vx, vy)
                    logger.info("eval %s: loss=%.5f", name, loss)
                evaluated.add(name)
            last_progress = time.time()
            if len(evaluated) >= epochs:
                return
        else:
            if time.time() - last_progress > IDLE_TIMEOUT_SECS:
                return
            time.sleep(KERAS_SLEEP_SECS)


def evaluator_fn(client: Optional[KVClient], experiment) -> None:
    """Reference ``:18-25``."""
    if isinstance(experiment, KerasExperiment):
        keras_evaluate(experiment, client)
    else:
        evaluate(experiment, client)


def main() -> None:
    """Reference ``:143-158``."""
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    cluster_tasks = task_common._prepare_container(client)
    event.init_event(client, task, "127.0.0.1:0")
    experiment = _task_commons._get_experiment(client)
    thread = task_common._execute_dispatched_function(
        client, lambda: evaluator_fn(client, experiment))
    thread.join()
    task_common._shutdown_container(client, cluster_tasks, None, thread)


if __name__ == "__main__":
    try:
        main()
    except Exception:
        logger.exception("evaluator failed")
        sys.exit(1)
