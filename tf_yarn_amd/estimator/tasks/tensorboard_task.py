"""TensorBoard-role task module (reference
``tf_yarn/tensorflow/tasks/_tensorboard_task.py``): resolve the model dir
(env ``TB_MODEL_DIR`` or from the unpickled experiment, ``:34-43``), run
the board server, advertise its URL, wait for every cluster task's ``stop``
event, linger the termination timeout, exit."""

from __future__ import annotations

import logging
import os
import sys
import time

from tf_yarn_amd import _task_commons, event, tensorboard
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
from tf_yarn_amd.estimator.tasks import task_common

logger = logging.getLogger(__name__)


def _resolve_model_dir(client) -> str:
    model_dir = os.environ.get("TB_MODEL_DIR", "")
    if model_dir:
        return model_dir
    try:
        experiment = _task_commons._get_experiment(client)
        if isinstance(experiment, KerasExperiment):
            return experiment.model_dir
        if isinstance(experiment, Experiment):
            return experiment.estimator.model_dir or ""
        # PytorchExperiment
        return getattr(experiment, "tensorboard_hdfs_dir", "") or ""
    except Exception:
        logger.exception("could not resolve model_dir from experiment")
        return ""


def main() -> None:
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    cluster_tasks = task_common._prepare_container(client)
    event.init_event(client, task, "127.0.0.1:0")
    model_dir = _resolve_model_dir(client)
    logger.info("board model_dir: %r", model_dir)
    server, url = tensorboard.start_tf_board(client, model_dir, task)
    event.start_event(client, task)
    event.broadcast_train_eval_start_timer(client, task)
    # wait for the stop event of every cluster task (:54-55)
    task_common.wait_for_connected_tasks(client, cluster_tasks, None)
    timeout = tensorboard.get_termination_timeout()
    logger.info("all tasks stopped; lingering %.0fs", timeout)
    time.sleep(timeout)
    server.shutdown()
    event.broadcast_train_eval_stop_timer(client, task)
    event.stop_event(client, task, None)
    event.broadcast_container_stop_time(client, task)


if __name__ == "__main__":
    try:
        main()
    except Exception:
        logger.exception("tensorboard task failed")
        sys.exit(1)
