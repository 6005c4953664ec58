"""Estimator/Keras flavor ``run_on_yarn`` (reference
``tf_yarn/tensorflow/client.py``): wraps the user fn with monitoring-hook
injection and delegates to the core spawner; default topology is
``single_server_topology()`` (reference ``tensorflow/client.py:14``)."""

from typing import Callable, Dict, Optional

from tf_yarn_amd import client as core_client
from tf_yarn_amd.estimator.metrics import _add_monitor_to_experiment
from tf_yarn_amd.metrics import Metrics
from tf_yarn_amd.topologies import TaskSpec, single_server_topology


def run_on_yarn(experiment_fn: Callable,
                task_specs: Optional[Dict[str, TaskSpec]] = None,
                nb_retries: int = 0,
                custom_task_module: Optional[str] = None,
                **kwargs) -> Optional[Metrics]:
    """Reference ``tensorflow/client.py:17-31``."""
    def monitored_experiment_fn():
        return _add_monitor_to_experiment(experiment_fn())

    return core_client.run_on_yarn(
        monitored_experiment_fn,
        task_specs or single_server_topology(),
        nb_retries=nb_retries,
        custom_task_module=custom_task_module,
        **kwargs)
