"""Framework-agnostic "distributed" flavor (reference
``tf_yarn/distributed/client.py``): the user function receives the task
parameters and manages its own process group; the launcher only provides
rank/master/world_size."""

from typing import Callable, Dict, Optional

from tf_yarn_amd import client as core_client
from tf_yarn_amd.metrics import Metrics
from tf_yarn_amd.topologies import TaskSpec

TASK_MODULE = "tf_yarn_amd.distributed.task"


def run_on_yarn(experiment_fn: Callable,
                task_specs: Dict[str, TaskSpec],
                nb_retries: int = 0,
                **kwargs) -> Optional[Metrics]:
    """Reference ``distributed/client.py:9-20``."""
    return core_client.run_on_yarn(
        experiment_fn,
        task_specs,
        nb_retries=nb_retries,
        custom_task_module=TASK_MODULE,
        **kwargs)
