"""PyTorch-flavor ``run_on_yarn`` (reference ``tf_yarn/pytorch/client.py``):
injects the DDP worker task module and delegates to the core spawner."""

from typing import Callable, Dict, Optional

from tf_yarn_amd import client as core_client
from tf_yarn_amd.metrics import Metrics
from tf_yarn_amd.pytorch.experiment import PytorchExperiment
from tf_yarn_amd.topologies import TaskSpec

TASK_MODULE = "tf_yarn_amd.pytorch.tasks.worker"


def run_on_yarn(experiment_fn: Callable[[], PytorchExperiment],
                task_specs: Dict[str, TaskSpec],
                nb_retries: int = 0,
                custom_task_module: Optional[str] = TASK_MODULE,
                **kwargs) -> Optional[Metrics]:
    """Reference ``pytorch/client.py:12-23``."""
    return core_client.run_on_yarn(
        experiment_fn,
        task_specs,
        nb_retries=nb_retries,
        custom_task_module=custom_task_module or TASK_MODULE,
        **kwargs)
