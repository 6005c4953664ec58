"""Task-type -> task-module selection (reference ``tf_yarn/_env.py``)."""

from typing import Optional

INDEPENDENT_WORKERS_MODULE = "tf_yarn_amd.estimator.tasks.independent_workers_task"
TENSORBOARD_MODULE = "tf_yarn_amd.estimator.tasks.tensorboard_task"


def gen_task_cmd(task_type: str,
                 custom_task_module: Optional[str] = None) -> str:
    """Return the python module a task process executes
    (reference ``_env.py:10-24``): tensorboard has a dedicated module,
    everything else runs the (possibly overridden) worker module."""
    if task_type == "tensorboard":
        module = TENSORBOARD_MODULE
    elif custom_task_module:
        module = custom_task_module
    else:
        module = INDEPENDENT_WORKERS_MODULE
    return module
