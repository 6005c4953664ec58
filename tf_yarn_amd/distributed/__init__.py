from tf_yarn_amd.distributed.client import run_on_yarn
from tf_yarn_amd.distributed.task import TaskParameters, get_task

__all__ = ["run_on_yarn", "TaskParameters", "get_task"]
