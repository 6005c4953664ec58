from tf_yarn_amd.pytorch.client import run_on_yarn
from tf_yarn_amd.pytorch.experiment import (DataLoaderArgs,
                                            DistributedDataParallelArgs,
                                            PytorchExperiment)
from tf_yarn_amd import Metrics, RunFailed, get_safe_experiment_fn
from tf_yarn_amd.topologies import NodeLabel, TaskSpec

__all__ = ["run_on_yarn", "PytorchExperiment", "DataLoaderArgs",
           "DistributedDataParallelArgs", "TaskSpec", "NodeLabel",
           # reference pytorch/__init__.py:11-25 re-exports
           "RunFailed", "Metrics", "get_safe_experiment_fn"]
