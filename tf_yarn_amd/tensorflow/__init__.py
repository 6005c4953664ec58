"""Import-path compatibility alias for tf-yarn users.

The reference exposes its Estimator/Keras flavor under
``tf_yarn.tensorflow``; this framework's torch-backed implementation lives
in :mod:`tf_yarn_amd.estimator`.  A user switching over can keep their
import shape::

    from tf_yarn_amd.tensorflow import run_on_yarn, Experiment

Submodules (``client``, ``experiment``, ``keras_experiment``, ``metrics``,
``cluster``, ``tasks``) alias the estimator package's modules 1:1.
"""

import sys

from tf_yarn_amd import (Metrics, NodeLabel, RunFailed, TaskSpec,
                         get_safe_experiment_fn, ps_strategy_topology,
                         single_server_topology)
from tf_yarn_amd.estimator import (DNNClassifier, Estimator, EvalSpec,
                                   Experiment, KerasExperiment, KerasModel,
                                   LinearClassifier, ModelCheckpoint,
                                   RunConfig, TrainSpec, load_model,
                                   run_on_yarn, train_and_evaluate)
from tf_yarn_amd.estimator import client, cluster, estimator, experiment
from tf_yarn_amd.estimator import keras_experiment, metrics, tasks

# module-path aliases so "import tf_yarn_amd.tensorflow.client" etc. work
for _name, _mod in [("client", client), ("cluster", cluster),
                    ("experiment", experiment),
                    ("keras_experiment", keras_experiment),
                    ("metrics", metrics), ("tasks", tasks),
                    ("estimator", estimator)]:
    sys.modules[f"{__name__}.{_name}"] = _mod

__all__ = ["run_on_yarn", "Experiment", "KerasExperiment", "Estimator",
           "DNNClassifier", "LinearClassifier", "TrainSpec", "EvalSpec",
           "RunConfig", "train_and_evaluate", "KerasModel",
           "ModelCheckpoint", "load_model",
           # reference tensorflow/__init__.py:1-15 re-exports
           "RunFailed", "Metrics", "TaskSpec", "NodeLabel",
           "single_server_topology", "ps_strategy_topology",
           "get_safe_experiment_fn"]
