"""Every import shape the reference's examples use must work here with
``tf_yarn`` -> ``tf_yarn_amd`` (collected by grepping the reference's
examples/ tree; see PARITY.md)."""


def test_reference_example_import_shapes():
    from tf_yarn_amd._task_commons import catchtime  # noqa: F401
    from tf_yarn_amd.distributed import client  # noqa: F401
    from tf_yarn_amd.distributed.task import get_task  # noqa: F401
    from tf_yarn_amd.pytorch import (DataLoaderArgs,  # noqa: F401
                                     NodeLabel, PytorchExperiment,
                                     TaskSpec, model_ckpt, run_on_yarn)
    from tf_yarn_amd.tensorflow import Experiment  # noqa: F401
    from tf_yarn_amd.tensorflow import (KerasExperiment,  # noqa: F401
                                        run_on_yarn as tf_run_on_yarn)
    from tf_yarn_amd.topologies import NodeLabel  # noqa: F401, F811
    from tf_yarn_amd.topologies import TaskSpec  # noqa: F401, F811


def test_reference_top_level_exports():
    import tf_yarn_amd as t
    for name in ("RunFailed", "Metrics", "TaskSpec", "NodeLabel",
                 "single_server_topology", "ps_strategy_topology",
                 "get_safe_experiment_fn"):
        assert hasattr(t, name), name


def test_reference_tensorflow_exports():
    import tf_yarn_amd.tensorflow as t
    for name in ("Experiment", "KerasExperiment", "run_on_yarn",
                 "RunFailed", "Metrics", "TaskSpec", "NodeLabel",
                 "single_server_topology", "ps_strategy_topology",
                 "get_safe_experiment_fn"):
        assert hasattr(t, name), name


def test_reference_pytorch_exports():
    import tf_yarn_amd.pytorch as t
    for name in ("PytorchExperiment", "DataLoaderArgs", "run_on_yarn",
                 "RunFailed", "Metrics", "TaskSpec", "NodeLabel",
                 "get_safe_experiment_fn", "DistributedDataParallelArgs"):
        assert hasattr(t, name), name


def test_tensorflow_alias_submodules_importable():
    """custom_task_module strings through the alias package must resolve
    (e.g. "tf_yarn_amd.tensorflow.tasks.evaluator_task")."""
    import importlib
    for name in ("tf_yarn_amd.tensorflow.client",
                 "tf_yarn_amd.tensorflow.cluster",
                 "tf_yarn_amd.tensorflow.metrics",
                 "tf_yarn_amd.tensorflow.tasks.evaluator_task"):
        assert importlib.import_module(name) is not None, name
