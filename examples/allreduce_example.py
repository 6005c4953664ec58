"""Estimator training on the ring-allreduce path (the reference's
``examples/collective_all_reduce_example.py``: Horovod-gloo →
here fused-bucket RCCL/gloo allreduce with rank-0 broadcast).

Run: python examples/allreduce_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd import TaskSpec
from tf_yarn_amd.estimator import run_on_yarn

EXAMPLES_DIR = os.path.dirname(os.path.abspath(__file__))


def experiment_fn(model_dir: str):
    def make():
        import sys
        if EXAMPLES_DIR not in sys.path:
            sys.path.insert(0, EXAMPLES_DIR)
        import tabular_data
        from tf_yarn_amd.estimator import (DNNClassifier, EvalSpec,
                                           RunConfig, TrainSpec)
        from tf_yarn_amd.estimator.experiment import Experiment
        est = DNNClassifier([32, 16], n_features=tabular_data.N_FEATURES,
                            model_dir=model_dir,
                            config=RunConfig(save_checkpoints_steps=25))
        return Experiment(
            est,
            TrainSpec(tabular_data.input_fn(), max_steps=50),
            EvalSpec(tabular_data.input_fn(seed=1), steps=5,
                     throttle_secs=0))
    return make


def main():
    model_dir = os.environ.get("MODEL_DIR", "/tmp/miyarn_allred_example")
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {
            "chief": TaskSpec(memory=1024, vcores=1),
            "worker": TaskSpec(memory=1024, vcores=1, instances=1),
        },
        custom_task_module="tf_yarn_amd.estimator.tasks.allred_task")
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
