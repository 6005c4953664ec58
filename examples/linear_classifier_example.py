"""LinearClassifier on the PS-strategy topology (the reference's
``examples/linear_classifier_example.py``: premade LinearClassifier over
winequality-shaped tabular data).

Run: python examples/linear_classifier_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from tf_yarn_amd import TaskSpec
from tf_yarn_amd.estimator import run_on_yarn

EXAMPLES_DIR = os.path.dirname(os.path.abspath(__file__))


def experiment_fn(model_dir: str):
    def make():
        import sys
        if EXAMPLES_DIR not in sys.path:
            sys.path.insert(0, EXAMPLES_DIR)
        import tabular_data
        from tf_yarn_amd.estimator import (EvalSpec, LinearClassifier,
                                           RunConfig, TrainSpec)
        from tf_yarn_amd.estimator.experiment import Experiment
        est = LinearClassifier(n_features=tabular_data.N_FEATURES,
                               model_dir=model_dir,
                               config=RunConfig(save_checkpoints_steps=20))
        return Experiment(
            est,
            TrainSpec(tabular_data.input_fn(), max_steps=60),
            EvalSpec(tabular_data.input_fn(seed=1), steps=10,
                     throttle_secs=0))
    return make


def main():
    model_dir = os.environ.get("MODEL_DIR", "/tmp/miyarn_linear_example")
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {
            "chief": TaskSpec(memory=1024, vcores=1),
            "ps": TaskSpec(memory=1024, vcores=1, instances=1),
            "worker": TaskSpec(memory=1024, vcores=1, instances=1),
        })
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
