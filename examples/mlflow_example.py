"""MLflow-integrated run (the reference's ``examples/mlflow_example.py``):
the run logs metrics through the mlflow facade; with no mlflow installed or
no tracking URI the facade no-ops and the run still succeeds — the example
asserts the end-to-end behavior either way.

Run: python examples/mlflow_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd import TaskSpec, mlflow
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
        est = DNNClassifier([16], n_features=tabular_data.N_FEATURES,
                            model_dir=model_dir,
                            config=RunConfig(save_checkpoints_steps=20))
        return Experiment(
            est,
            TrainSpec(tabular_data.input_fn(), max_steps=30),
            EvalSpec(tabular_data.input_fn(seed=1), steps=5,
                     throttle_secs=0))
    return make


def main():
    print("mlflow active:", mlflow.use_mlflow)
    mlflow.log_metric("example_started", 1.0)  # no-op without mlflow
    model_dir = os.environ.get("MODEL_DIR", "/tmp/miyarn_mlflow_example")
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {"chief": TaskSpec(memory=1024, vcores=1)})
    assert metrics is not None
    if metrics.total_training_duration is not None:
        mlflow.log_metric("training_duration",
                          metrics.total_training_duration)
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
