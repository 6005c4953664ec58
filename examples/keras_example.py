"""KerasExperiment on the single-server topology (the reference's
``examples/keras_example.py``): a 2-layer MLP on synthetic MNIST —
BASELINE config 1, runs without a GPU.

Run: python examples/keras_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd import TaskSpec
from tf_yarn_amd.estimator import run_on_yarn


def experiment_fn(model_dir: str):
    def make():
        import torch
        from torch import nn

        from tf_yarn_amd.estimator.keras import KerasModel, ModelCheckpoint
        from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
        from tf_yarn_amd.models.mlp import MLP

        torch.manual_seed(0)
        model = KerasModel(MLP(in_dim=784, hidden=(128, 64), n_classes=10))
        # Adadelta: the reference README's Keras optimizer (README.md:106)
        model.compile(optimizer="adadelta",
                      loss="sparse_categorical_crossentropy")

        def input_data_fn():
            torch.manual_seed(1)
            return torch.randn(1024, 784)  # synthetic MNIST

        def target_data_fn():
            torch.manual_seed(2)
            return torch.randint(0, 10, (1024,))

        return KerasExperiment(
            model=model,
            model_dir=model_dir,
            train_params={
                "epochs": 2,
                "batch_size": 64,
                "callbacks": [ModelCheckpoint(
                    os.path.join(model_dir, "checkpoint-{epoch}"))],
            },
            input_data_fn=input_data_fn,
            target_data_fn=target_data_fn,
            validation_data_fn=lambda: (input_data_fn(),
                                        target_data_fn()),
        )
    return make


def main():
    model_dir = os.environ.get("MODEL_DIR", "/tmp/miyarn_keras_example")
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {
            "chief": TaskSpec(memory=1024, vcores=1),
            "worker": TaskSpec(memory=1024, vcores=1, instances=1),
        },
        custom_task_module="tf_yarn_amd.estimator.tasks.allred_task")
    print("run metrics:", metrics)
    assert os.path.exists(os.path.join(model_dir, "checkpoint-1"))


if __name__ == "__main__":
    main()
