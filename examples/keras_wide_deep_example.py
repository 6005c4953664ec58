"""Keras wide-and-deep on synthetic Criteo-shaped data through the
ring-allreduce task (BASELINE config 3's API flavor): a
``KerasExperiment`` whose model wraps the CTR WideAndDeep network, run
with the Horovod-style path (``hvd.DistributedOptimizer`` + rank-0
broadcast inside ``allred_task``) — the reference's
``native_keras_with_gloo_example.py`` flow on RCCL/gloo.

Run (CPU plumbing): python examples/keras_wide_deep_example.py
Run (GPU):          USE_GPU=1 N_GPUS=8 python examples/keras_wide_deep_example.py

Note: the raw-performance flagship for this model is ``bench.py``
(sharded embeddings + custom reducer); this example demonstrates the
same model through the reference-compatible Keras API.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd import NodeLabel, TaskSpec
from tf_yarn_amd.estimator import run_on_yarn

N_DENSE, N_SPARSE = 13, 26
TABLE_ROWS = 1000


def experiment_fn(model_dir: str):
    def make():
        import torch

        from tf_yarn_amd.estimator.keras import KerasModel, ModelCheckpoint
        from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
        from tf_yarn_amd.models.wide_deep import FlatInputWideAndDeep

        torch.manual_seed(0)
        model = KerasModel(FlatInputWideAndDeep(
            dense_dim=N_DENSE, table_sizes=[TABLE_ROWS] * N_SPARSE,
            embedding_dim=8, hidden=(64, 32)))
        model.compile(optimizer="sgd", loss="binary_crossentropy")

        def input_data_fn():
            g = torch.Generator().manual_seed(1)
            dense = torch.randn(2048, N_DENSE, generator=g)
            ids = torch.randint(0, TABLE_ROWS, (2048, N_SPARSE),
                                generator=g).float()
            return torch.cat([dense, ids], dim=1)

        def target_data_fn():
            g = torch.Generator().manual_seed(2)
            return (torch.rand(2048, generator=g) < 0.3).float()

        return KerasExperiment(
            model=model,
            model_dir=model_dir,
            train_params={
                "epochs": 2,
                "batch_size": 256,
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
    model_dir = os.environ.get("MODEL_DIR",
                               "/tmp/miyarn_keras_wide_deep_example")
    use_gpu = os.environ.get("USE_GPU") == "1"
    n_workers = int(os.environ.get("N_GPUS", "2")) - 1 if use_gpu else 1
    label = NodeLabel.GPU if use_gpu else NodeLabel.CPU
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {
            "chief": TaskSpec(memory=2048, vcores=4, label=label),
            "worker": TaskSpec(memory=2048, vcores=4,
                               instances=max(1, n_workers), label=label),
        },
        custom_task_module="tf_yarn_amd.estimator.tasks.allred_task",
    )
    print("training wall time:", metrics.total_training_duration)


if __name__ == "__main__":
    main()
