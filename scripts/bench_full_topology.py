#!/usr/bin/env python3
"""BASELINE config 5: the full topology — chief + 6 workers (ring
allreduce) + evaluator + tensorboard co-task, with MLflow tracking
through the optional shim (no-ops cleanly when mlflow is absent, the
reference's contract).  Exercises topologies.py validation, the
evaluator checkpoint-scan loop, the tensorboard URL event and the
lifecycle-metrics aggregation end to end, and prints the same JSON
line shape as the other benches.

Run: python scripts/bench_full_topology.py [--steps N] [--batch B]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

N_FEATURES = 39
HIDDEN = [256, 128]


def experiment_fn_factory(model_dir: str, steps: int, batch: int):
    def make():
        from tf_yarn_amd.estimator import (DNNClassifier, EvalSpec,
                                           RunConfig, TrainSpec)
        from tf_yarn_amd.estimator.experiment import Experiment

        est = DNNClassifier(
            HIDDEN, n_features=N_FEATURES, model_dir=model_dir,
            config=RunConfig(save_checkpoints_steps=max(steps // 2, 1)))

        def input_fn():
            gen = torch.Generator().manual_seed(
                int(os.environ.get("MIYARN_CONTAINER_ID", "worker_0")
                    .split("_")[-1]) + 1)
            w = torch.randn(N_FEATURES, generator=gen)
            for _ in range(steps + 1):
                x = torch.randn(batch, N_FEATURES, generator=gen)
                y = ((x @ w) > 0).long()
                yield x, y

        return Experiment(
            est,
            TrainSpec(input_fn, max_steps=steps),
            EvalSpec(input_fn, steps=4, throttle_secs=0))
    return make


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=120)
    ap.add_argument("--batch", type=int, default=1024)
    args = ap.parse_args()

    from tf_yarn_amd import TaskSpec
    from tf_yarn_amd.estimator import run_on_yarn
    from tf_yarn_amd.topologies import NodeLabel

    model_dir = os.environ.get("MODEL_DIR") or tempfile.mkdtemp(
        prefix="miyarn_bench_full_")
    label = (NodeLabel.GPU if torch.cuda.is_available() else NodeLabel.CPU)
    n_train = 7  # chief + 6 workers
    t0 = time.perf_counter()
    metrics = run_on_yarn(
        experiment_fn_factory(model_dir, args.steps, args.batch),
        {
            "chief": TaskSpec(memory=1024, vcores=4, label=label),
            "worker": TaskSpec(memory=1024, vcores=4, instances=6,
                               label=label),
            "evaluator": TaskSpec(memory=1024, vcores=2),
            "tensorboard": TaskSpec(memory=512, vcores=1,
                                    tb_termination_timeout_seconds=1,
                                    tb_model_dir=model_dir),
        },
        custom_task_module="tf_yarn_amd.estimator.tasks.allred_task",
    )
    wall = time.perf_counter() - t0
    train_s = metrics.total_training_duration or wall
    examples = args.steps * args.batch * n_train
    print(json.dumps({
        "metric": "examples/sec, full topology (chief+6w+evaluator+tb)",
        "value": examples / train_s,
        "unit": "examples/s",
        "n_gpus": torch.cuda.device_count() or 0,
        "steps": args.steps,
        "warmup": 0,
        "ms_per_step": train_s / args.steps * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {"model": "dnn_classifier_criteo_tabular",
                   "topology": "chief+6workers+evaluator+tensorboard",
                   "global_batch": args.batch * n_train,
                   "parallelism": "ring-allreduce",
                   "mlflow": __import__("tf_yarn_amd.mlflow", fromlist=["use_mlflow"]).use_mlflow,
                   "wall_s": round(wall, 2)},
    }))


if __name__ == "__main__":
    main()
