#!/usr/bin/env python3
"""PS-strategy throughput (BASELINE config 2): Estimator DNNClassifier on
synthetic Criteo tabular, 1 chief + 1 ps + 2 workers (async push/pull
through the p2p PS engine, tf_yarn_amd/parallel/ps.py).

Run: python scripts/bench_ps.py [--steps N] [--batch B]
Prints one JSON line (same field set as bench.py; metric name marks the
config).  On a CPU-only box this measures the gloo/CPU data plane —
config 2's quoted placement (4 MI355X) is the driver's to run.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

N_FEATURES = 39  # 13 dense + 26 categorical (hashed) — Criteo tabular
HIDDEN = [512, 256, 128]


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
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--batch", type=int, default=2048)
    args = ap.parse_args()

    import tempfile

    from tf_yarn_amd import TaskSpec
    from tf_yarn_amd.topologies import NodeLabel
    from tf_yarn_amd.estimator import run_on_yarn

    # FRESH model dir: a stale one makes estimator.train no-op at
    # global_step >= max_steps and the bench reports garbage.
    model_dir = os.environ.get("MODEL_DIR") or tempfile.mkdtemp(
        prefix="miyarn_bench_ps_")
    # GPU-label the PS world when the box has a GPU (config 2 places all
    # four tasks on MI355X; a 1-GPU box timeshares cuda:0).
    label = (NodeLabel.GPU if torch.cuda.is_available()
             and os.environ.get("MIYARN_PS_CPU", "") != "1"
             else NodeLabel.CPU)
    n_train_tasks = 3  # chief + 2 workers train concurrently (async PS)
    t0 = time.perf_counter()
    metrics = run_on_yarn(
        experiment_fn_factory(model_dir, args.steps, args.batch),
        {
            "chief": TaskSpec(memory=2048, vcores=8, label=label),
            "ps": TaskSpec(memory=2048, vcores=8, instances=1,
                           label=label),
            "worker": TaskSpec(memory=2048, vcores=8, instances=2,
                               label=label),
        })
    wall = time.perf_counter() - t0
    train_s = metrics.total_training_duration or wall
    examples = args.steps * args.batch * n_train_tasks
    n_gpus = torch.cuda.device_count() or 0
    print(json.dumps({
        "metric": "examples/sec, Estimator DNNClassifier PS strategy",
        "value": examples / train_s,
        "unit": "examples/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": 0,
        "ms_per_step": train_s / args.steps * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {"model": "dnn_classifier_criteo_tabular",
                   "topology": "chief+1ps+2workers",
                   "global_batch": args.batch * n_train_tasks,
                   "n_features": N_FEATURES, "hidden": HIDDEN,
                   "parallelism": "ps-async",
                   "wall_s": round(wall, 2)},
    }))


if __name__ == "__main__":
    main()
