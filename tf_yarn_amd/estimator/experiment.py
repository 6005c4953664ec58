"""Experiment descriptor (reference ``tf_yarn/tensorflow/experiment.py``)."""

from typing import NamedTuple

from tf_yarn_amd.estimator.estimator import (Estimator, EvalSpec, RunConfig,
                                             TrainSpec)


class Experiment(NamedTuple):
    """Reference ``tensorflow/experiment.py:6-14``."""
    estimator: Estimator
    train_spec: TrainSpec
    eval_spec: EvalSpec

    @property
    def config(self) -> RunConfig:
        return self.estimator.config
