"""Evaluator-task logic tests (reference tests/tensorflow/test_evaluator_task.py)."""

import os

import pytest
import torch
from torch import nn

from tf_yarn_amd.estimator.estimator import checkpoint_step
from tf_yarn_amd.estimator.tasks.evaluator_task import (get_ckpt_to_eval,
                                                        stop_cond_reached)


def _write_ckpt(model_dir, step):
    os.makedirs(model_dir, exist_ok=True)
    path = os.path.join(model_dir, f"model.ckpt-{step}")
    torch.save({"model": {}, "global_step": step}, path)
    return path


@pytest.mark.parametrize("steps,evaluated,expected", [
    ([5, 10, 15], set(), 5),
    ([5, 10, 15], {5}, 10),
    ([5, 10, 15], {5, 10}, 15),
    ([5, 10, 15], {5, 10, 15}, None),
    ([10, 5], {10}, 5),  # oldest unevaluated first
])
def test_get_ckpt_to_eval(tmp_path, steps, evaluated, expected):
    model_dir = str(tmp_path)
    for s in steps:
        _write_ckpt(model_dir, s)
    ckpt = get_ckpt_to_eval(model_dir, evaluated)
    if expected is None:
        assert ckpt is None
    else:
        assert checkpoint_step(ckpt) == expected


@pytest.mark.parametrize("max_steps,evaluated,expected", [
    (None, {5, 10}, False),
    (10, set(), False),
    (10, {5}, False),
    (10, {5, 10}, True),
    (10, {15}, True),
])
def test_stop_cond(max_steps, evaluated, expected):
    assert stop_cond_reached(max_steps, evaluated) is expected


def test_checkpoint_step_parse():
    assert checkpoint_step("/a/b/model.ckpt-1234") == 1234
    with pytest.raises(ValueError):
        checkpoint_step("/a/b/other-12")


def test_keras_evaluate_reloads_each_checkpoint(tmp_path, caplog):
    """Reference evaluator_task.py:54-74: reload the whole Keras model
    per checkpoint-{epoch}, evaluate on validation_data_fn, stop once
    every epoch's checkpoint is evaluated."""
    import logging

    import torch
    from torch import nn

    from tf_yarn_amd.estimator import tasks
    from tf_yarn_amd.estimator.keras import KerasModel
    from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
    from tf_yarn_amd.estimator.tasks import evaluator_task

    model_dir = tmp_path / "m"
    model_dir.mkdir()
    torch.manual_seed(0)
    for epoch in range(2):  # two saved whole-model checkpoints
        m = KerasModel(nn.Linear(4, 1))
        m.compile(optimizer="sgd", loss="mse")
        m.save(str(model_dir / f"checkpoint-{epoch}"))

    def validation_data_fn():
        torch.manual_seed(1)
        x = torch.randn(16, 4)
        return x, x.sum(1, keepdim=True)

    exp = KerasExperiment(
        model=None, model_dir=str(model_dir),
        train_params={"epochs": 2}, input_data_fn=None,
        target_data_fn=None, validation_data_fn=validation_data_fn)
    with caplog.at_level(logging.INFO,
                         logger=evaluator_task.__name__):
        evaluator_task.keras_evaluate(exp)  # returns once both evaluated
    evals = [m for m in caplog.messages if m.startswith("eval ")]
    assert len(evals) == 2
    assert any("checkpoint-0" in m for m in evals)
    assert any("checkpoint-1" in m for m in evals)
