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
