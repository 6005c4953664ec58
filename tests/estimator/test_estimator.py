"""Estimator driver tests: train/eval/checkpoint layout parity."""

import os

import pytest
import torch
from torch import nn

from tf_yarn_amd.estimator.estimator import (DNNClassifier, Estimator,
                                             EvalSpec, RunConfig, TrainSpec,
                                             checkpoint_step,
                                             evaluated_steps,
                                             latest_checkpoint,
                                             list_checkpoints,
                                             train_and_evaluate)


def _make_estimator(model_dir, save_every=5):
    def module_fn():
        torch.manual_seed(0)
        return nn.Linear(4, 2)

    def optimizer_fn(params):
        return torch.optim.SGD(params, lr=0.1)

    def loss_fn(out, labels):
        return nn.functional.cross_entropy(out, labels.long())

    return Estimator(module_fn, optimizer_fn, loss_fn,
                     model_dir=model_dir,
                     config=RunConfig(save_checkpoints_steps=save_every),
                     device="cpu")


def _input_fn():
    torch.manual_seed(1)
    for _ in range(10):
        yield torch.randn(8, 4), torch.randint(0, 2, (8,))


def test_train_creates_ckpt_layout(tmp_path):
    model_dir = str(tmp_path / "m")
    est = _make_estimator(model_dir)
    est.train(_input_fn, max_steps=10)
    assert est.global_step == 10
    # model.ckpt-N files + checkpoint state file
    ckpts = list_checkpoints(model_dir)
    assert any(p.endswith("model.ckpt-5") for p in ckpts)
    assert any(p.endswith("model.ckpt-10") for p in ckpts)
    assert os.path.exists(os.path.join(model_dir, "checkpoint"))
    assert latest_checkpoint(model_dir).endswith("model.ckpt-10")
    assert checkpoint_step(latest_checkpoint(model_dir)) == 10


def test_train_resumes_from_ckpt(tmp_path):
    model_dir = str(tmp_path / "m")
    est = _make_estimator(model_dir)
    est.train(_input_fn, max_steps=10)
    w10 = est._module.weight.detach().clone()
    # fresh estimator resumes at step 10
    est2 = _make_estimator(model_dir)
    est2.train(_input_fn, max_steps=10)  # already done: no-op
    assert est2.global_step == 10
    assert torch.allclose(est2._module.weight.detach(), w10)
    est2.train(_input_fn, max_steps=15)
    assert est2.global_step == 15


def test_evaluate_writes_eval_events(tmp_path):
    model_dir = str(tmp_path / "m")
    est = _make_estimator(model_dir)
    est.train(_input_fn, max_steps=5)
    result = est.evaluate(_input_fn, steps=3)
    assert "loss" in result and result["global_step"] == 5
    assert evaluated_steps(est.eval_dir()) == [5]


def test_evaluate_specific_checkpoint(tmp_path):
    model_dir = str(tmp_path / "m")
    est = _make_estimator(model_dir)
    est.train(_input_fn, max_steps=10)
    ckpt5 = os.path.join(model_dir, "model.ckpt-5")
    result = est.evaluate(_input_fn, steps=2, checkpoint_path=ckpt5)
    assert result["global_step"] == 5


def test_dnn_classifier_learns(tmp_path):
    est = DNNClassifier([16, 8], n_features=4, n_classes=2,
                        model_dir=str(tmp_path / "dnn"), device="cpu")

    def input_fn():
        torch.manual_seed(2)
        for _ in range(20):
            x = torch.randn(32, 4)
            y = (x.sum(dim=1) > 0).long()
            yield x, y

    est.train(input_fn, max_steps=60)
    result = est.evaluate(input_fn, steps=10)
    assert result["accuracy"] > 0.7


def test_train_and_evaluate(tmp_path):
    est = _make_estimator(str(tmp_path / "m"))
    train_and_evaluate(est, TrainSpec(_input_fn, max_steps=5),
                       EvalSpec(_input_fn, steps=2))
    assert est.global_step == 5
    assert evaluated_steps(est.eval_dir()) == [5]


def test_get_all_metrics(tmp_path):
    from tf_yarn_amd.estimator.metrics import get_all_metrics
    from tf_yarn_amd.utils import tb
    w = tb.SummaryWriter(str(tmp_path))
    w.add_scalar("loss", 0.5, step=1)
    w.add_scalar("loss", 0.25, step=2)
    w.close()
    m = get_all_metrics(str(tmp_path))
    assert m["name"] == ["loss", "loss"]
    assert m["value"] == [0.5, 0.25]
    assert m["step"] == [1, 2]


def test_exporter_writes_export(tmp_path):
    from tf_yarn_amd.estimator.estimator import Exporter
    est = _make_estimator(str(tmp_path / "m"))
    est.train(_input_fn, max_steps=5)
    exporter = Exporter(name="best")
    export_path = os.path.join(str(tmp_path / "m"), exporter.name)
    ckpt = os.path.join(str(tmp_path / "m"), "model.ckpt-5")
    exporter.export(est, export_path, ckpt, {"loss": 0.1})
    assert os.path.exists(os.path.join(export_path, "exported-5.pt"))


def test_eval_monitor_hook_publishes_stats(tmp_path):
    from tf_yarn_amd.estimator.metrics import EvalMonitorHook
    from tf_yarn_amd.kv import KVClient, KVServer
    server = KVServer()
    try:
        client = KVClient(server.address)
        est = _make_estimator(str(tmp_path / "m"))
        est.train(_input_fn, max_steps=3)
        hook = EvalMonitorHook(client)
        hook._task = "evaluator:0"
        est.evaluate(_input_fn, steps=2, hooks=[hook])
        stats = client.list("evaluator:0/")
        assert "evaluator:0/nb_eval_steps" in stats
        assert stats["evaluator:0/nb_eval_steps"] == b"2"
        assert "evaluator:0/last_training_step" in stats
        assert stats["evaluator:0/last_training_step"] == b"3"
    finally:
        server.stop()


def test_step_per_second_hook_counts(tmp_path):
    from tf_yarn_amd.estimator.metrics import StepPerSecondHook
    est = _make_estimator(str(tmp_path / "m"))
    hook = StepPerSecondHook(every_n_steps=2)
    est.train(_input_fn, max_steps=6, hooks=[hook])
    assert est.global_step == 6


def test_linear_classifier_train_eval(tmp_path):
    """Premade LinearClassifier (reference linear_classifier_example.py
    family): learns a separable problem, writes model.ckpt-N."""
    import torch

    from tf_yarn_amd.estimator import LinearClassifier, RunConfig

    est = LinearClassifier(
        n_features=6, model_dir=str(tmp_path / "m"),
        config=RunConfig(save_checkpoints_steps=20))

    def input_fn():
        torch.manual_seed(4)
        w = torch.randn(6)
        for _ in range(40):
            x = torch.randn(64, 6)
            yield x, ((x @ w) > 0).long()

    est.train(input_fn, max_steps=80)
    result = est.evaluate(input_fn, steps=10)
    assert result["accuracy"] > 0.9, result
    assert est.latest_checkpoint().endswith("model.ckpt-80")
