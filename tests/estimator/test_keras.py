"""Keras shim tests (compile/fit/evaluate/save + ModelCheckpoint)."""

import os

import torch
from torch import nn

from tf_yarn_amd.estimator.keras import (KerasModel, ModelCheckpoint,
                                         load_model)


def _model():
    torch.manual_seed(0)
    return KerasModel(nn.Sequential(nn.Linear(4, 8), nn.ReLU(),
                                    nn.Linear(8, 1))).to("cpu")


def _data(n=128):
    torch.manual_seed(1)
    x = torch.randn(n, 4)
    y = x.sum(dim=1, keepdim=True)
    return x, y


def test_fit_reduces_loss():
    model = _model()
    model.compile(optimizer="sgd", loss="mse")
    x, y = _data()
    hist = model.fit(x, y, epochs=5, batch_size=16)
    assert hist["loss"][-1] < hist["loss"][0]


def test_fused_optimizers_by_name():
    for opt in ("adam", "adagrad", "adadelta"):
        model = _model()
        model.compile(optimizer=opt, loss="mse")
        x, y = _data(64)
        model.fit(x, y, epochs=1, batch_size=16)


def test_model_checkpoint_and_load(tmp_path):
    model = _model()
    model.compile(optimizer="sgd", loss="mse")
    x, y = _data(64)
    ckpt_path = str(tmp_path / "checkpoint-{epoch}")
    model.fit(x, y, epochs=2, batch_size=16,
              callbacks=[ModelCheckpoint(ckpt_path)])
    assert os.path.exists(str(tmp_path / "checkpoint-0"))
    assert os.path.exists(str(tmp_path / "checkpoint-1"))
    loaded = load_model(str(tmp_path / "checkpoint-1"))
    loaded.compile(optimizer="sgd", loss="mse")
    # loaded model evaluates the same as the live one
    assert abs(loaded.evaluate(x, y) - model.evaluate(x, y)) < 1e-5


def test_predict_shape():
    model = _model()
    model.compile(optimizer="sgd", loss="mse")
    x, _ = _data(32)
    out = model.predict(x)
    assert out.shape == (32, 1)


def test_evaluate_validation_data():
    model = _model()
    model.compile(optimizer="sgd", loss="mse")
    x, y = _data(64)
    hist = model.fit(x, y, epochs=1, batch_size=16,
                     validation_data=(x, y))
    assert "loss" in hist
