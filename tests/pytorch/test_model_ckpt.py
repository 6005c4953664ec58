"""Checkpoint save/load roundtrip (reference tests/pytorch/test_model_ckpt.py)."""

import os

import torch
from torch import nn

from tf_yarn_amd.pytorch import model_ckpt


def test_save_load_roundtrip(tmp_path):
    model_dir = str(tmp_path / "ckpts")
    model = nn.Linear(4, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.5)
    path = model_ckpt.save_ckpt(model_dir, model, opt, epoch=3, note="hi")
    assert os.path.basename(path) == "model_3.pt"
    assert os.path.exists(path)

    model2 = nn.Linear(4, 2)
    opt2 = torch.optim.SGD(model2.parameters(), lr=0.5)
    state = model_ckpt.load_ckpt(path, model2, opt2)
    assert state["epoch"] == 3
    assert state["note"] == "hi"
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)


def test_find_latest_ckpt(tmp_path):
    model_dir = str(tmp_path)
    assert model_ckpt.find_latest_ckpt(model_dir) is None
    model = nn.Linear(2, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    model_ckpt.save_ckpt(model_dir, model, opt, epoch=1)
    model_ckpt.save_ckpt(model_dir, model, opt, epoch=10)
    model_ckpt.save_ckpt(model_dir, model, opt, epoch=2)
    latest = model_ckpt.find_latest_ckpt(model_dir)
    assert latest.endswith("model_10.pt")


def test_load_latest_ckpt(tmp_path):
    model_dir = str(tmp_path)
    assert model_ckpt.load_latest_ckpt(model_dir, nn.Linear(2, 2)) is None
    model = nn.Linear(2, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    model_ckpt.save_ckpt(model_dir, model, opt, epoch=7)
    state = model_ckpt.load_latest_ckpt(model_dir, nn.Linear(2, 2), opt)
    assert state["epoch"] == 7


def test_unwrap_ddp_like():
    class Wrap:
        def __init__(self, m):
            self.module = m

    m = nn.Linear(2, 2)
    assert model_ckpt._unwrap_model(Wrap(m)) is m
    assert model_ckpt._unwrap_model(m) is m
