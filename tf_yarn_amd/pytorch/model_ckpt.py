"""Epoch-based PyTorch checkpointing (parity with reference
``tf_yarn/pytorch/model_ckpt.py``): layout ``model_dir/model_{epoch}.pt``
holding ``{'model','optimizer','epoch', **extra}``, resolved through the
``resolve_filesystem_and_path`` seam so the layout code matches the
reference's shape (``model_ckpt.py:15-77``)."""

from __future__ import annotations

import logging
import os
import pickle
import re
import tempfile
from typing import Any, Dict, Optional

import torch

from tf_yarn_amd.utils.fs import resolve_filesystem_and_path

logger = logging.getLogger(__name__)

_CKPT_RE = re.compile(r".*model_(\d+)\.pt$")


def find_latest_ckpt(model_dir: str) -> Optional[str]:
    """Reference ``model_ckpt.py:15-28``."""
    fs, path = resolve_filesystem_and_path(model_dir)
    if not fs.exists(path):
        return None
    best_epoch = -1
    best = None
    for p in fs.ls(path):
        m = _CKPT_RE.match(p)
        if m and int(m.group(1)) > best_epoch:
            best_epoch = int(m.group(1))
            best = p
    return best


def load_latest_ckpt(model_dir: str, model, optimizer=None,
                     device: str = "cpu") -> Optional[Dict[str, Any]]:
    """Reference ``model_ckpt.py:31-39``."""
    ckpt = find_latest_ckpt(model_dir)
    if ckpt is None:
        return None
    return load_ckpt(ckpt, model, optimizer, device)


def load_ckpt(ckpt_path: str, model, optimizer=None,
              device: str = "cpu", weights_only: bool = True) -> Dict[str, Any]:
    """Reference ``model_ckpt.py:42-52``.

    ``weights_only=True`` by default (safe unpickling); pass ``False``
    only for trusted checkpoints carrying arbitrary non-tensor extras.
    """
    fs, path = resolve_filesystem_and_path(ckpt_path)
    with fs.open(path, "rb") as fd:
        try:
            state = torch.load(fd, map_location=device,
                               weights_only=weights_only)
        except pickle.UnpicklingError:
            if weights_only:
                raise RuntimeError(
                    f"{ckpt_path} contains non-tensor objects; reload with "
                    "weights_only=False if you trust its source") from None
            raise
    _unwrap_model(model).load_state_dict(state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    return state


def save_ckpt(model_dir: str, model, optimizer, epoch: int,
              **kwargs: Any) -> str:
    """Reference ``model_ckpt.py:55-73``: torch.save to a tempdir, then
    fs.put to the destination."""
    state = {
        "model": _unwrap_model(model).state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else {},
        "epoch": epoch,
        **kwargs,
    }
    fs, path = resolve_filesystem_and_path(model_dir)
    fs.mkdir(path)
    dest = os.path.join(path, f"model_{epoch}.pt")
    with tempfile.TemporaryDirectory() as tmp:
        local = os.path.join(tmp, f"model_{epoch}.pt")
        torch.save(state, local)
        fs.put(local, dest)
    return dest


def _unwrap_model(model):
    """DDP-unwrap (reference ``model_ckpt.py:76-77``)."""
    return model.module if hasattr(model, "module") else model
