"""Keras-style model shim over torch modules.

The reference's Keras path (``keras_experiment.py``,
``native_keras_with_gloo_example.py``) needs ``compile``/``fit``/
``evaluate``/``save`` plus the ``ModelCheckpoint`` callback writing
``checkpoint-{epoch}`` whole-model files reloaded with ``load_model``
(reference ``evaluator_task.py:54-65``).  Implemented over torch with the
framework's fused optimizers where available.
"""

from __future__ import annotations

import logging
import os
from typing import Callable, Dict, List, Optional, Sequence

import torch
from torch import nn

logger = logging.getLogger(__name__)


def _resolve_optimizer(opt, params):
    if isinstance(opt, str):
        name = opt.lower()
        from tf_yarn_amd.ops.optim import (FusedAdadelta, FusedAdagrad,
                                           FusedAdam, FusedSGD)
        table = {
            "sgd": lambda p: FusedSGD(p, lr=0.01),
            "adam": lambda p: FusedAdam(p, lr=1e-3),
            "adagrad": lambda p: FusedAdagrad(p, lr=0.01),
            "adadelta": lambda p: FusedAdadelta(p, lr=1.0),
        }
        if name not in table:
            raise ValueError(f"unknown optimizer {opt!r}")
        return table[name](list(params))
    if callable(opt) and not isinstance(opt, torch.optim.Optimizer):
        return opt(params)
    return opt


def _resolve_loss(loss) -> Callable:
    if callable(loss):
        return loss
    table = {
        "mse": nn.functional.mse_loss,
        "mean_squared_error": nn.functional.mse_loss,
        "binary_crossentropy":
            lambda o, t: nn.functional.binary_cross_entropy_with_logits(
                o.float().squeeze(-1), t.float()),
        "categorical_crossentropy":
            lambda o, t: nn.functional.cross_entropy(o, t.long()),
        "sparse_categorical_crossentropy":
            lambda o, t: nn.functional.cross_entropy(o, t.long()),
    }
    if loss not in table:
        raise ValueError(f"unknown loss {loss!r}")
    return table[loss]


class Callback:
    def on_epoch_end(self, epoch: int, logs: Dict,
                     model: "KerasModel") -> None: ...

    def on_batch_end(self, batch: int, logs: Dict,
                     model: "KerasModel") -> None: ...


class ModelCheckpoint(Callback):
    """Whole-model per-epoch checkpoints ``checkpoint-{epoch}``
    (reference native_keras_with_gloo_example.py:75-77)."""

    def __init__(self, filepath: str):
        self.filepath = filepath  # e.g. "<model_dir>/checkpoint-{epoch}"

    def on_epoch_end(self, epoch: int, logs: Dict,
                     model: "KerasModel") -> None:
        path = self.filepath.format(epoch=epoch)
        model.save(path)


class KerasModel:
    """Keras-surface wrapper around an ``nn.Module``."""

    def __init__(self, module: nn.Module, name: str = "model"):
        self.module = module
        self.name = name
        self.optimizer = None
        self.loss_fn: Optional[Callable] = None
        self.metrics: List[str] = []
        self.stop_training = False
        self._device = "cuda" if torch.cuda.is_available() else "cpu"

    def to(self, device: str) -> "KerasModel":
        self._device = device
        self.module.to(device)
        return self

    def compile(self, optimizer="sgd", loss="mse",
                metrics: Sequence[str] = ()) -> None:
        self.module.to(self._device)
        self.optimizer = _resolve_optimizer(optimizer,
                                            self.module.parameters())
        self.loss_fn = _resolve_loss(loss)
        self.metrics = list(metrics)

    def _batches(self, x, y, batch_size: int, shuffle: bool = True):
        n = x.shape[0]
        idx = torch.randperm(n) if shuffle else torch.arange(n)
        for i in range(0, n - batch_size + 1, batch_size):
            sel = idx[i:i + batch_size]
            yield x[sel], y[sel]

    def fit(self, x=None, y=None, *, epochs: int = 1,
            batch_size: int = 32,
            validation_data=None,
            callbacks: Sequence[Callback] = (),
            steps_per_epoch: Optional[int] = None,
            initial_epoch: int = 0,
            verbose: int = 1) -> Dict[str, List[float]]:
        assert self.optimizer is not None, "call compile() first"
        x = torch.as_tensor(x).to(self._device)
        y = torch.as_tensor(y).to(self._device)
        history: Dict[str, List[float]] = {"loss": []}
        for epoch in range(initial_epoch, epochs):
            if self.stop_training:
                break
            self.module.train()
            epoch_loss, n_b = 0.0, 0
            for b, (bx, by) in enumerate(
                    self._batches(x, y, batch_size)):
                self.optimizer.zero_grad()
                out = self.module(bx)
                loss = self.loss_fn(out, by)
                loss.backward()
                self.optimizer.step()
                self._apply_sparse()
                loss_val = float(loss.detach())
                epoch_loss += loss_val
                n_b += 1
                for cb in callbacks:
                    cb.on_batch_end(b, {"loss": loss_val}, self)
                if steps_per_epoch is not None and n_b >= steps_per_epoch:
                    break
            logs = {"loss": epoch_loss / max(1, n_b)}
            if validation_data is not None:
                vx, vy = validation_data
                logs["val_loss"] = self.evaluate(vx, vy,
                                                 batch_size=batch_size)
            history["loss"].append(logs["loss"])
            if verbose:
                logger.info("epoch %d: %s", epoch, logs)
            for cb in callbacks:
                cb.on_epoch_end(epoch, logs, self)
        return history

    def _apply_sparse(self) -> None:
        """CTR models (WideAndDeep) route embedding grads through a
        sink instead of p.grad; apply the fused sparse update with the
        optimizer's lr (and clear the sink so eval passes don't leak).
        No-op for ordinary modules."""
        m = self.module
        if hasattr(m, "apply_sparse_updates"):
            lr = self.optimizer.param_groups[0].get("lr", 0.0)
            m.apply_sparse_updates(lr)

    @torch.no_grad()
    def evaluate(self, x, y, batch_size: int = 32) -> float:
        self.module.eval()
        x = torch.as_tensor(x).to(self._device)
        y = torch.as_tensor(y).to(self._device)
        total, n_b = 0.0, 0
        for bx, by in self._batches(x, y, batch_size, shuffle=False):
            total += float(self.loss_fn(self.module(bx), by))
            n_b += 1
        self.module.train()
        return total / max(1, n_b)

    @torch.no_grad()
    def predict(self, x, batch_size: int = 256) -> torch.Tensor:
        self.module.eval()
        x = torch.as_tensor(x).to(self._device)
        outs = [self.module(x[i:i + batch_size])
                for i in range(0, x.shape[0], batch_size)]
        self.module.train()
        return torch.cat(outs).cpu()

    def save(self, path: str) -> None:
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        torch.save({"module": self.module, "name": self.name}, path)

    def save_weights(self, path: str) -> None:
        torch.save(self.module.state_dict(), path)

    def load_weights(self, path: str) -> None:
        self.module.load_state_dict(
            torch.load(path, map_location=self._device,
                       weights_only=False))


def load_model(path: str) -> KerasModel:
    """Reference ``evaluator_task.py:65`` (tf.keras.models.load_model)."""
    state = torch.load(path, map_location="cpu", weights_only=False)
    return KerasModel(state["module"], state.get("name", "model"))
