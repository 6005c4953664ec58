"""Torch-backed Estimator with the reference's train/eval/checkpoint
contract.

The reference's Experiment path is TF Estimator
(``tensorflow/experiment.py:6-14``, ``evaluator_task.py:38-140``); this
module keeps the load-bearing surface — ``TrainSpec``/``EvalSpec``,
``estimator.train(input_fn)``/``estimator.evaluate(checkpoint_path=...)``,
``model_dir`` with ``model.ckpt-<step>`` files + a ``checkpoint`` state
file + an ``eval/`` event dir the evaluator derives "already evaluated"
steps from — implemented over torch modules (the environment is
PyTorch-ROCm; SURVEY §7 "Keras/Estimator descriptor parity without TF").
"""

from __future__ import annotations

import logging
import os
import re
from typing import Callable, Dict, List, NamedTuple, Optional, Sequence

import torch
from torch import nn

from tf_yarn_amd.utils import tb

logger = logging.getLogger(__name__)

CKPT_RE = re.compile(r".*model\.ckpt-(\d+)$")


class RunConfig(NamedTuple):
    """Subset of tf.estimator.RunConfig the reference exercises."""
    model_dir: Optional[str] = None
    save_checkpoints_steps: int = 1000
    log_step_count_steps: int = 100
    # carrier for device_filters (the reference's stop-barrier scoping,
    # tf_task_common.py:102-118)
    session_config: Optional[dict] = None


class TrainSpec(NamedTuple):
    input_fn: Callable[[], Sequence]
    max_steps: Optional[int] = None


class EvalSpec(NamedTuple):
    input_fn: Callable[[], Sequence]
    steps: Optional[int] = None
    name: str = ""
    exporters: Sequence = ()
    start_delay_secs: int = 0
    throttle_secs: int = 30


class SessionRunHook:
    """Minimal hook protocol (the reference injects StepPerSecondHook /
    EvalMonitorHook, tensorflow/metrics.py:18-71)."""

    def begin(self, estimator: "Estimator") -> None: ...

    def after_step(self, step: int, loss: float,
                   estimator: "Estimator") -> None: ...

    def end(self, estimator: "Estimator") -> None: ...


class Exporter:
    """Exporter parity (reference evaluator_task.py:118-121): exports under
    model_dir/<exporter.name>."""

    def __init__(self, name: str = "export"):
        self.name = name

    def export(self, estimator: "Estimator", export_path: str,
               checkpoint_path: str, eval_result: Dict) -> None:
        os.makedirs(export_path, exist_ok=True)
        model = estimator._build_module()
        state = torch.load(checkpoint_path, map_location="cpu",
                           weights_only=False)
        model.load_state_dict(state["model"])
        torch.save(model.state_dict(),
                   os.path.join(export_path,
                                f"exported-{state['global_step']}.pt"))


def latest_checkpoint(model_dir: str) -> Optional[str]:
    """Resolve the newest model.ckpt-N via the ``checkpoint`` state file,
    falling back to a directory scan."""
    state_file = os.path.join(model_dir, "checkpoint")
    if os.path.exists(state_file):
        with open(state_file) as fd:
            for line in fd:
                m = re.match(r'model_checkpoint_path:\s*"(.*)"', line)
                if m:
                    path = os.path.join(model_dir, m.group(1))
                    if os.path.exists(path):
                        return path
    candidates = list_checkpoints(model_dir)
    return candidates[-1] if candidates else None


def list_checkpoints(model_dir: str) -> List[str]:
    if not os.path.isdir(model_dir):
        return []
    out = []
    for name in os.listdir(model_dir):
        m = CKPT_RE.match(name)
        if m:
            out.append((int(m.group(1)), os.path.join(model_dir, name)))
    return [p for _, p in sorted(out)]


def checkpoint_step(path: str) -> int:
    """Parse N from model.ckpt-N (reference evaluator_task.py:130)."""
    m = CKPT_RE.match(path)
    if not m:
        raise ValueError(f"not a checkpoint path: {path}")
    return int(m.group(1))


class Estimator:
    """train/evaluate driver over a torch module.

    ``module_fn() -> nn.Module``; ``optimizer_fn(params) -> Optimizer``;
    ``loss_fn(outputs, labels) -> scalar``; optional
    ``metrics_fn(outputs, labels) -> dict``.
    ``input_fn() -> iterable of (features, labels)``.
    """

    def __init__(self,
                 module_fn: Callable[[], nn.Module],
                 optimizer_fn: Callable,
                 loss_fn: Callable,
                 model_dir: Optional[str] = None,
                 config: Optional[RunConfig] = None,
                 metrics_fn: Optional[Callable] = None,
                 device: Optional[str] = None,
                 train_step_fn: Optional[Callable] = None):
        self.config = config or RunConfig()
        self.model_dir = model_dir or self.config.model_dir
        self._module_fn = module_fn
        self._optimizer_fn = optimizer_fn
        self._loss_fn = loss_fn
        self._metrics_fn = metrics_fn
        self._train_step_fn = train_step_fn
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu")
        self._module: Optional[nn.Module] = None
        self._optimizer = None
        self.global_step = 0
        # set by the allreduce task to wrap grads sync
        self.grad_sync_hook: Optional[Callable] = None

    # -- internals ----------------------------------------------------------

    def _build_module(self) -> nn.Module:
        return self._module_fn()

    def _ensure_built(self) -> None:
        if self._module is None:
            self._module = self._build_module().to(self.device)
            self._optimizer = self._optimizer_fn(self._module.parameters())
            ckpt = self.latest_checkpoint()
            if ckpt is not None:
                self._restore(ckpt)

    def _restore(self, ckpt_path: str) -> None:
        state = torch.load(ckpt_path, map_location=self.device,
                           weights_only=False)
        self._module.load_state_dict(state["model"])
        if state.get("optimizer") and self._optimizer is not None:
            try:
                self._optimizer.load_state_dict(state["optimizer"])
            except ValueError:
                logger.warning("optimizer state incompatible; reset")
        self.global_step = state["global_step"]
        logger.info("restored %s (step %d)", ckpt_path, self.global_step)

    def save_checkpoint(self) -> Optional[str]:
        if not self.model_dir:
            return None
        os.makedirs(self.model_dir, exist_ok=True)
        name = f"model.ckpt-{self.global_step}"
        path = os.path.join(self.model_dir, name)
        torch.save({
            "model": self._module.state_dict(),
            "optimizer": self._optimizer.state_dict()
            if self._optimizer else {},
            "global_step": self.global_step,
        }, path)
        with open(os.path.join(self.model_dir, "checkpoint"), "w") as fd:
            fd.write(f'model_checkpoint_path: "{name}"\n')
        logger.info("saved %s", path)
        return path

    def latest_checkpoint(self) -> Optional[str]:
        if not self.model_dir:
            return None
        return latest_checkpoint(self.model_dir)

    # -- public API ----------------------------------------------------------

    def train(self, input_fn: Callable, max_steps: Optional[int] = None,
              steps: Optional[int] = None,
              hooks: Sequence[SessionRunHook] = (),
              save_checkpoints: bool = True) -> "Estimator":
        self._ensure_built()
        module, optimizer = self._module, self._optimizer
        module.train()
        for h in hooks:
            h.begin(self)
        if max_steps is not None and self.global_step >= max_steps:
            logger.info("already at max_steps=%d", max_steps)
            return self
        done = False
        start_step = self.global_step
        while not done:
            for batch in input_fn():
                features, labels = batch
                features = _to_device(features, self.device)
                labels = _to_device(labels, self.device)
                if self._train_step_fn is not None:
                    loss = self._train_step_fn(
                        module, optimizer, features, labels)
                else:
                    optimizer.zero_grad()
                    outputs = module(features)
                    loss = self._loss_fn(outputs, labels)
                    loss.backward()
                    if self.grad_sync_hook is not None:
                        self.grad_sync_hook(module)
                    optimizer.step()
                    if hasattr(module, "apply_sparse_updates"):
                        # sink-based embedding grads (CTR models)
                        module.apply_sparse_updates(
                            optimizer.param_groups[0].get("lr", 0.0))
                self.global_step += 1
                lval = float(loss.detach())
                if self.global_step % self.config.log_step_count_steps == 0:
                    logger.info("step %d loss %.5f", self.global_step, lval)
                for h in hooks:
                    h.after_step(self.global_step, lval, self)
                if (save_checkpoints and self.model_dir and
                        self.global_step %
                        self.config.save_checkpoints_steps == 0):
                    self.save_checkpoint()
                if max_steps is not None and self.global_step >= max_steps:
                    done = True
                    break
                if steps is not None and \
                        self.global_step - start_step >= steps:
                    done = True
                    break
            else:
                # epoch exhausted; without step bounds train one pass
                if max_steps is None and steps is None:
                    done = True
            if max_steps is None and steps is None:
                done = True
        if save_checkpoints and self.model_dir:
            self.save_checkpoint()
        for h in hooks:
            h.end(self)
        return self

    @torch.no_grad()
    def evaluate(self, input_fn: Callable, steps: Optional[int] = None,
                 checkpoint_path: Optional[str] = None, name: str = "",
                 hooks: Sequence[SessionRunHook] = ()) -> Dict[str, float]:
        self._ensure_built()
        if checkpoint_path is not None:
            self._restore(checkpoint_path)
        module = self._module
        module.eval()
        for h in hooks:
            h.begin(self)
        total_loss = 0.0
        n_batches = 0
        metric_sums: Dict[str, float] = {}
        for batch in input_fn():
            features, labels = batch
            features = _to_device(features, self.device)
            labels = _to_device(labels, self.device)
            outputs = module(features)
            loss = self._loss_fn(outputs, labels)
            total_loss += float(loss)
            n_batches += 1
            if self._metrics_fn is not None:
                for k, v in self._metrics_fn(outputs, labels).items():
                    metric_sums[k] = metric_sums.get(k, 0.0) + float(v)
            for h in hooks:
                h.after_step(n_batches, float(loss), self)
            if steps is not None and n_batches >= steps:
                break
        for h in hooks:
            h.end(self)
        result = {"loss": total_loss / max(1, n_batches),
                  "global_step": self.global_step}
        for k, v in metric_sums.items():
            result[k] = v / max(1, n_batches)
        self._write_eval_events(result, name)
        module.train()
        return result

    def _write_eval_events(self, result: Dict[str, float],
                           name: str) -> None:
        """Write eval events under model_dir/eval so the evaluator task can
        derive which ckpt steps are done (reference
        evaluator_task.py:134-140)."""
        if not self.model_dir:
            return
        eval_dir = os.path.join(self.model_dir,
                                f"eval_{name}" if name else "eval")
        writer = tb.SummaryWriter(eval_dir)
        for k, v in result.items():
            writer.add_scalar(k, v, step=result["global_step"])
        writer.close()

    def eval_dir(self, name: str = "") -> str:
        return os.path.join(self.model_dir or "",
                            f"eval_{name}" if name else "eval")


def _to_device(x, device):
    if isinstance(x, torch.Tensor):
        return x.to(device)
    if isinstance(x, (list, tuple)):
        return type(x)(_to_device(v, device) for v in x)
    if isinstance(x, dict):
        return {k: _to_device(v, device) for k, v in x.items()}
    return x


def evaluated_steps(eval_dir: str) -> List[int]:
    """Steps already evaluated, derived from the eval event files."""
    steps = set()
    for ev in tb.read_events(eval_dir):
        if ev.get("tag") == "global_step" or ev.get("step") is not None:
            if ev.get("step") is not None:
                steps.add(int(ev["step"]))
    return sorted(steps)


def train_and_evaluate(estimator: Estimator, train_spec: TrainSpec,
                       eval_spec: Optional[EvalSpec] = None) -> None:
    """Single-process train_and_evaluate (in cluster mode the task modules
    drive per-role behavior; reference _independent_workers_task.py)."""
    estimator.train(train_spec.input_fn, max_steps=train_spec.max_steps)
    if eval_spec is not None:
        estimator.evaluate(eval_spec.input_fn, steps=eval_spec.steps,
                           name=eval_spec.name)


class LinearClassifier(Estimator):
    """Premade linear classifier (the reference's
    ``examples/linear_classifier_example.py`` uses
    ``tf.estimator.LinearClassifier`` over winequality)."""

    def __init__(self, n_features: int, n_classes: int = 2,
                 optimizer_fn: Optional[Callable] = None,
                 model_dir: Optional[str] = None,
                 config: Optional[RunConfig] = None,
                 device: Optional[str] = None):
        def module_fn() -> nn.Module:
            return nn.Linear(n_features, n_classes)

        def loss_fn(outputs, labels):
            return nn.functional.cross_entropy(outputs, labels.long())

        def metrics_fn(outputs, labels):
            acc = (outputs.argmax(dim=1) == labels.long()).float().mean()
            return {"accuracy": float(acc)}

        if optimizer_fn is None:
            def optimizer_fn(params):
                return torch.optim.SGD(params, lr=0.1)

        super().__init__(module_fn, optimizer_fn, loss_fn,
                         model_dir=model_dir, config=config,
                         metrics_fn=metrics_fn, device=device)


class DNNClassifier(Estimator):
    """Premade DNN classifier (the reference's Estimator examples use
    ``tf.estimator.DNNClassifier`` over winequality/Criteo tabular data)."""

    def __init__(self, hidden_units: Sequence[int], n_features: int,
                 n_classes: int = 2,
                 optimizer_fn: Optional[Callable] = None,
                 model_dir: Optional[str] = None,
                 config: Optional[RunConfig] = None,
                 device: Optional[str] = None):
        def module_fn() -> nn.Module:
            layers: List[nn.Module] = []
            d = n_features
            for h in hidden_units:
                layers += [nn.Linear(d, h), nn.ReLU()]
                d = h
            layers.append(nn.Linear(d, n_classes))
            return nn.Sequential(*layers)

        def loss_fn(outputs, labels):
            return nn.functional.cross_entropy(outputs, labels.long())

        def metrics_fn(outputs, labels):
            acc = (outputs.argmax(dim=1) == labels.long()).float().mean()
            return {"accuracy": float(acc)}

        if optimizer_fn is None:
            def optimizer_fn(params):
                return torch.optim.Adagrad(params, lr=0.05)

        super().__init__(module_fn, optimizer_fn, loss_fn,
                         model_dir=model_dir, config=config,
                         metrics_fn=metrics_fn, device=device)
