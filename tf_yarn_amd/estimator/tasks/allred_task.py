"""Ring-allreduce Estimator/Keras task module.

Parity with reference ``tf_yarn/tensorflow/tasks/gloo_allred_task.py``
(selected via ``custom_task_module="tf_yarn_amd.estimator.tasks.allred_task"``):
synchronous data-parallel training where the chief doubles as rendezvous
driver.  The Horovod-gloo machinery (RendezvousServer, HOROVOD_* env,
per-tensor gloo allreduce) is replaced by the framework's KV rendezvous +
fused-bucket RCCL ring allreduce over xGMI
(:mod:`tf_yarn_amd.parallel.hvd`); non-chief workers strip
model_dir/checkpointing so only the chief writes (reference
``gloo_allred_task.py:59-83``)."""

from __future__ import annotations

import logging
import os
import sys
import torch

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.estimator.estimator import Estimator
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
from tf_yarn_amd.estimator.metrics import StepPerSecondHook
from tf_yarn_amd.estimator.tasks import task_common
from tf_yarn_amd.estimator.tasks.evaluator_task import evaluator_fn
from tf_yarn_amd.parallel import comm, hvd

logger = logging.getLogger(__name__)

INIT_SEED = 20240913


def _worker_fn(client, cluster_tasks, experiment, task_type: str,
               rank: int, world_size: int, device: str) -> None:
    """Reference ``gloo_allred_task.py:36-92``."""
    # RCCL needs a DISTINCT GPU per rank (same rule as the reference's
    # worker backend pick, ``pytorch/tasks/worker.py:171-174``): when
    # ranks share a device (e.g. a 1-GPU box running a multi-task
    # topology), fall back to gloo — compute stays on the GPU, only the
    # collectives stage through CPU.
    use_rccl = (device.startswith("cuda")
                and world_size <= torch.cuda.device_count())
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend="nccl" if use_rccl else "gloo",
                            device=device, kv_client=client,
                            group_name="allred")
    try:
        if isinstance(experiment, KerasExperiment):
            _keras_fit(experiment, task_type, rank)
        else:
            _estimator_train(experiment, task_type, rank)
    finally:
        comm.destroy_process_group()


def _estimator_train(experiment: Experiment, task_type: str,
                     rank: int) -> None:
    estimator: Estimator = experiment.estimator
    torch.manual_seed(INIT_SEED + 0)  # equal init then rank-0 broadcast
    estimator._ensure_built()
    hvd.broadcast_parameters(estimator._module, root_rank=0)

    def sync(module):
        grads = [p.grad for p in module.parameters()
                 if p.grad is not None
                 and not getattr(p, "_miyarn_sparse", False)]
        hvd.allreduce_tensors(grads, average=True)

    estimator.grad_sync_hook = sync
    if task_type != "chief":
        estimator.model_dir = None  # only chief checkpoints (:59-68)
    hooks = [StepPerSecondHook()] if task_type == "chief" else []
    estimator.train(experiment.train_spec.input_fn,
                    max_steps=experiment.train_spec.max_steps,
                    hooks=hooks)


def _keras_fit(experiment: KerasExperiment, task_type: str,
               rank: int) -> None:
    """Keras path: DistributedOptimizer + rank-0 broadcast callback
    (reference ``gloo_allred_task.py:77-89``)."""
    model = experiment.model
    torch.manual_seed(INIT_SEED)
    params = dict(experiment.train_params)
    if task_type != "chief":
        # drop ModelCheckpoint callbacks so only chief writes (:77-83)
        from tf_yarn_amd.estimator.keras import ModelCheckpoint
        params["callbacks"] = [
            cb for cb in params.get("callbacks", [])
            if not isinstance(cb, ModelCheckpoint)]
    if model.optimizer is None:
        raise ValueError("KerasExperiment model must be compiled")
    hvd.broadcast_parameters(model.module, root_rank=0)
    model.optimizer = hvd.DistributedOptimizer(model.optimizer)
    x = experiment.input_data_fn() if experiment.input_data_fn else None
    y = experiment.target_data_fn() if experiment.target_data_fn else None
    model.fit(x, y, **params)


def main() -> None:
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    task_key = _task_commons.get_task_key()
    cluster_tasks = task_common._prepare_container(client)
    event.init_event(client, task, "127.0.0.1:0")
    experiment = _task_commons._get_experiment(client)

    training = [t for t in cluster_tasks if t.type in ("chief", "worker")]
    world_size = sum(t.nb_proc for t in training)
    rank = 0
    for t in training:
        if t.type == task_key.type and t.id == task_key.id:
            break
        rank += t.nb_proc

    gpu_ids = [int(x) for x in os.environ.get("MIYARN_GPU_IDS",
                                              "").split(",") if x]
    device = (f"cuda:{gpu_ids[0]}"
              if gpu_ids and torch.cuda.is_available() else "cpu")
    if device.startswith("cuda"):
        torch.cuda.set_device(torch.device(device))

    thread = None
    try:
        if task_key.type in ("chief", "worker"):
            # chief doubles as rendezvous driver: rank 0 elects the master
            # through the KV store (the RendezvousServer equivalent,
            # gloo_allred_task.py:94-123)
            _task_commons.choose_master(client, rank)

            def run():
                _worker_fn(client, cluster_tasks, experiment,
                           task_key.type, rank, world_size, device)

            thread = task_common._execute_dispatched_function(client, run)
            thread.join()
        elif task_key.type == "evaluator":
            thread = task_common._execute_dispatched_function(
                client, lambda: evaluator_fn(client, experiment))
            thread.join()
        else:
            raise ValueError(f"unexpected task type {task_key.type}")
    finally:
        task_common._shutdown_container(client, cluster_tasks, None,
                                        thread)


if __name__ == "__main__":
    try:
        main()
    except Exception:
        logger.exception("task failed")
        sys.exit(1)
