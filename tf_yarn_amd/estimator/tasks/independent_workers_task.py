"""Default Estimator task module: ParameterServerStrategy on RCCL p2p.

Parity with reference ``tf_yarn/tensorflow/tasks/_independent_workers_task.py``:
chief/worker train independently against parameter shards held by ps tasks
(async push/pull, no inter-worker barrier); ps tasks serve until every
worker says goodbye; KerasExperiment is rejected on this path
(reference ``:28-29``); the shutdown stop-barrier releases everyone
(``tf_task_common.py:102-107``).

MI355X-native: the TF gRPC server + TF_CONFIG dance
(``tensorflow/cluster.py``) is replaced by a torch process group over the
framework's KV store, per-(worker, ps) RCCL communicators over xGMI and the
fused HIP optimizer apply on the ps shard.
"""

from __future__ import annotations

import logging
import sys
import torch
from torch import nn

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.estimator.estimator import Estimator
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment
from tf_yarn_amd.estimator.metrics import StepPerSecondHook
from tf_yarn_amd.estimator.tasks import task_common
from tf_yarn_amd.parallel import comm, ps as ps_mod

logger = logging.getLogger(__name__)

INIT_SEED = 20240913  # every rank builds the same initial module


def _device_filters(experiment: Experiment, task_type: str):
    """session_config.device_filters determine the stop-barrier scope
    (reference tf_task_common.py:102-118): workers wait for ps+chief only."""
    sc = experiment.config.session_config or {}
    filters = sc.get("device_filters")
    if filters:
        return filters
    if task_type in ("chief", "worker"):
        return ["/job:ps", f"/job:{task_type}"]
    return None


def main() -> None:
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    task_key = _task_commons.get_task_key()
    cluster_tasks = task_common._prepare_container(client)

    event.init_event(client, task, "127.0.0.1:0")
    experiment = _task_commons._get_experiment(client)
    if isinstance(experiment, KerasExperiment):
        raise ValueError(
            "KerasExperiment is not supported on the PS path; use the "
            "allreduce task module (reference "
            "_independent_workers_task.py:28-29)")

    if task_key.type == "evaluator":
        # Side-car continuous evaluator (reference evaluator_task.py)
        from tf_yarn_amd.estimator.tasks.evaluator_task import evaluator_fn
        thread = task_common._execute_dispatched_function(
            client, lambda: evaluator_fn(client, experiment))
        thread.join()
        task_common._shutdown_container(client, cluster_tasks, None,
                                        thread)
        return

    topo = ps_mod.PsTopology(cluster_tasks, task_key.type, task_key.id)
    gpu_ids = [int(x) for x in
               __import__("os").environ.get("MIYARN_GPU_IDS", "").split(",")
               if x]
    device = (f"cuda:{gpu_ids[0]}"
              if gpu_ids and torch.cuda.is_available() else "cpu")
    if device.startswith("cuda"):
        torch.cuda.set_device(torch.device(device))

    estimator: Estimator = experiment.estimator
    estimator.device = device

    if topo.n_ps == 0:
        # Single-server topology (chief trains locally, reference
        # single_server_topology): no PS world to build.
        thread = _run_local(client, experiment, task_key.type)
        thread.join()
        task_common._shutdown_container(
            client, cluster_tasks,
            _device_filters(experiment, task_key.type), thread)
        return

    _task_commons.choose_master(client, topo.rank)
    # RCCL p2p pair groups need one DISTINCT GPU per rank (RCCL refuses
    # duplicate devices in a communicator).  With fewer GPUs than PS
    # ranks, fall back to gloo transport with CPU wire buffers while
    # keeping shards + fused optimizer apply on the GPU (staged mode —
    # also the reference's CPU-ps/GPU-worker deployment shape).
    use_rccl = (device.startswith("cuda")
                and topo.world_size <= torch.cuda.device_count())
    backend = "nccl" if use_rccl else "gloo"
    comm_device = None if (use_rccl or not device.startswith("cuda")) \
        else "cpu"
    comm.init_process_group(rank=topo.rank, world_size=topo.world_size,
                            backend=backend,
                            device=device, kv_client=client,
                            group_name="ps_world", need_subgroups=True)
    pair_groups = ps_mod.build_pair_groups(topo)

    torch.manual_seed(INIT_SEED)
    ref_module = estimator._build_module()
    params = [p for p in ref_module.parameters() if p.requires_grad]
    layout = ps_mod._ShardLayout(params, topo.n_ps)

    thread = None
    try:
        if topo.is_ps:
            _run_ps(client, topo, layout, pair_groups, device, estimator,
                    comm_device)
        else:
            thread = _run_training(client, topo, layout, pair_groups,
                                   device, experiment, task_key.type,
                                   comm_device)
            # ps never joins in the reference (:38-40); training tasks do
            thread.join()
    finally:
        task_common._shutdown_container(
            client, cluster_tasks,
            _device_filters(experiment, task_key.type), thread)


def _run_local(client, experiment: Experiment, task_type: str):
    """Chief-only local training (no ps in the topology)."""
    estimator = experiment.estimator
    torch.manual_seed(INIT_SEED)
    hooks = [StepPerSecondHook()] if task_type == "chief" else []

    def train():
        estimator.train(experiment.train_spec.input_fn,
                        max_steps=experiment.train_spec.max_steps,
                        hooks=hooks)

    return task_common._execute_dispatched_function(client, train)


def _run_ps(client, topo, layout, pair_groups, device,
            estimator: Estimator, comm_device=None) -> None:
    """Serve a parameter shard with the user's optimizer applied to the
    flat shard (fused HIP step on GPU)."""
    shard_index = topo.rank - topo.n_workers
    numel = layout.shard_numel[shard_index]
    shard_param = nn.Parameter(torch.zeros(numel, device=device))
    opt = estimator._optimizer_fn([shard_param])

    def optimizer_step(shard: torch.Tensor, grad: torch.Tensor) -> None:
        shard_param.grad = grad
        opt.step()

    server = ps_mod.PsShardServer(topo, layout, pair_groups, device,
                                  optimizer_step,
                                  comm_device=comm_device)
    server.shard = shard_param.data  # optimizer updates this in place
    server.receive_initial(src_rank=0)
    event.start_event(client, _task_commons.get_task())
    event.broadcast_train_eval_start_timer(client, _task_commons.get_task())
    server.serve()
    event.broadcast_train_eval_stop_timer(client, _task_commons.get_task())


def _run_training(client, topo, layout, pair_groups, device,
                  experiment: Experiment, task_type: str,
                  comm_device=None):
    estimator: Estimator = experiment.estimator
    torch.manual_seed(INIT_SEED)
    estimator._ensure_built()
    params = [p for p in estimator._module.parameters()
              if p.requires_grad]
    channel = ps_mod.PsWorkerChannel(topo, layout, pair_groups, device,
                                     params, comm_device=comm_device)
    if topo.rank == 0:
        channel.send_initial()

    loss_fn = estimator._loss_fn

    def ps_train_step(module, optimizer, features, labels):
        module.zero_grad()
        loss = loss_fn(module(features), labels)
        loss.backward()
        channel.push_pull()  # async PS: push grads, pull fresh weights
        return loss

    estimator._train_step_fn = ps_train_step
    if task_type != "chief":
        estimator.model_dir = None  # only the chief writes checkpoints

    hooks = [StepPerSecondHook()] if task_type == "chief" else []

    def train():
        try:
            estimator.train(experiment.train_spec.input_fn,
                            max_steps=experiment.train_spec.max_steps,
                            hooks=hooks)
            if task_type == "chief" and experiment.eval_spec is not None:
                estimator.evaluate(experiment.eval_spec.input_fn,
                                   steps=experiment.eval_spec.steps,
                                   name=experiment.eval_spec.name)
        finally:
            channel.goodbye()

    return task_common._execute_dispatched_function(client, train)


if __name__ == "__main__":
    try:
        main()
    except Exception:
        logger.exception("task failed")
        sys.exit(1)
