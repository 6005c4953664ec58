"""PyTorch DDP worker task (container entrypoint).

Parity with reference ``tf_yarn/pytorch/tasks/worker.py``: read cluster
tasks from the KV store, compute ``world_size`` as the sum of per-task
process counts (``worker.py:186``), spawn ``nb_proc_per_worker`` local
processes — one per GPU — elect the master through the KV store
(``worker.py:155``), bring up the process group, wrap the model in the
framework's own reducer and hand everything to the user ``main_fn``
(``worker.py:94-121``).

MI355X-native deltas: GPU ids come from the spawner's static assignment
(``MIYARN_GPU_IDS``) instead of an in-worker round-robin; the process group
is RCCL over xGMI rendezvoused through the control-plane store; DDP is
:class:`~tf_yarn_amd.parallel.ddp.BucketedDataParallel`.
"""

from __future__ import annotations

import dataclasses
import logging
import os
import sys
from typing import List, Optional

import torch
import torch.multiprocessing as mp

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.parallel import comm
from tf_yarn_amd.parallel.ddp import BucketedDataParallel
from tf_yarn_amd.pytorch.experiment import (DataLoaderArgs,
                                            DistributedDataParallelArgs,
                                            PytorchExperiment)
from tf_yarn_amd.topologies import ContainerTask
from tf_yarn_amd.utils import tb
from tf_yarn_amd.utils.fs import resolve_filesystem_and_path

logger = logging.getLogger(__name__)


def _get_gpu_ids() -> List[int]:
    raw = os.environ.get("MIYARN_GPU_IDS", "")
    return [int(x) for x in raw.split(",") if x != ""]


def _get_device(gpu_ids: List[int], local_rank: int) -> str:
    """Reference ``worker.py:162-168`` maps worker_id % n_gpus; here the
    spawner pre-assigned this container's GPUs."""
    if gpu_ids and torch.cuda.is_available():
        return f"cuda:{gpu_ids[local_rank % len(gpu_ids)]}"
    return "cpu"


def _get_collective_ops_backend(device: str,
                                world_size: int = 1) -> str:
    """nccl (=RCCL) only when every process can own a distinct GPU,
    gloo otherwise — the reference's procs-vs-GPUs rule
    (``worker.py:171-174``); RCCL refuses duplicate devices in one
    communicator, so ranks sharing a GPU must stage collectives
    through gloo (compute stays on the GPU)."""
    if device.startswith("cuda") and             world_size <= torch.cuda.device_count():
        return "nccl"
    return "gloo"


def _create_dataloader(dataset,
                       dataloader_args: DataLoaderArgs,
                       rank: int, world_size: int):
    """Reference ``worker.py:50-92``: DistributedSampler for map-style
    datasets; iterable datasets are passed through (they shard by rank
    themselves, like ParquetDataset)."""
    kwargs = dataclasses.asdict(dataloader_args)
    shuffle = kwargs.pop("shuffle")
    if kwargs.get("prefetch_factor") is None:
        kwargs.pop("prefetch_factor")
        if kwargs.get("num_workers", 0) == 0:
            kwargs["pin_memory"] = kwargs.get("pin_memory", False)
    if kwargs.get("batch_size") is None:
        # pre-batched dataset (e.g. ParquetDataset yields whole batches):
        # auto-batching off; drop_last/ sampler options don't apply
        kwargs.pop("drop_last", None)
    if isinstance(dataset, torch.utils.data.IterableDataset):
        return torch.utils.data.DataLoader(dataset, **kwargs)
    sampler = torch.utils.data.distributed.DistributedSampler(
        dataset, num_replicas=world_size, rank=rank, shuffle=shuffle)
    return torch.utils.data.DataLoader(dataset, sampler=sampler, **kwargs)


def _upload_tensorboard_logs(local_dir: str, dest_dir: str,
                             rank: int) -> None:
    """Flush per-worker TB logs to the shared dir
    (reference ``worker.py:145-152``)."""
    try:
        fs, path = resolve_filesystem_and_path(dest_dir)
        fs.put(local_dir, os.path.join(path, f"worker_{rank}"))
    except Exception:
        logger.exception("failed to upload tensorboard logs")


def _train(client: KVClient,
           device: str,
           rank: int,
           world_size: int) -> None:
    """Per-process training body (reference ``worker.py:94-121``)."""
    backend = _get_collective_ops_backend(device, world_size)
    _task_commons.choose_master(client, rank)
    if device.startswith("cuda"):
        torch.cuda.set_device(torch.device(device))
    comm.init_process_group(rank=rank, world_size=world_size,
                            backend=backend, device=device,
                            kv_client=client)
    try:
        # Materialize the experiment AFTER the process group is up, like
        # the reference (worker.py:101-103): datasets that detect
        # rank/world at construction (ParquetDataset) see the live group.
        experiment = _task_commons._get_experiment(client)
        assert isinstance(experiment, PytorchExperiment), type(experiment)
        model = experiment.model.to(device)
        ddp_args = experiment.ddp_args or DistributedDataParallelArgs()
        ddp_model = BucketedDataParallel(
            model,
            broadcast_buffers=ddp_args.broadcast_buffers,
            bucket_cap_mb=ddp_args.bucket_cap_mb,
            find_unused_parameters=ddp_args.find_unused_parameters,
            gradient_as_bucket_view=ddp_args.gradient_as_bucket_view)
        loader = _create_dataloader(
            experiment.train_dataset, experiment.dataloader_args,
            rank, world_size)
        tb_dir = None
        writer = None
        if experiment.tensorboard_hdfs_dir:
            tb_dir = os.path.join(
                os.environ.get("MIYARN_APP_DIR", "/tmp"),
                f"tb_worker_{rank}")
            writer = tb.SummaryWriter(tb_dir)
        experiment.main_fn(ddp_model, loader, device, rank, writer)
        if writer is not None:
            writer.close()
            _upload_tensorboard_logs(
                tb_dir, experiment.tensorboard_hdfs_dir, rank)
    finally:
        comm.destroy_process_group()


def main() -> None:
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    task_key = _task_commons.get_task_key()
    _task_commons._setup_container_logs(client)
    cluster_tasks = _task_commons._get_cluster_tasks(client)
    world_size = _task_commons._compute_world_size(cluster_tasks)
    n_workers = next(
        (t.nb_proc for t in cluster_tasks
         if t.type == task_key.type and t.id == task_key.id), 1)
    gpu_ids = _get_gpu_ids()
    # Rank base: processes of tasks ordered as in cluster_instances.
    rank_base = 0
    for t in cluster_tasks:
        if t.type == task_key.type and t.id == task_key.id:
            break
        rank_base += t.nb_proc
    event.init_event(client, task, "127.0.0.1:0")
    event.start_event(client, task)
    event.broadcast_train_eval_start_timer(client, task)
    exc: Optional[BaseException] = None
    try:
        if n_workers == 1:
            _run_single(client, rank_base, world_size, gpu_ids)
        else:
            mp.start_processes(
                _spawned_entry,
                args=(client.address, rank_base, world_size, gpu_ids),
                nprocs=n_workers,
                start_method="spawn")
    except BaseException as e:  # noqa: BLE001
        exc = e
    event.broadcast_train_eval_stop_timer(client, task)
    event.stop_event(client, task, exc)
    event.broadcast_container_stop_time(client, task)
    if exc is not None:
        logger.error("task %s failed", task, exc_info=exc)
        sys.exit(1)


def _run_single(client: KVClient, rank: int, world_size: int,
                gpu_ids: List[int]) -> None:
    device = _get_device(gpu_ids, 0)
    _train(client, device, rank, world_size)


def _spawned_entry(local_rank: int, kv_addr: str, rank_base: int,
                   world_size: int, gpu_ids: List[int]) -> None:
    _task_commons.setup_logging()
    client = KVClient(kv_addr)
    rank = rank_base + local_rank
    device = _get_device(gpu_ids, local_rank)
    _train(client, device, rank, world_size)


if __name__ == "__main__":
    main()
