"""Task module for the framework-agnostic flavor (reference
``tf_yarn/distributed/task.py``): spawn one process per local rank, give
the user fn a :class:`TaskParameters` (task_type, rank, world_size,
master address/port, n_workers_per_executor) and let it drive everything."""

from __future__ import annotations

import logging
import os
import sys
from typing import List, NamedTuple, Optional

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.kv import KVClient

logger = logging.getLogger(__name__)


class TaskParameters(NamedTuple):
    """Reference ``distributed/task.py:37-55``."""
    task_type: str
    rank: int
    world_size: int
    master_address: str
    master_port: int
    n_workers_per_executor: int
    gpu_id: Optional[int] = None


def get_task(client: KVClient, local_rank: int = 0) -> TaskParameters:
    """Compute this process's task parameters.  ``world_size`` counts only
    tasks whose type contains "worker" plus the chief (reference
    ``distributed/task.py:49`` counts "worker"-named tasks)."""
    task_key = _task_commons.get_task_key()
    cluster_tasks = _task_commons._get_cluster_tasks(client)
    training = [t for t in cluster_tasks
                if "worker" in t.type or t.type == "chief"]
    world_size = sum(t.nb_proc for t in training)
    rank_base = 0
    for t in training:
        if t.type == task_key.type and t.id == task_key.id:
            break
        rank_base += t.nb_proc
    rank = rank_base + local_rank
    addr = _task_commons.choose_master(client, rank)
    host, port = addr.rsplit(":", 1)
    n_proc = next((t.nb_proc for t in training
                   if t.type == task_key.type and t.id == task_key.id), 1)
    gpu_ids = [int(x) for x in
               os.environ.get("MIYARN_GPU_IDS", "").split(",") if x]
    gpu = gpu_ids[local_rank % len(gpu_ids)] if gpu_ids else None
    return TaskParameters(task_key.type, rank, world_size, host, int(port),
                          n_proc, gpu)


def _parallel_run(fn, client: KVClient, n_proc: int) -> None:
    """Reference ``distributed/task.py:63-79``."""
    if n_proc == 1:
        fn(get_task(client, 0))
        return
    import torch.multiprocessing as mp
    mp.start_processes(
        _spawned, args=(client.address,), nprocs=n_proc,
        start_method="spawn")


def _spawned(local_rank: int, kv_addr: str) -> None:
    _task_commons.setup_logging()
    client = KVClient(kv_addr)
    fn = _get_fn(client)
    fn(get_task(client, local_rank))


def _get_fn(client: KVClient):
    import cloudpickle
    from tf_yarn_amd import constants
    blob = client.wait(constants.KV_EXPERIMENT_FN)
    return cloudpickle.loads(blob)


def main() -> None:
    """Reference ``distributed/task.py:81-98``."""
    _task_commons.setup_logging()
    client = _task_commons.get_client()
    task = _task_commons.get_task()
    task_key = _task_commons.get_task_key()
    _task_commons._setup_container_logs(client)
    cluster_tasks = _task_commons._get_cluster_tasks(client)
    n_proc = next((t.nb_proc for t in cluster_tasks
                   if t.type == task_key.type and t.id == task_key.id), 1)
    event.init_event(client, task, "127.0.0.1:0")
    event.start_event(client, task)
    event.broadcast_train_eval_start_timer(client, task)
    exc: Optional[BaseException] = None
    try:
        fn = _get_fn(client)
        _parallel_run(fn, client, n_proc)
    except BaseException as e:  # noqa: BLE001
        exc = e
    event.broadcast_train_eval_stop_timer(client, task)
    event.stop_event(client, task, exc)
    event.broadcast_container_stop_time(client, task)
    if exc is not None:
        logger.error("task %s failed", task, exc_info=exc)
        sys.exit(1)


if __name__ == "__main__":
    main()
