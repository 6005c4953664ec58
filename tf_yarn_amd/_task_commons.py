"""Task-side bootstrap helpers shared by every task module.

Parity with the reference's ``tf_yarn/_task_commons.py``: task identity from
the container env var, cluster-task discovery through the KV store, the
pickled-experiment hand-off (with the hang-avoidance start+stop emission on
failure, reference ``_task_commons.py:58-62``), KV-based master election and
the ``task_id * n_workers + local_rank`` rank convention
(``_task_commons.py:111``).
"""

from __future__ import annotations

import json
import logging
import logging.config
import os
import time
from contextlib import contextmanager
from typing import List, Optional

import cloudpickle

from tf_yarn_amd import constants, event
from tf_yarn_amd._internal import reserve_sock_addr
from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.topologies import ContainerKey, ContainerTask

logger = logging.getLogger(__name__)


def setup_logging() -> None:
    """Load the packaged logging config (reference ``_task_commons.py:19-23``)."""
    conf = os.path.join(os.path.dirname(__file__), "default.log.conf")
    if os.path.exists(conf):
        logging.config.fileConfig(conf, disable_existing_loggers=False)
    else:  # pragma: no cover - packaging fallback
        logging.basicConfig(level=logging.INFO)


def get_client() -> KVClient:
    """Connect to the application KV store (the ``from_current()`` of skein)."""
    addr = os.environ.get(constants.ENV_KV_ADDR)
    if not addr:
        raise RuntimeError(
            f"{constants.ENV_KV_ADDR} not set: task started outside a cluster")
    return KVClient(addr)


def get_task_key() -> ContainerKey:
    """Parse task identity from the container env
    (reference ``_task_commons.py:70-72``, ``SKEIN_CONTAINER_ID`` with
    ``_`` -> ``:``)."""
    cid = os.environ[constants.ENV_CONTAINER_ID]
    task_type, task_id = cid.rsplit("_", 1)
    return ContainerKey(task_type, int(task_id))


def get_task() -> str:
    return get_task_key().to_kv_str()


def is_worker(task_type: Optional[str] = None) -> bool:
    return _task_type(task_type) == "worker"


def is_evaluator(task_type: Optional[str] = None) -> bool:
    return _task_type(task_type) == "evaluator"


def is_chief(task_type: Optional[str] = None) -> bool:
    return _task_type(task_type) == "chief"


def _task_type(task_type: Optional[str]) -> str:
    return task_type if task_type is not None else get_task_key().type


def _setup_container_logs(client: KVClient) -> None:
    """Broadcast the container log location + start time
    (reference ``_task_commons.py:26-34``)."""
    task = get_task()
    log_path = os.environ.get("MIYARN_LOG_FILE", "")
    event.logs_event(client, task, log_path)
    event.broadcast_container_start_time(client, task)


def _get_cluster_tasks(client: KVClient) -> List[ContainerTask]:
    """Read ``cluster_instances`` from the KV store
    (reference ``_task_commons.py:37-41``)."""
    raw = client.wait(constants.KV_CLUSTER_INSTANCES).decode()
    return [ContainerTask(t, i, n) for t, i, n in json.loads(raw)]


def _compute_world_size(cluster_tasks: List[ContainerTask]) -> int:
    """world_size = sum of nb_proc over all tasks
    (reference ``_task_commons.py:43-45``)."""
    return sum(t.nb_proc for t in cluster_tasks)


def _get_nb_workers(task_id: int,
                    cluster_tasks: List[ContainerTask]) -> int:
    """nb_proc of this worker instance (reference ``_task_commons.py:47-52``)."""
    for t in cluster_tasks:
        if t.type == "worker" and t.id == task_id:
            return t.nb_proc
    raise ValueError(f"worker:{task_id} not in cluster tasks")


def _get_experiment(client: KVClient):
    """Unpickle and call the experiment closure.  On failure, still emit
    start+stop events so the run does not hang
    (reference ``_task_commons.py:55-68``)."""
    try:
        blob = client.wait(constants.KV_EXPERIMENT_FN)
        experiment = cloudpickle.loads(blob)()
    except Exception as e:
        task = get_task()
        event.start_event(client, task)
        event.stop_event(client, task, e)
        raise
    return experiment


def choose_master(client: KVClient, rank: int) -> str:
    """KV-based master election: rank 0 reserves a port and broadcasts
    ``host:port``; other ranks wait (reference ``_task_commons.py:95-108``)."""
    if rank == 0:
        with reserve_sock_addr() as (host, port):
            addr = f"{host}:{port}"
            event.broadcast(client, "master_addr", addr)
    else:
        addr = event.wait(client, "master_addr")
    host, port = addr.rsplit(":", 1)
    os.environ[constants.ENV_MASTER_ADDR] = host
    os.environ[constants.ENV_MASTER_PORT] = port
    return addr


def compute_rank(task_id: int, local_rank: int, n_workers: int) -> int:
    """Reference ``_task_commons.py:111-113``."""
    return task_id * n_workers + local_rank


@contextmanager
def catchtime(message: str):
    """Timing context manager (reference ``_task_commons.py:117-125``)."""
    start = time.time()
    yield
    logger.info("%s took %.3f s", message, time.time() - start)
