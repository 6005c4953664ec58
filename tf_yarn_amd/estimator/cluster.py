"""Cluster-spec assembly over the KV store (parity with reference
``tf_yarn/tensorflow/cluster.py``): every task broadcasts its ``init``
address; ``aggregate_spec`` KV-waits all of them and builds the
type -> [addr...] mapping ordered by task id (reference ``cluster.py:14-21``).

The TF-specific parts (TF_CONFIG env, the "fake Google env" trick to stop
Estimator auto-starting a server, ``tf.distribute.Server``) have no
MI355X equivalent — the process-group bootstrap in
:mod:`tf_yarn_amd.parallel.comm` replaces the gRPC server mesh — but the
spec-exchange protocol is kept because barriers and diagnostics use it.
"""

from __future__ import annotations

import json
import logging
import os
from typing import Dict, List, Optional

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.topologies import ContainerTask

logger = logging.getLogger(__name__)

CLUSTER_CONFIG_ENV = "MIYARN_CLUSTER_CONFIG"  # the TF_CONFIG analog


def aggregate_spec(client: KVClient,
                   cluster_tasks: List[ContainerTask]
                   ) -> Dict[str, List[str]]:
    """KV-wait every task's /init address, ordered by task id
    (reference ``cluster.py:14-21``)."""
    spec: Dict[str, List[str]] = {}
    for task in sorted(cluster_tasks, key=lambda t: (t.type, t.id)):
        addr = event.wait(client, f"{task.type}:{task.id}/init")
        spec.setdefault(task.type, []).append(addr)
    return spec


def start_cluster(client: KVClient,
                  cluster_tasks: List[ContainerTask],
                  sock_addr: str) -> Dict[str, List[str]]:
    """Broadcast own init address, then aggregate everyone's
    (reference ``cluster.py:24-38``)."""
    task = _task_commons.get_task()
    event.init_event(client, task, sock_addr)
    return aggregate_spec(client, cluster_tasks)


def setup_cluster_config(spec: Dict[str, List[str]],
                         task_type: Optional[str] = None,
                         task_id: Optional[int] = None) -> None:
    """Export the cluster spec into the environment (the TF_CONFIG analog,
    reference ``cluster.py:41-50``)."""
    if task_type is None:
        key = _task_commons.get_task_key()
        task_type, task_id = key.type, key.id
    os.environ[CLUSTER_CONFIG_ENV] = json.dumps({
        "cluster": spec,
        "task": {"type": task_type, "index": task_id},
    })


def get_cluster_config() -> Optional[dict]:
    raw = os.environ.get(CLUSTER_CONFIG_ENV)
    return json.loads(raw) if raw else None
