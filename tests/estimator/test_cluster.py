"""Cluster-spec glue unit tests (reference tests/tensorflow/
test_cluster.py: aggregate_spec via dict-KV, cluster config export) and
the task-common dispatch/shutdown paths (reference tests/tensorflow/
test_tf_task_common.py)."""

import os
from unittest import mock

import pytest

from tf_yarn_amd import event
from tf_yarn_amd.constants import ENV_CONTAINER_ID
from tf_yarn_amd.estimator import cluster
from tf_yarn_amd.estimator.tasks import task_common
from tf_yarn_amd.topologies import ContainerTask


def test_aggregate_spec_orders_by_task_id(kv_client):
    tasks = [ContainerTask("worker", 1, 1), ContainerTask("chief", 0, 1),
             ContainerTask("worker", 0, 1)]
    kv_client.put("chief:0/init", b"host0:1000")
    kv_client.put("worker:0/init", b"host1:2000")
    kv_client.put("worker:1/init", b"host2:3000")
    spec = cluster.aggregate_spec(kv_client, tasks)
    assert spec == {"chief": ["host0:1000"],
                    "worker": ["host1:2000", "host2:3000"]}


def test_start_cluster_broadcasts_own_init(kv_client):
    tasks = [ContainerTask("chief", 0, 1)]
    with mock.patch.dict(os.environ, {ENV_CONTAINER_ID: "chief_0"}):
        spec = cluster.start_cluster(kv_client, tasks, "me:4242")
    assert spec == {"chief": ["me:4242"]}
    assert kv_client.get("chief:0/init") == b"me:4242"


def test_setup_and_get_cluster_config(kv_client):
    spec = {"chief": ["a:1"], "worker": ["b:2", "c:3"]}
    with mock.patch.dict(os.environ, {ENV_CONTAINER_ID: "worker_1"}):
        cluster.setup_cluster_config(spec)
        cfg = cluster.get_cluster_config()
    assert cfg["cluster"] == spec
    assert cfg["task"] == {"type": "worker", "index": 1}


def test_execute_dispatched_function_emits_start_and_timers(kv_client):
    """Reference test__execute_dispatched_function: the fn runs in a
    MonitoredThread, `start` + train/eval timer events appear."""
    ran = []
    with mock.patch.dict(os.environ, {ENV_CONTAINER_ID: "chief_0"}):
        thread = task_common._execute_dispatched_function(
            kv_client, lambda: ran.append(1))
        thread.join(timeout=30)
    assert ran == [1]
    assert thread.exception is None
    assert kv_client.get("chief:0/start") is not None
    assert kv_client.get("chief:0/train_eval_start_time") is not None
    assert kv_client.get("chief:0/train_eval_stop_time") is not None


def test_shutdown_container_reraises_thread_exception(kv_client):
    """Reference test__shutdown_container: stop event carries the
    formatted exception and the shutdown re-raises it."""
    boom = ValueError("boom")
    with mock.patch.dict(os.environ, {ENV_CONTAINER_ID: "worker_0"}):
        thread = task_common._execute_dispatched_function(
            kv_client, lambda: (_ for _ in ()).throw(boom))
        thread.join(timeout=30)
        assert isinstance(thread.exception, ValueError)
        tasks = [ContainerTask("worker", 0, 1)]
        # the barrier waits on worker:0/stop which _shutdown_container
        # itself emits first, so this returns
        with pytest.raises(ValueError):
            task_common._shutdown_container(kv_client, tasks,
                                            ["/job:worker"], thread)
    stop = kv_client.get("worker:0/stop")
    assert stop is not None and b"boom" in stop


def test_prepare_container_reads_cluster_tasks(kv_client):
    event_tasks = [ContainerTask("chief", 0, 1),
                   ContainerTask("worker", 0, 2)]
    import json

    from tf_yarn_amd import constants
    kv_client.put(constants.KV_CLUSTER_INSTANCES,
                  json.dumps([list(t) for t in event_tasks]).encode())
    with mock.patch.dict(os.environ, {ENV_CONTAINER_ID: "chief_0"}):
        tasks = task_common._prepare_container(kv_client)
    assert tasks == event_tasks
    # container log URL + start time broadcast
    assert kv_client.get("chief:0/logs") is not None
    assert kv_client.get("chief:0/container_start_time") is not None
