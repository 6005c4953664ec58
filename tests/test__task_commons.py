"""Task-bootstrap tests (reference tests/test__task_commons.py)."""

import json
import os
from unittest import mock

import cloudpickle
import pytest

from tf_yarn_amd import _task_commons, constants
from tf_yarn_amd.topologies import ContainerKey, ContainerTask


def test_get_task_key(monkeypatch):
    monkeypatch.setenv(constants.ENV_CONTAINER_ID, "worker_3")
    assert _task_commons.get_task_key() == ContainerKey("worker", 3)
    assert _task_commons.get_task() == "worker:3"


def test_role_predicates(monkeypatch):
    monkeypatch.setenv(constants.ENV_CONTAINER_ID, "chief_0")
    assert _task_commons.is_chief()
    assert not _task_commons.is_worker()
    assert not _task_commons.is_evaluator()
    assert _task_commons.is_worker("worker")


def test_get_cluster_tasks(kv_client):
    payload = [["chief", 0, 1], ["worker", 0, 2], ["worker", 1, 2]]
    kv_client.put(constants.KV_CLUSTER_INSTANCES,
                  json.dumps(payload).encode())
    tasks = _task_commons._get_cluster_tasks(kv_client)
    assert tasks == [ContainerTask("chief", 0, 1),
                     ContainerTask("worker", 0, 2),
                     ContainerTask("worker", 1, 2)]
    assert _task_commons._compute_world_size(tasks) == 5


def test_get_nb_workers():
    tasks = [ContainerTask("chief", 0, 1), ContainerTask("worker", 1, 4)]
    assert _task_commons._get_nb_workers(1, tasks) == 4
    with pytest.raises(ValueError):
        _task_commons._get_nb_workers(9, tasks)


def test_get_experiment_success(kv_client):
    kv_client.put(constants.KV_EXPERIMENT_FN,
                  cloudpickle.dumps(lambda: "the-experiment"))
    assert _task_commons._get_experiment(kv_client) == "the-experiment"


def test_get_experiment_failure_emits_events(kv_client, monkeypatch):
    """On a failing experiment_fn the task must still emit start+stop so
    the run does not hang (reference _task_commons.py:58-62)."""
    monkeypatch.setenv(constants.ENV_CONTAINER_ID, "worker_0")

    def bad_fn():
        raise RuntimeError("broken closure")

    kv_client.put(constants.KV_EXPERIMENT_FN, cloudpickle.dumps(bad_fn))
    with pytest.raises(RuntimeError, match="broken closure"):
        _task_commons._get_experiment(kv_client)
    assert kv_client.get("worker:0/start") == b""
    assert b"broken closure" in kv_client.get("worker:0/stop")


def test_choose_master_rank0_broadcasts(kv_client, monkeypatch):
    monkeypatch.delenv(constants.ENV_MASTER_ADDR, raising=False)
    monkeypatch.delenv(constants.ENV_MASTER_PORT, raising=False)
    addr = _task_commons.choose_master(kv_client, rank=0)
    assert kv_client.get("master_addr").decode() == addr
    assert os.environ[constants.ENV_MASTER_ADDR] == "127.0.0.1"
    assert int(os.environ[constants.ENV_MASTER_PORT]) > 0


def test_choose_master_other_rank_waits(kv_client, monkeypatch):
    monkeypatch.setenv(constants.ENV_MASTER_ADDR, "stale")
    kv_client.put("master_addr", b"127.0.0.1:4242")
    addr = _task_commons.choose_master(kv_client, rank=1)
    assert addr == "127.0.0.1:4242"
    assert os.environ[constants.ENV_MASTER_PORT] == "4242"


def test_compute_rank():
    assert _task_commons.compute_rank(task_id=2, local_rank=1,
                                      n_workers=4) == 9
