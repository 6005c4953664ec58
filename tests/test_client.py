"""Core client / spawner tests (reference tests/test_client.py)."""

import json
import os
from unittest import mock

import pytest

from tf_yarn_amd import client as client_mod
from tf_yarn_amd import constants
from tf_yarn_amd.client import (ContainerLogStatus, RunFailed,
                                _allocate_gpus, _handle_events,
                                _setup_cluster_spec, get_safe_experiment_fn,
                                run_on_yarn)
from tf_yarn_amd.event import (CONTAINER_START_TIME, CONTAINER_STOP_TIME,
                               TRAIN_EVAL_START_TIME, TRAIN_EVAL_STOP_TIME)
from tf_yarn_amd.topologies import ContainerTask, NodeLabel, TaskSpec


def test_setup_cluster_spec_excludes_side_tasks(kv_client):
    """cluster_instances excludes evaluator+tensorboard
    (reference client.py:174-176)."""
    tasks = [ContainerTask("chief", 0, 1), ContainerTask("worker", 0, 1),
             ContainerTask("evaluator", 0, 1),
             ContainerTask("tensorboard", 0, 1)]
    _setup_cluster_spec(tasks, kv_client)
    spec = json.loads(kv_client.get(constants.KV_CLUSTER_INSTANCES).decode())
    assert spec == [["chief", 0, 1], ["worker", 0, 1]]


def test_allocate_gpus_round_robin():
    tasks = [ContainerTask("chief", 0, 2), ContainerTask("worker", 0, 2),
             ContainerTask("worker", 1, 2), ContainerTask("ps", 0, 1),
             ContainerTask("evaluator", 0, 1)]
    specs = {
        "chief": TaskSpec(vcores=2, nb_proc_per_worker=2,
                          label=NodeLabel.GPU),
        "worker": TaskSpec(vcores=2, nb_proc_per_worker=2, instances=2,
                           label=NodeLabel.GPU),
        "ps": TaskSpec(),
        "evaluator": TaskSpec(),
    }
    gpus = _allocate_gpus(tasks, specs)
    assert gpus["chief:0"] == [0, 1]
    assert gpus["worker:0"] == [2, 3]
    assert gpus["worker:1"] == [4, 5]
    assert gpus["ps:0"] == []
    assert gpus["evaluator:0"] == []


def test_allocate_gpus_cpu_label_gets_none():
    tasks = [ContainerTask("chief", 0, 1)]
    specs = {"chief": TaskSpec()}
    assert _allocate_gpus(tasks, specs) == {"chief:0": []}


def test_get_safe_experiment_fn():
    fn = get_safe_experiment_fn("os.path.join", "a", "b")
    assert fn() == os.path.join("a", "b")


def test_handle_events_training_span():
    """training time = max(stop) - min(start) across chief+workers
    (reference client.py:703-712)."""
    tasks = [ContainerTask("chief", 0, 1), ContainerTask("worker", 0, 1),
             ContainerTask("evaluator", 0, 1)]
    events = {
        "chief:0": {TRAIN_EVAL_START_TIME: "100.0",
                    TRAIN_EVAL_STOP_TIME: "200.0",
                    CONTAINER_START_TIME: "90.0",
                    CONTAINER_STOP_TIME: "210.0"},
        "worker:0": {TRAIN_EVAL_START_TIME: "110.0",
                     TRAIN_EVAL_STOP_TIME: "230.0"},
        "evaluator:0": {TRAIN_EVAL_START_TIME: "150.0",
                        TRAIN_EVAL_STOP_TIME: "260.0"},
    }
    metrics = _handle_events(events, tasks)
    assert metrics.total_training_duration == 130.0  # 230 - 100
    assert metrics.total_eval_duration == 110.0  # 260 - 150
    assert metrics.container_duration["chief:0"] == 120.0
    assert metrics.container_duration["worker:0"] is None
    assert metrics.train_eval_time_per_node["worker:0"] == 120.0


def test_handle_events_missing_gives_none():
    metrics = _handle_events({}, [ContainerTask("chief", 0, 1)])
    assert metrics.total_training_duration is None
    assert metrics.total_eval_duration is None


def test_container_log_status_by_container_id():
    status = ContainerLogStatus({"chief:0": "/tmp/x.log"},
                                {"chief:0": "FAILED"})
    assert status.by_container_id() == {"chief:0": ("/tmp/x.log", "FAILED")}


@pytest.mark.parametrize("nb_retries,nb_failures,expect_raise", [
    (0, 0, False), (0, 1, True), (1, 1, False), (2, 3, True),
])
def test_run_on_yarn_retries(nb_retries, nb_failures, expect_raise,
                             monkeypatch):
    """Retry semantics (reference tests/test_client.py:165-198)."""
    calls = {"n": 0}

    def fake_setup(task_specs, n_try, custom, hook, base_dir,
                   extra_env=None):
        return mock.MagicMock()

    def fake_execute(cluster, fn, thresholds, n_try):
        if calls["n"] < nb_failures:
            calls["n"] += 1
            raise RunFailed("boom")
        return None, ContainerLogStatus()

    monkeypatch.setattr(client_mod, "_setup_cluster", fake_setup)
    monkeypatch.setattr(client_mod, "_execute_and_await_termination",
                        fake_execute)
    monkeypatch.setattr(client_mod, "_shutdown_cluster", lambda c: None)
    monkeypatch.setattr(client_mod, "_log_container_tails", lambda s: None)

    if expect_raise:
        with pytest.raises(RunFailed):
            run_on_yarn(lambda: None, {"chief": TaskSpec()},
                        nb_retries=nb_retries)
    else:
        run_on_yarn(lambda: None, {"chief": TaskSpec()},
                    nb_retries=nb_retries)


def test_run_on_yarn_validates_topology():
    with pytest.raises(ValueError):
        run_on_yarn(lambda: None, {"worker": TaskSpec()})


def test_tensorflow_alias_package():
    """tf-yarn users import tf_yarn.tensorflow.*; the alias keeps that
    import shape working against the estimator flavor."""
    from tf_yarn_amd.tensorflow import (Experiment, KerasExperiment,
                                        run_on_yarn)
    import tf_yarn_amd.tensorflow.client as alias_client
    import tf_yarn_amd.estimator.client as real_client
    assert alias_client is real_client
    assert run_on_yarn is real_client.run_on_yarn
    from tf_yarn_amd.estimator.experiment import Experiment as E2
    assert Experiment is E2


def test_run_on_yarn_env_reaches_tasks(tmp_path):
    """The env= kwarg (reference client.py:306) must land in every task's
    environment."""
    import cloudpickle
    import sys
    cloudpickle.register_pickle_by_value(sys.modules[__name__])
    from tf_yarn_amd import TaskSpec, run_on_yarn

    def experiment_fn(task_params):
        import os
        assert os.environ["MIYARN_TEST_FLAG"] == "42"

    metrics = run_on_yarn(
        experiment_fn,
        {"chief": TaskSpec(memory=512, vcores=1)},
        custom_task_module="tf_yarn_amd.distributed.task",
        env={"MIYARN_TEST_FLAG": 42},
        base_dir=str(tmp_path))
    assert metrics is not None
