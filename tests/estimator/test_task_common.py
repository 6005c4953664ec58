"""task_common tests: device-filter matching + shutdown barrier
(reference tests/tensorflow/test_tf_task_common.py)."""

import threading
import time

import pytest

from tf_yarn_amd import event
from tf_yarn_amd.estimator.tasks.task_common import (
    matches_device_filters, wait_for_connected_tasks)
from tf_yarn_amd.topologies import ContainerTask


@pytest.mark.parametrize("task,filters,expected", [
    ("worker:0", None, True),
    ("worker:0", [], True),
    ("worker:0", ["/job:ps", "/job:worker"], True),
    ("ps:1", ["/job:ps", "/job:worker"], True),
    ("chief:0", ["/job:ps", "/job:worker"], False),
    ("worker:1", ["/job:worker/task:1"], True),
    ("worker:0", ["/job:worker/task:1"], False),
    ("evaluator:0", ["/job:ps", "/job:worker"], False),
])
def test_matches_device_filters(task, filters, expected):
    assert matches_device_filters(task, filters) is expected


def test_wait_for_connected_tasks_barrier(kv_client):
    tasks = [ContainerTask("chief", 0, 1), ContainerTask("ps", 0, 1),
             ContainerTask("worker", 0, 1)]
    filters = ["/job:ps", "/job:worker"]
    done = threading.Event()

    def waiter():
        wait_for_connected_tasks(kv_client, tasks, filters)
        done.set()

    t = threading.Thread(target=waiter, daemon=True)
    t.start()
    # chief's stop is NOT required by the filters
    event.stop_event(kv_client, "ps:0", None)
    time.sleep(0.2)
    assert not done.is_set()
    event.stop_event(kv_client, "worker:0", None)
    assert done.wait(timeout=5)
