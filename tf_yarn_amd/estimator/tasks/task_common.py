"""Shared Estimator-task logic (reference
``tf_yarn/tensorflow/tasks/tf_task_common.py``): container prep, the
monitored train thread with train/eval timers, and the shutdown stop-barrier
that propagates exceptions and releases never-terminating ps tasks."""

from __future__ import annotations

import logging
import re
from typing import List, Optional

from tf_yarn_amd import _task_commons, event
from tf_yarn_amd._internal import MonitoredThread
from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.topologies import ContainerTask

logger = logging.getLogger(__name__)


def _prepare_container(client: KVClient):
    """Reference ``tf_task_common.py:21-35``: broadcast log location +
    start time, read cluster tasks."""
    _task_commons._setup_container_logs(client)
    cluster_tasks = _task_commons._get_cluster_tasks(client)
    return cluster_tasks


def _gen_monitored_train_and_evaluate(fn, client: KVClient, task: str):
    """Wrap the training fn with train/eval start/stop timers
    (reference ``tf_task_common.py:38-53``)."""

    def wrapped():
        event.broadcast_train_eval_start_timer(client, task)
        try:
            fn()
        finally:
            event.broadcast_train_eval_stop_timer(client, task)

    return wrapped


def _execute_dispatched_function(client: KVClient, fn) -> MonitoredThread:
    """Run the dispatched function in a MonitoredThread and emit the
    ``start`` event (reference ``tf_task_common.py:56-74``)."""
    task = _task_commons.get_task()
    thread = MonitoredThread(
        name=f"{task}-train",
        target=_gen_monitored_train_and_evaluate(fn, client, task),
        daemon=True)
    thread.start()
    event.start_event(client, task)
    return thread


def matches_device_filters(task: str,
                           device_filters: Optional[List[str]]) -> bool:
    """Map the reference's ``/job:x/task:n`` device-filter semantics onto
    ``type:id`` task keys (reference ``tf_task_common.py:109-118``)."""
    if not device_filters:
        return True
    task_type, task_id = task.split(":")
    for f in device_filters:
        m = re.match(r"/job:([a-z_]+)(?:/task:(\d+))?", f)
        if m is None:
            continue
        if m.group(1) == task_type and (
                m.group(2) is None or m.group(2) == task_id):
            return True
    return False


def wait_for_connected_tasks(client: KVClient,
                             cluster_tasks: List[ContainerTask],
                             device_filters: Optional[List[str]],
                             message: str = "stop") -> None:
    """Stop-barrier: wait for /stop from every device-filter-matched task
    (reference ``tf_task_common.py:102-107``)."""
    for t in cluster_tasks:
        task = f"{t.type}:{t.id}"
        if matches_device_filters(task, device_filters):
            event.wait(client, f"{task}/{message}")


def _shutdown_container(client: KVClient,
                        cluster_tasks: List[ContainerTask],
                        device_filters: Optional[List[str]],
                        thread: Optional[MonitoredThread]) -> None:
    """Emit ``stop`` (with the captured exception), run the stop-barrier,
    then re-raise so the process exits non-zero
    (reference ``tf_task_common.py:77-99``)."""
    exc = thread.exception if thread is not None else None
    task = _task_commons.get_task()
    event.stop_event(client, task, exc)
    wait_for_connected_tasks(client, cluster_tasks, device_filters)
    event.broadcast_container_stop_time(client, task)
    if exc is not None:
        raise exc
