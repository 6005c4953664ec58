"""Primitive tests (reference tests/test__internal.py:15-48)."""

import errno
import os
import socket

import pytest

from tf_yarn_amd._internal import (MonitoredThread, ThreadState, iter_tasks,
                                   reserve_sock_addr, xset_environ)
from tf_yarn_amd.topologies import ContainerTask, TaskSpec


def test_monitored_thread_captures_exception():
    def boom():
        raise ValueError("kaboom")

    t = MonitoredThread(target=boom)
    t.start()
    t.join()
    assert t.state == ThreadState.FAILED
    assert isinstance(t.exception, ValueError)


def test_monitored_thread_success():
    t = MonitoredThread(target=lambda: None)
    t.start()
    t.join()
    assert t.state == ThreadState.SUCCEEDED
    assert t.exception is None


def test_reserve_sock_addr_holds_port():
    """Binding the reserved port again (without SO_REUSEPORT) must fail
    while inside the context (reference semantics: the socket stays open)."""
    with reserve_sock_addr() as (host, port):
        assert host == "127.0.0.1"
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as probe:
            with pytest.raises(OSError) as exc_info:
                probe.bind((host, port))
            assert exc_info.value.errno == errno.EADDRINUSE


def test_iter_tasks_expands_instances():
    tasks = iter_tasks({
        "chief": TaskSpec(instances=1, nb_proc_per_worker=2, vcores=2),
        "worker": TaskSpec(instances=3, nb_proc_per_worker=1),
    })
    assert tasks == [
        ContainerTask("chief", 0, 2),
        ContainerTask("worker", 0, 1),
        ContainerTask("worker", 1, 1),
        ContainerTask("worker", 2, 1),
    ]


def test_xset_environ_exclusive(monkeypatch):
    monkeypatch.delenv("MIYARN_TEST_VAR", raising=False)
    xset_environ(MIYARN_TEST_VAR="1")
    assert os.environ["MIYARN_TEST_VAR"] == "1"
    with pytest.raises(RuntimeError):
        xset_environ(MIYARN_TEST_VAR="2")
    del os.environ["MIYARN_TEST_VAR"]
