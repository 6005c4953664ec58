"""Low-level utilities (parity with reference ``tf_yarn/_internal.py``)."""

from __future__ import annotations

import logging
import os
import platform
import socket
import threading
from contextlib import contextmanager
from enum import Enum
from typing import Iterator, List, Tuple

from tf_yarn_amd.topologies import ContainerTask, TaskSpecs

logger = logging.getLogger(__name__)


class ThreadState(Enum):
    RUNNING = "RUNNING"
    FAILED = "FAILED"
    SUCCEEDED = "SUCCEEDED"


class MonitoredThread(threading.Thread):
    """Thread that captures the exception raised by its target
    (reference ``_internal.py:22-45``): the task wrapper inspects
    ``.exception`` / ``.state`` to propagate remote tracebacks through
    ``stop`` events instead of dying silently."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._exc: BaseException | None = None
        self._done = False

    @property
    def state(self) -> ThreadState:
        if not self._done:
            return ThreadState.RUNNING
        return ThreadState.FAILED if self._exc else ThreadState.SUCCEEDED

    @property
    def exception(self) -> BaseException | None:
        return self._exc

    def run(self) -> None:
        try:
            super().run()
        except BaseException as exc:  # noqa: BLE001 - deliberate capture
            self._exc = exc
        finally:
            self._done = True


def get_so_reuseport() -> int | None:
    """SO_REUSEPORT if the kernel supports it (reference ``_internal.py:48``)."""
    if hasattr(socket, "SO_REUSEPORT"):
        return socket.SO_REUSEPORT
    if platform.system() == "Linux":
        major, minor, *_ = platform.release().split(".")
        if (int(major), int(minor.split("-")[0])) >= (3, 9):
            return 15  # SO_REUSEPORT value on Linux
    return None


@contextmanager
def reserve_sock_addr() -> Iterator[Tuple[str, int]]:
    """Reserve an ephemeral port and KEEP the socket open while the address
    is broadcast, shrinking the port-hijack race window
    (reference ``_internal.py:61-80``).  Yields ``(host, port)``.

    Single-node: the host is always 127.0.0.1 (container hostnames may not
    resolve in this environment)."""
    so_reuseport = get_so_reuseport()
    if so_reuseport is None:
        raise RuntimeError("SO_REUSEPORT unsupported on this kernel")
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as sock:
        sock.setsockopt(socket.SOL_SOCKET, so_reuseport, 1)
        sock.bind(("127.0.0.1", 0))
        _, port = sock.getsockname()
        yield ("127.0.0.1", port)


def iter_tasks(task_specs: TaskSpecs) -> List[ContainerTask]:
    """Expand task specs into per-instance ContainerTasks
    (reference ``_internal.py:83-88``)."""
    tasks = []
    for task_type, spec in sorted(task_specs.items()):
        for i in range(spec.instances):
            tasks.append(
                ContainerTask(task_type, i, spec.nb_proc_per_worker))
    return tasks


def xset_environ(**kwargs: str) -> None:
    """Exclusively set env vars; raise if any already set
    (reference ``_internal.py:90-96``)."""
    for key, value in kwargs.items():
        if key in os.environ:
            raise RuntimeError(f"{key} already set in environment")
        os.environ[key] = value
