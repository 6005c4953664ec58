"""torch.distributed Store adapter over the framework KV control plane.

The reference rendezvouses NCCL through torch's TCPStore on the KV-elected
master address (``pytorch/tasks/worker.py:101,155``).  Here the rendezvous
itself flows through our own store: RCCL communicator bootstrap
(ncclUniqueId exchange inside ProcessGroupNCCL) happens via this adapter,
so the whole control plane is one system.
"""

from __future__ import annotations

import time
from datetime import timedelta
from typing import List

import torch.distributed as dist

from tf_yarn_amd.kv import KVClient


class KVRendezvousStore(dist.Store):
    """c10d Store backed by :class:`~tf_yarn_amd.kv.KVServer`.

    Implements the subset ProcessGroupNCCL/Gloo rendezvous uses:
    set/get/add/wait/check/compare_set/delete_key/num_keys.
    """

    def __init__(self, client: KVClient, prefix: str = "c10d/",
                 timeout: timedelta = timedelta(minutes=30)):
        super().__init__()
        self._client = client
        self._prefix = prefix
        self._timeout = timeout.total_seconds()

    def _k(self, key: str) -> str:
        return self._prefix + key

    def set(self, key: str, value) -> None:
        self._client.put(self._k(key), bytes(value))

    def get(self, key: str) -> bytes:
        return self._client.wait(self._k(key), timeout=self._timeout)

    def add(self, key: str, amount: int) -> int:
        return self._client.add(self._k(key), amount)

    def compare_set(self, key: str, expected, desired) -> bytes:
        return self._client.compare_set(
            self._k(key), bytes(expected), bytes(desired))

    def delete_key(self, key: str) -> bool:
        self._client.delete(self._k(key))
        return True

    def num_keys(self) -> int:
        return len(self._client.list(self._prefix))

    def wait(self, keys: List[str], timeout: timedelta = None) -> None:
        t = timeout.total_seconds() if timeout else self._timeout
        deadline = time.time() + t
        for key in keys:
            remaining = deadline - time.time()
            if remaining <= 0:
                raise RuntimeError(f"store wait timed out on {key}")
            self._client.wait(self._k(key), timeout=remaining)

    def check(self, keys: List[str]) -> bool:
        return all(self._client.get(self._k(k)) is not None for k in keys)

    def set_timeout(self, timeout: timedelta) -> None:
        self._timeout = timeout.total_seconds()
