"""Process-group bootstrap over the framework control plane.

One process per GPU, ``torch.distributed`` backend ``"nccl"`` (= RCCL on
ROCm) over xGMI; ``"gloo"`` for CPU plumbing configs (reference backend
choice ``pytorch/tasks/worker.py:171-174``).  Rendezvous goes through
:class:`~tf_yarn_amd.parallel.store.KVRendezvousStore` instead of a separate
TCPStore, so RCCL communicator bootstrap uses the same KV plane as every
barrier and election.
"""

from __future__ import annotations

import logging
import os
from datetime import timedelta
from typing import Optional

import torch
import torch.distributed as dist

from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.parallel.store import KVRendezvousStore

logger = logging.getLogger(__name__)


def get_backend_for_device(device: torch.device | str) -> str:
    device = torch.device(device)
    return "nccl" if device.type == "cuda" else "gloo"


def init_process_group(rank: int,
                       world_size: int,
                       backend: Optional[str] = None,
                       device: Optional[torch.device | str] = None,
                       kv_client: Optional[KVClient] = None,
                       timeout_minutes: float = 30,
                       group_name: str = "default",
                       need_subgroups: bool = False) -> None:
    """Initialize torch.distributed through the KV control plane.

    Falls back to env:// (MASTER_ADDR/MASTER_PORT, the reference's
    ``worker.py:101`` contract) when no KV client is available (e.g. under
    ``torch.distributed.run``) or when the caller needs ``dist.new_group``
    sub-groups (``need_subgroups=True``): c10d cannot derive sub-group
    stores from a Python Store subclass, so the PS pair-group path
    rendezvouses on a TCPStore at the KV-elected master address instead.
    """
    if backend is None:
        backend = get_backend_for_device(device or (
            "cuda" if torch.cuda.is_available() else "cpu"))
    timeout = timedelta(minutes=timeout_minutes)
    if need_subgroups:
        kv_client = None
    if kv_client is not None:
        store = KVRendezvousStore(kv_client, prefix=f"c10d/{group_name}/",
                                  timeout=timeout)
        dist.init_process_group(backend, store=store, rank=rank,
                                world_size=world_size, timeout=timeout)
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend, rank=rank, world_size=world_size,
                                timeout=timeout)
    logger.info("process group up: backend=%s rank=%d world_size=%d",
                backend, rank, world_size)


def destroy_process_group() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
