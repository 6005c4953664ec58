"""Distributed data planes on RCCL over xGMI.

Replaces the three data planes the reference wires up but does not implement
(SURVEY §2.2 N1-N4): torch DDP -> :class:`~tf_yarn_amd.parallel.ddp.BucketedDataParallel`
(own bucketed ring-allreduce reducer overlapped with backward),
Horovod-gloo -> :mod:`tf_yarn_amd.parallel.hvd` (optimizer-level allreduce hook),
TF ParameterServerStrategy -> :mod:`tf_yarn_amd.parallel.ps` (RCCL p2p push/pull).
Rendezvous flows through the framework's own KV control plane
(:class:`~tf_yarn_amd.parallel.store.KVRendezvousStore`).
"""

from tf_yarn_amd.parallel.comm import (get_backend_for_device,
                                       init_process_group)
from tf_yarn_amd.parallel.ddp import BucketedDataParallel

__all__ = ["BucketedDataParallel", "init_process_group",
           "get_backend_for_device"]
