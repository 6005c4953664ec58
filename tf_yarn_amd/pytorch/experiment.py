"""PyTorch experiment descriptors (parity with reference
``tf_yarn/pytorch/experiment.py``)."""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, NamedTuple, Optional, Union

import torch

logger = logging.getLogger(__name__)


@dataclass
class DataLoaderArgs:
    """Reference ``pytorch/experiment.py:6-21``.  ``drop_last`` defaults to
    True: unequal batch sizes across ranks hang the allreduce (the
    reference's warning, ``pytorch/experiment.py:10-15``)."""
    batch_size: int = 1
    num_workers: int = 0
    pin_memory: bool = True
    drop_last: bool = True
    prefetch_factor: Optional[int] = None
    shuffle: bool = False

    def __post_init__(self):
        if not self.drop_last:
            logger.warning(
                "drop_last=False can produce unequal batch counts across "
                "workers and freeze the allreduce; keep it True unless the "
                "dataset guarantees equal shards")


@dataclass
class DistributedDataParallelArgs:
    """Reference ``pytorch/experiment.py:23-28``, with ``bucket_cap_mb``
    retuned from the NVLink-era 25 MB to the xGMI default (see
    :mod:`tf_yarn_amd.parallel.ddp`)."""
    broadcast_buffers: bool = True
    bucket_cap_mb: float = 32
    find_unused_parameters: bool = False
    gradient_as_bucket_view: bool = True


class PytorchExperiment(NamedTuple):
    """Reference ``pytorch/experiment.py:30-56``."""
    # Model to train
    model: torch.nn.Module
    # main_fn(model, train_loader, device, rank, tb_writer)
    main_fn: Callable[[torch.nn.Module, Any, str, int, Any], None]
    # Training dataset (map-style or iterable)
    train_dataset: Any
    dataloader_args: DataLoaderArgs = DataLoaderArgs()
    # Where per-worker tensorboard logs are collected (local or URI)
    tensorboard_hdfs_dir: Optional[str] = None
    ddp_args: Optional[DistributedDataParallelArgs] = None
    # Number of workers used to compute the number of batches per worker
    n_workers_per_executor: int = 1
