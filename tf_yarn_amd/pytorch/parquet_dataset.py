"""Distributed IterableDataset over parquet files (parity with reference
``tf_yarn/pytorch/parquet_dataset.py``): each rank reads a contiguous slice
of batches per file; the last batch of each file is dropped so every rank
yields the same batch count (allreduce shape-equality,
``parquet_dataset.py:37-43``)."""

from __future__ import annotations

import logging
from typing import List

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


class ParquetDataset(torch.utils.data.IterableDataset):
    """Reference ``parquet_dataset.py:15-72``."""

    def __init__(self, dataset_path, batch_size: int,
                 num_samples: int = None, columns: List[str] = None):
        super().__init__()
        import pyarrow.parquet as pq  # local import: pyarrow optional
        self._pq = pq
        if isinstance(dataset_path, str):
            from tf_yarn_amd.utils.fs import resolve_filesystem_and_path
            fs, path = resolve_filesystem_and_path(dataset_path)
            files = [p for p in fs.ls(path) if p.endswith(".parquet")] \
                if not path.endswith(".parquet") else [path]
        else:
            files = list(dataset_path)
        self.files = files
        self.batch_size = batch_size
        self.columns = columns
        # rank/world detection (reference parquet_dataset.py:29-30)
        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()
        else:
            self.rank = 0
            self.world_size = 1

    def __iter__(self):
        # Defensive: a dataset constructed before init_process_group
        # (default rank 0 / world 1) picks up the live group at first
        # iteration, so sharding is correct even when the experiment was
        # materialized early.
        if self.world_size == 1 and dist.is_available() \
                and dist.is_initialized():
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()
        worker_info = torch.utils.data.get_worker_info()
        n_loaders = worker_info.num_workers if worker_info else 1
        loader_id = worker_info.id if worker_info else 0
        shard = self.rank * n_loaders + loader_id
        n_shards = self.world_size * n_loaders
        for fname in self.files:
            pf = self._pq.ParquetFile(fname)
            n_rows = pf.metadata.num_rows
            # Drop the last (ragged) batch per file; split the rest evenly,
            # dropping the remainder so every shard gets the same count.
            n_batches = n_rows // self.batch_size
            per_shard = n_batches // n_shards
            if per_shard == 0:
                continue
            start = shard * per_shard
            stop = start + per_shard
            for i, batch in enumerate(pf.iter_batches(
                    batch_size=self.batch_size, columns=self.columns)):
                if i >= stop:
                    break
                if i >= start and batch.num_rows == self.batch_size:
                    yield batch
