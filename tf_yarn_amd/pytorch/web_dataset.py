"""Sharded streaming dataset — the webdataset-branch analog of the
reference worker (``tf_yarn/pytorch/tasks/worker.py:50-65`` routes
``wds.WebDataset``/``wds.DataPipeline`` through ``wds.WebLoader``; the
load-bearing semantics are shard-list splitting by rank and by dataloader
worker, a bounded shuffle buffer, and ``with_epoch`` step capping).

``webdataset`` itself (tar-shard streaming) is not in this environment;
this module provides the same sharding contract over arbitrary shard
"urls" plus a user ``read_shard`` function, so ``_create_dataloader``'s
iterable-dataset pass-through serves it exactly like a WebDataset.
"""

from __future__ import annotations

import random
from typing import Callable, Iterable, Iterator, List, Optional

import torch
import torch.distributed as dist


def split_by_rank(items: List, rank: Optional[int] = None,
                  world_size: Optional[int] = None) -> List:
    """wds.split_by_node equivalent: round-robin shard slice per rank."""
    if rank is None or world_size is None:
        if dist.is_available() and dist.is_initialized():
            rank, world_size = dist.get_rank(), dist.get_world_size()
        else:
            rank, world_size = 0, 1
    return items[rank::world_size]


def split_by_worker(items: List) -> List:
    """wds.split_by_worker equivalent: slice per DataLoader worker."""
    info = torch.utils.data.get_worker_info()
    if info is None:
        return items
    return items[info.id::info.num_workers]


class ShardedIterableDataset(torch.utils.data.IterableDataset):
    """Streams samples from a list of shards, split by rank then by
    DataLoader worker (each sample is seen by exactly one consumer).

    Parameters
    ----------
    shards: shard identifiers (paths/urls); order defines the split.
    read_shard: callable(shard) -> iterable of samples.
    shuffle_buffer: >0 enables a bounded reservoir shuffle (wds.shuffle).
    shuffle_shards: shuffle the shard list each epoch (seeded per epoch).
    seed: base seed for the per-epoch shard shuffle + buffer.
    """

    def __init__(self, shards: List, read_shard: Callable[..., Iterable],
                 shuffle_buffer: int = 0, shuffle_shards: bool = False,
                 seed: int = 0):
        super().__init__()
        self.shards = list(shards)
        self.read_shard = read_shard
        self.shuffle_buffer = shuffle_buffer
        self.shuffle_shards = shuffle_shards
        self.seed = seed
        self._epoch = 0
        self._steps_per_epoch: Optional[int] = None

    def with_epoch(self, n: int) -> "ShardedIterableDataset":
        """Cap one iteration pass at n samples (wds.with_epoch): makes an
        infinite/uneven stream yield fixed-size epochs so every rank
        steps the same number of times (the allreduce equal-batch rule)."""
        self._steps_per_epoch = n
        return self

    def set_epoch(self, epoch: int) -> None:
        self._epoch = epoch

    def _shard_list(self) -> List:
        shards = list(self.shards)
        if self.shuffle_shards:
            random.Random(self.seed + self._epoch).shuffle(shards)
        return split_by_worker(split_by_rank(shards))

    def _raw_iter(self) -> Iterator:
        for shard in self._shard_list():
            for sample in self.read_shard(shard):
                yield sample

    def __iter__(self) -> Iterator:
        it = self._raw_iter()
        if self.shuffle_buffer > 0:
            it = _buffered_shuffle(it, self.shuffle_buffer,
                                   self.seed + self._epoch)
        if self._steps_per_epoch is not None:
            it = _take(it, self._steps_per_epoch)
        return it


def _buffered_shuffle(it: Iterator, buffer_size: int,
                      seed: int) -> Iterator:
    rng = random.Random(seed)
    buf: List = []
    for sample in it:
        if len(buf) < buffer_size:
            buf.append(sample)
            continue
        idx = rng.randrange(buffer_size)
        out, buf[idx] = buf[idx], sample
        yield out
    rng.shuffle(buf)
    yield from buf


def _take(it: Iterator, n: int) -> Iterator:
    for i, sample in enumerate(it):
        if i >= n:
            return
        yield sample
