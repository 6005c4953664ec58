"""ShardedIterableDataset: the webdataset-branch analog (reference
worker.py:50-65) — shard splitting, epoch capping, bounded shuffle."""

import torch

from tf_yarn_amd.pytorch.web_dataset import (ShardedIterableDataset,
                                             split_by_rank)


def _read(shard):
    base = shard * 10
    return range(base, base + 10)


def test_split_by_rank_round_robin():
    shards = list(range(7))
    parts = [split_by_rank(shards, r, 3) for r in range(3)]
    assert parts == [[0, 3, 6], [1, 4], [2, 5]]
    flat = [s for p in parts for s in p]
    assert sorted(flat) == shards


def test_streams_all_samples_single_consumer():
    ds = ShardedIterableDataset([0, 1, 2], _read)
    assert list(ds) == list(range(30))


def test_with_epoch_caps_samples():
    ds = ShardedIterableDataset([0, 1, 2], _read).with_epoch(12)
    assert len(list(ds)) == 12
    assert len(list(ds)) == 12  # repeatable


def test_shuffle_buffer_is_permutation_and_seeded():
    ds = ShardedIterableDataset([0, 1], _read, shuffle_buffer=8)
    run1 = list(ds)
    run2 = list(ds)
    assert sorted(run1) == list(range(20))
    assert run1 == run2  # same epoch -> same permutation
    ds.set_epoch(1)
    run3 = list(ds)
    assert sorted(run3) == list(range(20))
    assert run3 != run1  # new epoch reseeds


def test_shard_shuffle_changes_order_not_content():
    ds = ShardedIterableDataset(list(range(5)), _read,
                                shuffle_shards=True, seed=3)
    run1 = list(ds)
    assert sorted(run1) == list(range(50))
    ds.set_epoch(1)
    assert list(ds) != run1


def test_dataloader_workers_split_disjoint():
    ds = ShardedIterableDataset(list(range(4)), _read)
    loader = torch.utils.data.DataLoader(ds, num_workers=2, batch_size=None)
    seen = sorted(int(x) for x in loader)
    assert seen == list(range(40))


def test_worker_task_passes_iterable_through():
    """The worker's _create_dataloader must serve it like a WebDataset:
    no sampler, dataset's own sharding."""
    from tf_yarn_amd.pytorch.experiment import DataLoaderArgs
    from tf_yarn_amd.pytorch.tasks.worker import _create_dataloader
    ds = ShardedIterableDataset([0, 1], _read)
    loader = _create_dataloader(
        ds, DataLoaderArgs(batch_size=5, pin_memory=False),
        rank=0, world_size=1)
    batches = list(loader)
    assert all(len(b) == 5 for b in batches)
    assert int(torch.cat(batches).max()) == 19
