"""PyTorch worker-task unit tests (reference tests/pytorch/tasks/test_worker.py)."""

import os
from unittest import mock

import pytest
import torch

from tf_yarn_amd.pytorch.experiment import DataLoaderArgs
from tf_yarn_amd.pytorch.tasks import worker


def test_get_device_round_robin(monkeypatch):
    monkeypatch.setenv("MIYARN_GPU_IDS", "4,5")
    gpu_ids = worker._get_gpu_ids()
    assert gpu_ids == [4, 5]
    with mock.patch.object(torch.cuda, "is_available", return_value=True):
        assert worker._get_device(gpu_ids, 0) == "cuda:4"
        assert worker._get_device(gpu_ids, 1) == "cuda:5"
        assert worker._get_device(gpu_ids, 2) == "cuda:4"
    assert worker._get_device([], 0) == "cpu"


def test_get_device_cpu_when_no_cuda():
    with mock.patch.object(torch.cuda, "is_available", return_value=False):
        assert worker._get_device([3], 0) == "cpu"


def test_backend_selection():
    import torch
    n = torch.cuda.device_count()
    # nccl iff every rank can own a distinct GPU (reference
    # worker.py:171-174 procs-vs-GPUs rule)
    assert worker._get_collective_ops_backend(
        "cuda:0", world_size=max(1, n)) == ("nccl" if n else "gloo")
    assert worker._get_collective_ops_backend(
        "cuda:0", world_size=n + 1) == "gloo"
    assert worker._get_collective_ops_backend("cpu") == "gloo"


def test_create_dataloader_map_dataset_uses_sampler():
    ds = torch.utils.data.TensorDataset(torch.arange(100).float())
    loader = worker._create_dataloader(
        ds, DataLoaderArgs(batch_size=10, pin_memory=False),
        rank=0, world_size=2)
    assert isinstance(loader.sampler,
                      torch.utils.data.distributed.DistributedSampler)
    batches = list(loader)
    assert len(batches) == 5  # 100 / 2 ranks / batch 10


def test_create_dataloader_iterable_passthrough():
    class It(torch.utils.data.IterableDataset):
        def __iter__(self):
            return iter(torch.arange(20).float().split(1))

    loader = worker._create_dataloader(
        It(), DataLoaderArgs(batch_size=5, pin_memory=False),
        rank=0, world_size=2)
    assert loader.sampler is None or not isinstance(
        loader.sampler, torch.utils.data.distributed.DistributedSampler)


def test_parquet_dataset_sharding(tmp_path):
    """ParquetDataset: per-rank contiguous batch slices, equal counts,
    ragged tail dropped (reference parquet_dataset.py semantics)."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq

    from tf_yarn_amd.pytorch.parquet_dataset import ParquetDataset

    n_rows = 103  # ragged: 103 rows, batch 10 -> 10 full batches
    table = pa.table({"x": list(range(n_rows))})
    path = str(tmp_path / "part-0.parquet")
    pq.write_table(table, path)

    ds = ParquetDataset([path], batch_size=10)
    batches = list(ds)
    assert len(batches) == 10  # single shard gets all full batches
    assert all(b.num_rows == 10 for b in batches)
    seen = [v for b in batches for v in b.column("x").to_pylist()]
    assert seen == list(range(100))  # ragged tail (3 rows) dropped

    # two-shard split: equal counts, contiguous, disjoint
    ds.world_size, ds.rank = 2, 0
    first = [v for b in ds for v in b.column("x").to_pylist()]
    ds.rank = 1
    second = [v for b in ds for v in b.column("x").to_pylist()]
    assert len(first) == len(second) == 50
    assert set(first).isdisjoint(second)
