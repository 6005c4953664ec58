"""ParquetDataset sharding semantics (reference parquet_dataset.py:15-72):
contiguous per-shard batch slices, ragged last batch dropped, equal batch
counts across shards (allreduce shape-equality)."""

import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

from tf_yarn_amd.pytorch.parquet_dataset import ParquetDataset  # noqa: E402


def _write_parquet(path, n_rows, start=0):
    table = pa.table({"x": list(range(start, start + n_rows)),
                      "y": [float(i) for i in range(n_rows)]})
    pq.write_table(table, str(path))


def test_single_rank_drops_ragged_tail(tmp_path):
    f = tmp_path / "a.parquet"
    _write_parquet(f, 105)  # 10 full batches of 10 + ragged 5
    ds = ParquetDataset([str(f)], batch_size=10)
    batches = list(ds)
    assert len(batches) == 10
    assert all(b.num_rows == 10 for b in batches)
    first = batches[0].column("x").to_pylist()
    assert first == list(range(10))


def test_shards_are_disjoint_and_equal(tmp_path):
    f = tmp_path / "a.parquet"
    _write_parquet(f, 100)  # 10 batches of 10
    seen = []
    for rank in range(3):
        ds = ParquetDataset([str(f)], batch_size=10)
        ds.rank, ds.world_size = rank, 3  # 10 // 3 = 3 batches each
        rows = [r for b in ds for r in b.column("x").to_pylist()]
        assert len(rows) == 30
        seen.append(set(rows))
    assert seen[0].isdisjoint(seen[1]) and seen[1].isdisjoint(seen[2])
    # contiguous slices in file order
    assert max(seen[0]) < min(seen[1]) < max(seen[1]) < min(seen[2])


def test_multiple_files_and_column_projection(tmp_path):
    f1, f2 = tmp_path / "a.parquet", tmp_path / "b.parquet"
    _write_parquet(f1, 40)
    _write_parquet(f2, 40, start=1000)
    ds = ParquetDataset([str(f1), str(f2)], batch_size=10, columns=["x"])
    batches = list(ds)
    assert len(batches) == 8
    assert batches[0].schema.names == ["x"]
    xs = [r for b in batches for r in b.column("x").to_pylist()]
    assert 1000 in xs and 0 in xs


def test_tiny_file_skipped_when_fewer_batches_than_shards(tmp_path):
    f = tmp_path / "a.parquet"
    _write_parquet(f, 15)  # 1 full batch < 2 shards -> skipped entirely
    ds = ParquetDataset([str(f)], batch_size=10)
    ds.rank, ds.world_size = 0, 2
    assert list(ds) == []


def test_directory_listing(tmp_path):
    d = tmp_path / "data"
    d.mkdir()
    _write_parquet(d / "p1.parquet", 20)
    _write_parquet(d / "p2.parquet", 20)
    (d / "junk.txt").write_text("nope")
    ds = ParquetDataset(str(d), batch_size=10)
    assert len(ds.files) == 2
    assert len(list(ds)) == 4
