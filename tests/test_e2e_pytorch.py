"""End-to-end: run_on_yarn (pytorch flavor) on CPU with chief + worker.

Exercises the full stack: spawner -> worker task -> KV master election ->
gloo process group via the KV rendezvous store -> BucketedDataParallel ->
user main_fn -> checkpoint -> event aggregation -> Metrics.
"""

import os
import sys

import cloudpickle
import pytest
import torch
from torch import nn

from tf_yarn_amd import TaskSpec

# Task processes cannot import this test module: pickle its closures by value.
cloudpickle.register_pickle_by_value(sys.modules[__name__])
from tf_yarn_amd.pytorch import (DataLoaderArgs, PytorchExperiment,
                                 run_on_yarn)


def _experiment_fn(model_dir):
    import torch
    from torch import nn

    from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
    from tf_yarn_amd.pytorch import model_ckpt

    def main_fn(model, loader, device, rank, tb_writer):
        import torch.distributed as dist
        opt = torch.optim.SGD(model.parameters(), lr=0.05)
        losses = []
        for epoch in range(2):
            for x, y in loader:
                opt.zero_grad()
                loss = nn.functional.mse_loss(model(x), y)
                loss.backward()
                opt.step()
                losses.append(loss.item())
        dist.barrier()
        if rank == 0:
            model_ckpt.save_ckpt(model_dir, model, opt, epoch=2,
                                 final_loss=losses[-1])
        if tb_writer is not None:
            tb_writer.add_scalar("loss", losses[-1], 2)

    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 1))
    x = torch.randn(64, 8)
    y = x.sum(dim=1, keepdim=True)
    dataset = torch.utils.data.TensorDataset(x, y)
    return PytorchExperiment(
        model=model,
        main_fn=main_fn,
        train_dataset=dataset,
        dataloader_args=DataLoaderArgs(batch_size=8, pin_memory=False),
        tensorboard_hdfs_dir=os.path.join(model_dir, "tb"),
    )


@pytest.mark.timeout(180)
def test_run_on_yarn_pytorch_cpu(tmp_path):
    model_dir = str(tmp_path / "model")
    from functools import partial
    metrics = run_on_yarn(
        partial(_experiment_fn, model_dir),
        {
            "chief": TaskSpec(memory=512, vcores=1),
            "worker": TaskSpec(memory=512, vcores=1, instances=1),
        },
        base_dir=str(tmp_path / "app"),
    )
    assert metrics is not None
    assert metrics.total_training_duration is not None
    assert metrics.total_training_duration > 0
    # rank 0 (chief) wrote the checkpoint
    assert os.path.exists(os.path.join(model_dir, "model_2.pt"))
    # per-worker tensorboard logs were uploaded for both ranks
    tb_dir = os.path.join(model_dir, "tb")
    assert os.path.isdir(os.path.join(tb_dir, "worker_0"))
    assert os.path.isdir(os.path.join(tb_dir, "worker_1"))


@pytest.mark.timeout(120)
def test_run_on_yarn_failure_propagates(tmp_path):
    from tf_yarn_amd import RunFailed

    def bad_experiment():
        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
        import torch
        from torch import nn

        def main_fn(model, loader, device, rank, tb_writer):
            raise RuntimeError("deliberate failure for test")

        return PytorchExperiment(
            model=nn.Linear(2, 2),
            main_fn=main_fn,
            train_dataset=torch.utils.data.TensorDataset(
                torch.randn(8, 2), torch.randn(8, 2)),
            dataloader_args=DataLoaderArgs(batch_size=4, pin_memory=False),
        )

    with pytest.raises(RunFailed):
        run_on_yarn(
            bad_experiment,
            {"chief": TaskSpec(memory=512, vcores=1)},
            base_dir=str(tmp_path / "app"),
        )


@pytest.mark.timeout(180)
def test_run_on_yarn_pytorch_worker_only(tmp_path):
    """The reference README's pytorch topology has NO chief
    (README.md:253-260: workers only, nb_proc_per_worker=2) — the
    flavor's injected task module must not require one."""
    model_dir = str(tmp_path / "model")
    from functools import partial
    metrics = run_on_yarn(
        partial(_experiment_fn, model_dir),
        {
            "worker": TaskSpec(memory=512, vcores=2, instances=1,
                               nb_proc_per_worker=2),
        },
        base_dir=str(tmp_path / "app"),
    )
    assert metrics is not None
    assert os.path.exists(os.path.join(model_dir, "model_2.pt"))


@pytest.mark.timeout(180)
def test_run_on_yarn_retry_recovers(tmp_path):
    """nb_retries end to end (not mocked): the task crashes on try 0 and
    succeeds on try 1 (reference client.py:431-466 whole-app retry)."""
    flag = str(tmp_path / "first_try_done")

    def experiment_fn():
        def main_fn(model, loader, device, rank, tb_writer):
            import os
            if not os.path.exists(flag):
                open(flag, "w").close()
                raise RuntimeError("deliberate first-try failure")

        import torch
        from torch import nn

        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
        torch.manual_seed(0)
        x = torch.randn(16, 4)
        dataset = torch.utils.data.TensorDataset(x, x.sum(1, keepdim=True))
        return PytorchExperiment(
            model=nn.Linear(4, 1), main_fn=main_fn,
            train_dataset=dataset,
            dataloader_args=DataLoaderArgs(batch_size=8, pin_memory=False))

    metrics = run_on_yarn(
        experiment_fn,
        {"chief": TaskSpec(memory=512, vcores=1)},
        nb_retries=1,
        base_dir=str(tmp_path / "app"))
    assert metrics is not None  # second try succeeded
    assert os.path.exists(flag)


@pytest.mark.timeout(180)
def test_run_on_yarn_with_parquet_dataset(tmp_path):
    """PytorchExperiment over a ParquetDataset: the iterable dataset
    shards itself by rank inside the spawned workers (reference
    worker.py:54-65 + parquet_dataset.py integration)."""
    pq = pytest.importorskip("pyarrow.parquet")
    import pyarrow as pa

    pfile = tmp_path / "data.parquet"
    table = pa.table({"x": [float(i) for i in range(160)]})
    pq.write_table(table, str(pfile))
    out_dir = tmp_path / "out"
    out_dir.mkdir()

    def experiment_fn():
        import torch
        from torch import nn

        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
        from tf_yarn_amd.pytorch.parquet_dataset import ParquetDataset

        def main_fn(model, loader, device, rank, tb_writer):
            rows = []
            for batch in loader:
                rows += [float(v) for v in batch.column("x").to_pylist()]
            (out_dir / f"rank{rank}.txt").write_text(
                ",".join(str(int(r)) for r in rows))

        ds = ParquetDataset([str(pfile)], batch_size=10)
        return PytorchExperiment(
            model=nn.Linear(2, 1), main_fn=main_fn, train_dataset=ds,
            dataloader_args=DataLoaderArgs(batch_size=None,
                                           pin_memory=False))

    metrics = run_on_yarn(
        experiment_fn,
        {"worker": TaskSpec(memory=512, vcores=1, instances=2)},
        base_dir=str(tmp_path / "app"))
    assert metrics is not None
    seen = []
    for rank in range(2):
        vals = (out_dir / f"rank{rank}.txt").read_text().split(",")
        assert len(vals) == 80, f"rank {rank} got {len(vals)} rows"
        seen.append(set(vals))
    assert seen[0].isdisjoint(seen[1])


@pytest.mark.timeout(180)
def test_run_on_yarn_with_sharded_iterable_dataset(tmp_path):
    """ShardedIterableDataset (the webdataset-branch analog) through the
    spawned flavor: two workers stream disjoint shards."""
    out_dir = tmp_path / "out"
    out_dir.mkdir()

    def experiment_fn():
        from torch import nn

        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
        from tf_yarn_amd.pytorch.web_dataset import ShardedIterableDataset

        def read_shard(shard):
            return range(shard * 10, shard * 10 + 10)

        def main_fn(model, loader, device, rank, tb_writer):
            seen = [int(v) for batch in loader for v in batch]
            (out_dir / f"rank{rank}.txt").write_text(
                ",".join(map(str, sorted(seen))))

        ds = ShardedIterableDataset(list(range(4)), read_shard)
        return PytorchExperiment(
            model=nn.Linear(2, 1), main_fn=main_fn, train_dataset=ds,
            dataloader_args=DataLoaderArgs(batch_size=5,
                                           pin_memory=False))

    metrics = run_on_yarn(
        experiment_fn,
        {"worker": TaskSpec(memory=512, vcores=1, instances=2)},
        base_dir=str(tmp_path / "app"))
    assert metrics is not None
    r0 = set((out_dir / "rank0.txt").read_text().split(","))
    r1 = set((out_dir / "rank1.txt").read_text().split(","))
    assert len(r0) == 20 and len(r1) == 20 and r0.isdisjoint(r1)


@pytest.mark.timeout(180)
def test_checkpoint_resume_across_runs(tmp_path):
    """The reference resume pattern (pytorch_distributed_example.py:82-98):
    run 2 resumes from the latest model_{epoch}.pt of run 1."""
    model_dir = str(tmp_path / "ckpts")
    out = tmp_path / "resumed_epoch.txt"

    def experiment_fn():
        import torch
        from torch import nn

        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment
        from tf_yarn_amd.pytorch import model_ckpt

        def main_fn(model, loader, device, rank, tb_writer):
            opt = torch.optim.SGD(model.parameters(), lr=0.05)
            ckpt = model_ckpt.load_latest_ckpt(model_dir, model, opt,
                                               device)
            start_epoch = (ckpt["epoch"] + 1) if ckpt else 0
            for epoch in range(start_epoch, start_epoch + 2):
                for x, y in loader:
                    opt.zero_grad()
                    nn.functional.mse_loss(model(x), y).backward()
                    opt.step()
            if rank == 0:
                model_ckpt.save_ckpt(model_dir, model, opt,
                                     epoch=start_epoch + 1)
                out.write_text(str(start_epoch))

        torch.manual_seed(0)
        x = torch.randn(32, 4)
        dataset = torch.utils.data.TensorDataset(x, x.sum(1, keepdim=True))
        return PytorchExperiment(
            model=nn.Linear(4, 1), main_fn=main_fn, train_dataset=dataset,
            dataloader_args=DataLoaderArgs(batch_size=8, pin_memory=False))

    for i in range(2):
        metrics = run_on_yarn(
            experiment_fn,
            {"chief": TaskSpec(memory=512, vcores=1)},
            base_dir=str(tmp_path / f"app{i}"))
        assert metrics is not None
    # run 1 started at 0, saved model_1.pt; run 2 resumed at epoch 2
    assert out.read_text() == "2"
    assert os.path.exists(os.path.join(model_dir, "model_3.pt"))
