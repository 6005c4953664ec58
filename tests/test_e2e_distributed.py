"""End-to-end tests for the raw `distributed` flavor and the env-check
CLI round trip (reference check_hadoop_env submits a 1-container app)."""

import sys

import cloudpickle
import pytest
import torch

from tf_yarn_amd import TaskSpec
from tf_yarn_amd.distributed import TaskParameters, run_on_yarn

cloudpickle.register_pickle_by_value(sys.modules[__name__])


def _train_fn(task_params: TaskParameters):
    import os

    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = task_params.master_address
    os.environ["MASTER_PORT"] = str(task_params.master_port)
    dist.init_process_group("gloo", rank=task_params.rank,
                            world_size=task_params.world_size)
    t = torch.tensor([float(task_params.rank + 1)])
    dist.all_reduce(t)
    expected = sum(range(1, task_params.world_size + 1))
    assert t.item() == expected, (t.item(), expected)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_flavor_allreduce(tmp_path):
    metrics = run_on_yarn(
        _train_fn,
        {
            "chief": TaskSpec(memory=512, vcores=1),
            "worker": TaskSpec(memory=512, vcores=1, instances=2),
        },
        base_dir=str(tmp_path / "app"),
    )
    assert metrics is not None
    assert metrics.total_training_duration is not None


@pytest.mark.timeout(180)
def test_check_env_round_trip(tmp_path):
    from tf_yarn_amd.bin.check_env import check_local_env, \
        launch_remote_check
    local = check_local_env()
    assert "torch" in local
    assert local["gloo_backend"] is True
    assert launch_remote_check(str(tmp_path / "app")) is True


@pytest.mark.timeout(180)
def test_distributed_flavor_multiproc_per_worker(tmp_path):
    """nb_proc_per_worker=2: rank math task_id * nb_proc + local_rank
    (reference distributed/task.py:37-55) across spawned sub-processes."""
    metrics = run_on_yarn(
        _train_fn,
        {
            "worker": TaskSpec(memory=512, vcores=2, instances=2,
                               nb_proc_per_worker=2),
        },
        base_dir=str(tmp_path / "app"),
    )
    assert metrics is not None  # 4-rank allreduce inside asserted 1+2+3+4


@pytest.mark.timeout(180)
def test_hard_crash_kills_hung_survivors(tmp_path):
    """A task dying WITHOUT publishing a stop event (SIGKILL/OOM shape:
    os._exit) while its peer blocks forever on a collective must fail
    the run in bounded time with the survivor killed — the realistic
    GPU-box failure the event protocol cannot see (reference fail-fast
    model: containers never restart, client.py:233)."""
    import time as _time
    from tf_yarn_amd import RunFailed, TaskSpec
    from tf_yarn_amd.distributed import run_on_yarn

    def crashing_fn():
        def fn(task_params):
            import os
            import time
            if task_params.rank == 0:
                os._exit(137)  # hard death: no stop event, no traceback
            time.sleep(3600)  # survivor hangs (e.g. stuck collective)
        return fn

    t0 = _time.perf_counter()
    with pytest.raises(RunFailed):
        run_on_yarn(
            crashing_fn(),
            {"worker": TaskSpec(memory=512, vcores=1, instances=2)},
            base_dir=str(tmp_path / "app"),
        )
    # bounded: poll period + shutdown grace, nowhere near the 3600 s hang
    assert _time.perf_counter() - t0 < 120
