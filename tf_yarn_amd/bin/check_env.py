"""Environment diagnostic CLI (parity with reference
``tf_yarn/bin/check_hadoop_env.py``: check env prerequisites locally, then
launch a 1-container application that reports back through the KV store).

Checks: ROCm/GPU visibility, the HIP kernel extension, RCCL backend
availability, and a full spawner round trip.
Run: ``python -m tf_yarn_amd.bin.check_env``
"""

from __future__ import annotations

import logging
import os
import sys

logger = logging.getLogger(__name__)


def check_local_env() -> dict:
    """Local prerequisite checks (reference ``check_hadoop_env.py:97-123``)."""
    results = {}
    import torch
    results["torch"] = torch.__version__
    results["hip"] = getattr(torch.version, "hip", None) or "absent"
    results["gpu_available"] = torch.cuda.is_available()
    if torch.cuda.is_available():
        results["n_gpus"] = torch.cuda.device_count()
        results["gpu_name"] = torch.cuda.get_device_name(0)
    import torch.distributed as dist
    results["rccl_backend"] = dist.is_nccl_available()
    results["gloo_backend"] = dist.is_gloo_available()
    from tf_yarn_amd import ops
    results["hip_extension"] = ops.HAVE_EXT
    if ops.HAVE_EXT:
        from tf_yarn_amd.ops import _C
        results["kernels"] = sorted(
            n for n in dir(_C) if not n.startswith("_"))
    import tf_yarn_amd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(
        tf_yarn_amd.__file__)))
    results["tuned_gemm_table"] = os.path.exists(
        os.path.join(repo, "tuned", "tunableop_wide_deep.csv"))
    results["tuned_miopen_db"] = os.path.isdir(
        os.path.join(repo, "tuned", "miopen"))
    try:
        from tf_yarn_amd import _kv_native  # noqa: F401
        results["kv_native"] = True
    except ImportError:
        results["kv_native"] = False
    return results


def launch_remote_check(tmp_dir: str = None) -> bool:
    """Submit a 1-task app whose worker writes a result key to the KV store
    (reference ``check_hadoop_env.py:56-94``)."""
    from tf_yarn_amd import TaskSpec, run_on_yarn

    def check_fn(task_params):
        # runs inside the spawned chief task
        from tf_yarn_amd import _task_commons, event
        client = _task_commons.get_client()
        event.broadcast(client, "check_result",
                        f"ok rank={task_params.rank}")
        return None

    try:
        run_on_yarn(
            check_fn,
            {"chief": TaskSpec(memory=512, vcores=1)},
            custom_task_module="tf_yarn_amd.distributed.task",
            base_dir=tmp_dir)
        return True
    except Exception:
        logger.exception("remote check failed")
        return False


def main() -> int:
    logging.basicConfig(level=logging.INFO)
    ok = True
    print("== local environment ==")
    for key, value in check_local_env().items():
        print(f"  {key}: {value}")
    print("== spawner round trip ==")
    if launch_remote_check():
        print("  spawner + KV store + task bootstrap: OK")
    else:
        print("  spawner round trip: FAILED")
        ok = False
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
