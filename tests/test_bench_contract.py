"""Guard the driver's bench.py contract: single JSON line on stdout with
the required fields, both single-process and under torch.distributed.run
(CPU/gloo here; the driver runs the same launch shape on MI355X)."""

import json
import os
import socket
import subprocess
import sys

import pytest


def _free_port() -> str:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return str(s.getsockname()[1])

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


def _check_json_line(stdout: str, n_gpus: int):
    lines = [l for l in stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1, f"expected 1 JSON line, got: {stdout!r}"
    result = json.loads(lines[0])
    for field in REQUIRED_FIELDS:
        assert field in result, f"missing field {field}"
    assert result["n_gpus"] == n_gpus
    assert result["value"] > 0
    assert result["higher_is_better"] is True
    assert result["scaling"] == "weak"
    assert result["data"] == "synthetic"
    assert result["config"]["parallelism"] == f"dp{n_gpus}"
    assert result["config"]["global_batch"] == \
        result["config"]["per_gpu_batch"] * n_gpus
    # round-2 diagnostics for the unattended scale run
    assert "alltoall_mode" in result["config"]
    assert "rank_spread_ms_per_step" in result["config"]
    assert "comm_breakdown" in result
    return result


@pytest.mark.timeout(300)
def test_bench_single_process():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "64", "--table-rows", "1000", "--dtype", "fp32"],
        cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json_line(out.stdout, 1)


@pytest.mark.timeout(300)
def test_bench_under_torchrun_2_ranks():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", _free_port(), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "64",
         "--table-rows", "1000", "--dtype", "fp32"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json_line(out.stdout, 2)


@pytest.mark.timeout(300)
def test_bench_under_torchrun_4_ranks_uneven_shards():
    """dp4: 26 features over 4 ranks (7,7,6,6) exercises the uneven
    all-to-all splits the round-end scale run hits."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", _free_port(), "bench.py", "--gpus", "4",
         "--steps", "2", "--warmup", "1", "--batch", "32",
         "--table-rows", "500", "--dtype", "fp32"],
        cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json_line(out.stdout, 4)


@pytest.mark.timeout(300)
def test_bench_under_torchrun_8_ranks():
    """dp8: the exact rank count of the driver's round-end scale run
    (26 features shard 4,4,3,3,3,3,3,3 over 8 ranks)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", _free_port(), "bench.py", "--gpus", "8",
         "--steps", "2", "--warmup", "1", "--batch", "16",
         "--table-rows", "200", "--dtype", "fp32"],
        cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json_line(out.stdout, 8)


@pytest.mark.timeout(300)
def test_bench_ps_contract(tmp_path):
    """scripts/bench_ps.py (BASELINE config 2): full chief+1ps+2workers
    topology end-to-end, one JSON line with the shared field set."""
    out = subprocess.run(
        [sys.executable, "scripts/bench_ps.py", "--steps", "40",
         "--batch", "256"],
        cwd=REPO, capture_output=True, text=True, timeout=280,
        env=dict(os.environ, MODEL_DIR=str(tmp_path / "ps_model")))
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-1000:]
    result = json.loads(lines[0])
    for field in REQUIRED_FIELDS:
        assert field in result, f"missing field {field}"
    assert result["value"] > 0
    assert result["config"]["parallelism"] == "ps-async"


@pytest.mark.timeout(300)
def test_bench_resnet_contract():
    """scripts/bench_resnet.py (BASELINE config 4): same JSON contract."""
    out = subprocess.run(
        [sys.executable, "scripts/bench_resnet.py", "--steps", "1",
         "--warmup", "0", "--batch", "2", "--nchw"],
        cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1
    result = json.loads(lines[0])
    for field in REQUIRED_FIELDS:
        assert field in result, f"missing field {field}"
    assert result["config"]["model"] == "resnet50"
    assert result["value"] > 0
