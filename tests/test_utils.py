"""Utils: phase timer, board server, fs seam, summary writer."""

import json
import time
import urllib.request

import pytest
import torch

from tf_yarn_amd import tensorboard
from tf_yarn_amd.utils import tb, trace
from tf_yarn_amd.utils.fs import LocalFs, resolve_filesystem_and_path


def test_phase_timer_cpu():
    timer = trace.PhaseTimer(use_cuda=False)
    for _ in range(3):
        with timer.phase("work"):
            time.sleep(0.01)
    s = timer.summary()
    assert s["work"]["count"] == 3
    assert s["work"]["median_ms"] >= 9.0


def test_phase_timer_chrome_trace(tmp_path):
    timer = trace.PhaseTimer(use_cuda=False)
    with timer.phase("a"):
        pass
    path = str(tmp_path / "trace.json")
    timer.export_chrome_trace(path)
    data = json.load(open(path))
    assert data["traceEvents"][0]["name"] == "a"


def test_board_server_serves_metrics(tmp_path):
    writer = tb.SummaryWriter(str(tmp_path))
    writer.add_scalar("loss", 1.5, step=3)
    writer.close()
    server, url = tensorboard.start_tf_board(None, str(tmp_path))
    try:
        with urllib.request.urlopen(f"{url}/metrics", timeout=10) as r:
            events = json.loads(r.read())
        assert events[0]["tag"] == "loss"
        assert events[0]["value"] == 1.5
        with urllib.request.urlopen(url, timeout=10) as r:
            html = r.read().decode()
        assert "loss" in html
    finally:
        server.shutdown()


def test_resolve_filesystem_and_path(tmp_path):
    fs, path = resolve_filesystem_and_path(f"file://{tmp_path}/x.txt")
    assert isinstance(fs, LocalFs)
    assert path == f"{tmp_path}/x.txt"
    fs2, path2 = resolve_filesystem_and_path("/plain/path")
    assert path2 == "/plain/path"
    with pytest.raises(ValueError):
        resolve_filesystem_and_path("hdfs://nn/path")


def test_localfs_roundtrip(tmp_path):
    fs = LocalFs()
    src = tmp_path / "a.txt"
    src.write_text("hello")
    fs.put(str(src), str(tmp_path / "sub" / "b.txt"))
    assert fs.exists(str(tmp_path / "sub" / "b.txt"))
    assert (tmp_path / "sub" / "b.txt").read_text() == "hello"
    assert str(tmp_path / "a.txt") in fs.ls(str(tmp_path))
    fs.rm(str(tmp_path / "sub"), recursive=True)
    assert not fs.exists(str(tmp_path / "sub"))


def test_summary_writer_read_events(tmp_path):
    with tb.SummaryWriter(str(tmp_path)) as w:
        w.add_scalars("m", {"a": 1.0, "b": 2.0}, step=1)
    events = tb.read_events(str(tmp_path))
    tags = {e["tag"] for e in events}
    assert tags == {"m/a", "m/b"}


def test_packaging_zip_path(tmp_path):
    """packaging.zip_path (reference packaging.py:23-37 facade)."""
    import zipfile

    from tf_yarn_amd import packaging
    src = tmp_path / "pkg"
    src.mkdir()
    (src / "a.py").write_text("x = 1\n")
    (src / "sub").mkdir()
    (src / "sub" / "b.py").write_text("y = 2\n")
    out = packaging.zip_path(str(src), False, str(tmp_path))
    assert out.endswith(".zip") and __import__("os").path.exists(out)
    names = zipfile.ZipFile(out).namelist()
    assert any(n.endswith("a.py") for n in names)
    assert any(n.endswith("b.py") for n in names)


def test_check_env_local_and_remote():
    """bin/check_env (reference bin/check_hadoop_env.py): local env
    report + 1-task spawner round trip reporting through the KV store."""
    from tf_yarn_amd.bin import check_env
    results = check_env.check_local_env()
    assert results["gloo_backend"] is True
    assert "torch" in results
    assert check_env.launch_remote_check() is True


def test_analyze_rocpd_tool(tmp_path):
    """scripts/analyze_rocpd.py parses a minimal rocpd schema."""
    import sqlite3
    import subprocess
    import sys
    db = tmp_path / "r.db"
    conn = sqlite3.connect(str(db))
    conn.execute("CREATE TABLE rocpd_info_kernel_symbol "
                 "(id INTEGER, display_name TEXT)")
    conn.execute("CREATE TABLE rocpd_kernel_dispatch "
                 "(kernel_id INTEGER, start INTEGER, end INTEGER)")
    conn.execute("INSERT INTO rocpd_info_kernel_symbol VALUES (1, 'k1')")
    conn.executemany("INSERT INTO rocpd_kernel_dispatch VALUES (1, ?, ?)",
                     [(0, 1000), (2000, 4000)])
    conn.commit()
    conn.close()
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "scripts/analyze_rocpd.py", str(db),
         "--steps", "2", "--markdown"],
        cwd=repo, capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    assert "k1" in out.stdout and "us/step" in out.stdout


# ---- round-2 additions -----------------------------------------------------

def test_commprobe_disabled_is_noop_and_enabled_records():
    from tf_yarn_amd.utils import commprobe
    commprobe.reset()
    commprobe.disable()
    with commprobe.span("x"):
        pass
    commprobe.add_host_time("y", 1.0)
    assert commprobe.summary() == {}
    commprobe.enable()
    with commprobe.span("x"):
        pass
    commprobe.add_host_time("y", 0.25)
    s = commprobe.summary()
    commprobe.disable()
    assert s["y"]["ms"] == 250.0 and s["y"]["count"] == 1
    assert s["x"]["count"] == 1 and s["x"]["ms"] >= 0.0
    commprobe.reset()


def test_pick_region_bits_heuristic_bounds():
    from tf_yarn_amd import ops
    # bench shape: 26M rows, 1.7M updates -> mid-range bits
    assert 10 <= ops.pick_region_bits(26_000_000, 1_700_000) <= 13
    assert ops.pick_region_bits(100, 1_000_000) == 7      # tiny table
    assert ops.pick_region_bits(1 << 30, 10) == 14        # huge/sparse
    assert ops.pick_region_bits(1000, 0) == 11            # degenerate


def test_emb_bwd_sgd_fused_wide_cpu_reference():
    import torch
    from tf_yarn_amd import ops
    torch.manual_seed(3)
    rows, batch, fan, dim = 500, 40, 5, 8
    table = torch.randn(rows, dim)
    wide = torch.randn(rows, 1)
    rt, rw = table.clone(), wide.clone()
    ids = torch.randint(0, rows, (batch * fan,))
    grad = torch.randn(batch * fan, dim)
    gw = torch.randn(batch)
    ops.emb_bwd_sgd_fused_wide(table, wide, ids, grad, gw, lr=0.1,
                               scale=0.5)
    rt.index_add_(0, ids, grad, alpha=-0.05)
    exp = gw.reshape(-1, 1).expand(-1, fan).reshape(-1)
    rw.reshape(-1).index_add_(0, ids, exp, alpha=-0.05)
    assert torch.allclose(table, rt, atol=1e-5)
    assert torch.allclose(wide, rw, atol=1e-5)


def test_binned_wrappers_cpu_fallback():
    import torch
    from tf_yarn_amd import ops
    torch.manual_seed(4)
    rows, n, dim = 300, 50, 16
    table = torch.randn(rows, dim)
    ref = table.clone()
    ids = torch.randint(0, rows, (n,))
    grad = torch.randn(n, dim)
    ops.emb_bwd_sgd_binned(table, ids, grad, lr=0.2, scale=1.0)
    ref.index_add_(0, ids, grad, alpha=-0.2)
    assert torch.allclose(table, ref, atol=1e-5)


def test_bench_master_port_rank0_picks_free_port(tmp_path, monkeypatch):
    import socket
    import bench
    monkeypatch.setenv("MIYARN_RDV_TAG", "utest")
    monkeypatch.delenv("MASTER_PORT", raising=False)
    # occupy 29500 so the fallback must move past it
    blocker = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    blocker.bind(("127.0.0.1", 29500))
    try:
        bench._ensure_master_port(0)
        port = int(__import__("os").environ["MASTER_PORT"])
        assert 29500 < port < 29600
    finally:
        blocker.close()
        __import__("os").environ.pop("MASTER_PORT", None)


def test_binned_mode_parsing(monkeypatch):
    from tf_yarn_amd.models import sharded_embedding as se
    monkeypatch.setenv("MIYARN_BINNED_SCATTER", "1")
    assert se._binned_mode() == "on"
    monkeypatch.setenv("MIYARN_BINNED_SCATTER", "0")
    assert se._binned_mode() == "off"
    monkeypatch.delenv("MIYARN_BINNED_SCATTER")
    assert se._binned_mode() == "auto"
    monkeypatch.setenv("MIYARN_BINNED_SCATTER", "auto")
    assert se._binned_mode() == "auto"


def test_negotiated_alltoall_mode_states():
    import torch.distributed as dist

    from tf_yarn_amd.models.sharded_embedding import \
        negotiated_alltoall_mode
    # no process group -> unprobed
    assert negotiated_alltoall_mode() == "unprobed"
    import os
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29561",
                      RANK="0", WORLD_SIZE="1")
    dist.init_process_group("gloo")
    try:
        assert negotiated_alltoall_mode() == "gloo-emulate"
    finally:
        dist.destroy_process_group()
        for k in ("MASTER_ADDR", "MASTER_PORT", "RANK", "WORLD_SIZE"):
            os.environ.pop(k, None)
