"""Concurrency stress for the control-plane KV servers (native C++ and
Python fallback must both survive many clients, mixed ops, watchers)."""

import threading
import time

import pytest

from tf_yarn_amd.kv import KVClient, KVServer, PyKVServer, _native_available


def _stress(server, n_clients=16, n_ops=50):
    errors = []
    watch_got = []
    stop_watch = threading.Event()

    def watcher():
        c = KVClient(server.address)
        for k, v in c.events("w:"):
            watch_got.append(k)
            if len(watch_got) >= n_clients:
                stop_watch.set()
                return

    wt = threading.Thread(target=watcher, daemon=True)
    wt.start()
    time.sleep(0.2)

    def client_body(cid):
        try:
            c = KVClient(server.address)
            for i in range(n_ops):
                c.put(f"k:{cid}:{i}", bytes([cid]) * (i + 1))
            for i in range(0, n_ops, 7):
                assert c.get(f"k:{cid}:{i}") == bytes([cid]) * (i + 1)
            total = c.add("counter", 1)
            assert 1 <= total <= n_clients
            # waiter rendezvous: every client waits for the last's marker
            c.put(f"w:{cid}", b"x")
            c.wait("w:0", timeout=30)
            assert len(c.list(f"k:{cid}:")) == n_ops
        except Exception as e:  # noqa: BLE001
            errors.append((cid, repr(e)))

    threads = [threading.Thread(target=client_body, args=(i,))
               for i in range(n_clients)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors
    assert stop_watch.wait(timeout=10), \
        f"watcher saw {len(watch_got)}/{n_clients}"
    c = KVClient(server.address)
    assert c.get("counter") == str(n_clients).encode()


@pytest.mark.timeout(180)
def test_stress_default_server():
    server = KVServer()
    try:
        _stress(server)
    finally:
        server.stop()


@pytest.mark.timeout(180)
def test_stress_python_fallback():
    server = PyKVServer()
    try:
        _stress(server)
    finally:
        server.stop()


def test_native_vs_python_throughput():
    """Report ops/s for both servers (native should not be slower)."""
    results = {}
    for name, server in [("python", PyKVServer())] + (
            [("native", KVServer())] if _native_available() else []):
        try:
            c = KVClient(server.address)
            n = 2000
            t0 = time.perf_counter()
            for i in range(n):
                c.put(f"p:{i % 97}", b"v" * 64)
            dt = time.perf_counter() - t0
            results[name] = n / dt
        finally:
            server.stop()
    print("kv put ops/s:", {k: f"{v:,.0f}" for k, v in results.items()})
    if "native" in results:
        assert results["native"] > 0.5 * results["python"]


def test_chaos_mixed_ops_server_stays_consistent():
    """N threads hammer one server with a random mix of put/get/wait/
    list/delete/add/cas for ~2 s; the server must stay alive and every
    ADD must be accounted for exactly once."""
    import random
    import threading

    from tf_yarn_amd.kv import KVClient, KVServer

    server = KVServer()
    n_threads = 8
    adds_per_thread = 50
    errors = []

    def worker(tid):
        try:
            client = KVClient(server.address)
            rng = random.Random(tid)
            for i in range(adds_per_thread):
                op = rng.randrange(5)
                key = f"chaos/{rng.randrange(20)}"
                if op == 0:
                    client.put(key, f"v{tid}:{i}".encode())
                elif op == 1:
                    client.get(key)
                elif op == 2:
                    client.list("chaos/")
                elif op == 3 and rng.random() < 0.3:
                    client.delete(key)
                elif op == 4:
                    client.compare_set(key, b"x", b"y")
                client.add("chaos_counter", 1)  # always
        except Exception as e:  # noqa: BLE001
            errors.append((tid, repr(e)))

    threads = [threading.Thread(target=worker, args=(t,))
               for t in range(n_threads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    try:
        assert not errors, errors
        client = KVClient(server.address)
        total = client.add("chaos_counter", 0)
        assert total == n_threads * adds_per_thread, total
    finally:
        server.stop()
