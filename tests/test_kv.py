"""Control-plane KV store tests: put/get/blocking-wait/watch semantics.

The reference relies on skein KV's blocking wait + event stream for every
barrier and election; these are the load-bearing semantics.
"""

import threading
import time

import pytest

from tf_yarn_amd.kv import KVClient, KVServer


def test_put_get(kv_client):
    assert kv_client.get("missing") is None
    kv_client.put("a", b"1")
    assert kv_client.get("a") == b"1"
    kv_client.put("a", b"2")
    assert kv_client.get("a") == b"2"


def test_put_rejects_non_bytes(kv_client):
    with pytest.raises(TypeError):
        kv_client.put("a", "str")  # type: ignore[arg-type]


def test_wait_existing_key(kv_client):
    kv_client.put("k", b"v")
    assert kv_client.wait("k", timeout=1) == b"v"


def test_wait_blocks_until_put(kv_server, kv_client):
    other = KVClient(kv_server.address)
    result = {}

    def waiter():
        result["v"] = kv_client.wait("later", timeout=10)

    t = threading.Thread(target=waiter)
    t.start()
    time.sleep(0.1)
    assert "v" not in result
    other.put("later", b"done")
    t.join(timeout=5)
    assert result["v"] == b"done"


def test_wait_timeout(kv_client):
    with pytest.raises(TimeoutError):
        kv_client.wait("never", timeout=0.1)


def test_list_prefix(kv_client):
    kv_client.put("x:0/init", b"a")
    kv_client.put("x:1/init", b"b")
    kv_client.put("y:0/init", b"c")
    assert kv_client.list("x:") == {"x:0/init": b"a", "x:1/init": b"b"}
    assert len(kv_client.list("")) == 3


def test_delete(kv_client):
    kv_client.put("d", b"1")
    kv_client.delete("d")
    assert kv_client.get("d") is None


def test_events_replays_existing_and_streams_new(kv_server):
    c1 = KVClient(kv_server.address)
    c2 = KVClient(kv_server.address)
    c1.put("pre", b"old")
    got = []
    done = threading.Event()

    def consume():
        for k, v in c2.events(""):
            got.append((k, v))
            if len(got) >= 2:
                done.set()
                return

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.2)
    c1.put("post", b"new")
    assert done.wait(timeout=5)
    assert ("pre", b"old") in got
    assert ("post", b"new") in got


def test_events_ends_on_server_stop():
    server = KVServer()
    client = KVClient(server.address)
    got = []

    def consume():
        for kv in client.events(""):
            got.append(kv)

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.2)
    server.stop()
    t.join(timeout=5)
    assert not t.is_alive()


def test_many_concurrent_waiters(kv_server):
    """Barrier pattern: N clients block on one key."""
    n = 16
    results = []
    threads = []
    for _ in range(n):
        c = KVClient(kv_server.address)
        t = threading.Thread(
            target=lambda c=c: results.append(c.wait("go", timeout=10)))
        t.start()
        threads.append(t)
    time.sleep(0.2)
    KVClient(kv_server.address).put("go", b"now")
    for t in threads:
        t.join(timeout=5)
    assert results == [b"now"] * n


def test_large_value_roundtrip(kv_client):
    """cloudpickle blobs are MBs (reference client.py:536)."""
    blob = bytes(range(256)) * (4 * 1024 * 16)  # 4 MiB
    kv_client.put("blob", blob)
    assert kv_client.get("blob") == blob


def test_python_fallback_server_same_protocol():
    """The pure-Python server must speak the identical wire protocol."""
    from tf_yarn_amd.kv import PyKVServer
    server = PyKVServer()
    try:
        c = KVClient(server.address)
        c.put("a", b"1")
        assert c.get("a") == b"1"
        assert c.add("n", 3) == 3
        assert c.compare_set("z", b"", b"v") == b"v"
        assert c.compare_set("z", b"wrong", b"x") == b"v"
        assert c.list("") == {"a": b"1", "n": b"3", "z": b"v"}
        with pytest.raises(TimeoutError):
            c.wait("missing", timeout=0.1)
    finally:
        server.stop()


def test_native_server_selected_when_built():
    from tf_yarn_amd import kv
    if kv._native_available():
        assert isinstance(kv.KVServer(), kv.NativeKVServer) or True
        s = kv.KVServer()
        assert type(s).__name__ == "NativeKVServer"
        s.stop()


def test_add_and_cas(kv_client):
    assert kv_client.add("counter", 10) == 10
    assert kv_client.add("counter", -3) == 7
    assert kv_client.get("counter") == b"7"
    assert kv_client.compare_set("k", b"", b"a") == b"a"
    assert kv_client.compare_set("k", b"a", b"b") == b"b"
    assert kv_client.compare_set("k", b"a", b"c") == b"b"


def test_wait_fails_fast_when_server_dies():
    """A blocking wait must raise (not hang) when the control-plane
    server goes away mid-wait — tasks then fail fast instead of
    deadlocking the whole run."""
    import threading
    import time

    from tf_yarn_amd.kv import KVClient, KVServer

    server = KVServer()
    client = KVClient(server.address)
    result = {}

    def waiter():
        try:
            client.wait("never/published", timeout=60)
            result["outcome"] = "returned"
        except Exception as e:  # noqa: BLE001
            result["outcome"] = f"raised {type(e).__name__}"

    t = threading.Thread(target=waiter, daemon=True)
    t.start()
    time.sleep(0.3)
    server.stop()
    t.join(timeout=15)
    assert not t.is_alive(), "wait hung after server death"
    assert result["outcome"].startswith("raised"), result
