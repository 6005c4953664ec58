"""Control-plane key-value store.

Single-node replacement for the skein application KV store the reference routes
every coordination primitive through (reference ``tf_yarn/event.py:13-79``,
``client.py:633-657``): barriers, master election, cluster-spec exchange,
exception propagation and lifecycle timing all become ``put`` / blocking
``wait`` / prefix ``watch`` against this server.

Design: one TCP server owned by the client (launcher) process; every task
process connects with :class:`KVClient`.  Values are opaque ``bytes`` (same
contract as skein's KV).  Blocking waits are served server-side with a
condition variable so clients need no polling.  ``watch`` upgrades a
connection into a push stream of ``(key, value)`` PUT events, which the
launcher's event-aggregation thread consumes (reference ``client.py:633``).

Wire protocol: 4-byte big-endian length + pickled tuple.  The store only ever
binds to 127.0.0.1 and is torn down with the application.
"""

from __future__ import annotations

import logging
import pickle
import socket
import struct
import threading
from typing import Dict, Generator, List, Optional, Tuple

logger = logging.getLogger(__name__)

_LEN = struct.Struct("!I")
MAX_FRAME = 1 << 30


def _send_frame(sock: socket.socket, obj) -> None:
    payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(_LEN.pack(len(payload)) + payload)


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("kv connection closed")
        buf += chunk
    return bytes(buf)


def _recv_frame(sock: socket.socket):
    (n,) = _LEN.unpack(_recv_exact(sock, 4))
    if n > MAX_FRAME:
        raise ValueError(f"kv frame too large: {n}")
    return pickle.loads(_recv_exact(sock, n))


class KVServer:
    """Threaded TCP KV server with blocking wait and prefix watch."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self._data: Dict[str, bytes] = {}
        self._cond = threading.Condition()
        # watcher: (prefix, sock, send_lock)
        self._watchers: List[Tuple[str, socket.socket, threading.Lock]] = []
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind((host, port))
        self._sock.listen(128)
        self._addr = f"{host}:{self._sock.getsockname()[1]}"
        self._running = True
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name="kv-accept", daemon=True)
        self._accept_thread.start()

    @property
    def address(self) -> str:
        return self._addr

    # -- server internals ---------------------------------------------------

    def _accept_loop(self) -> None:
        while self._running:
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    def _serve_conn(self, conn: socket.socket) -> None:
        keep_open = False
        try:
            while True:
                req = _recv_frame(conn)
                op = req[0]
                if op == "put":
                    self._put(req[1], req[2])
                    _send_frame(conn, ("ok", None))
                elif op == "get":
                    with self._cond:
                        _send_frame(conn, ("ok", self._data.get(req[1])))
                elif op == "wait":
                    key, timeout = req[1], req[2]
                    with self._cond:
                        ok = self._cond.wait_for(
                            lambda: key in self._data or not self._running,
                            timeout=timeout)
                        if ok and key in self._data:
                            _send_frame(conn, ("ok", self._data[key]))
                        else:
                            _send_frame(conn, ("timeout", None))
                elif op == "add":
                    # Atomic counter (c10d Store.add semantics): value is
                    # stored as its decimal-string bytes.
                    key, amount = req[1], req[2]
                    with self._cond:
                        current = int(self._data.get(key, b"0") or b"0")
                        current += amount
                        self._data[key] = str(current).encode()
                        self._cond.notify_all()
                    self._notify_watchers(key, str(current).encode())
                    _send_frame(conn, ("ok", current))
                elif op == "cas":
                    # compare_set (c10d semantics): set to desired iff the
                    # current value equals expected, or the key is absent
                    # and expected is empty.  Returns the resulting value.
                    key, expected, desired = req[1], req[2], req[3]
                    with self._cond:
                        cur = self._data.get(key)
                        if (cur == expected
                                or (cur is None and expected == b"")):
                            self._data[key] = desired
                            self._cond.notify_all()
                            result = desired
                            changed = True
                        else:
                            result = cur if cur is not None else expected
                            changed = False
                    if changed:
                        self._notify_watchers(key, desired)
                    _send_frame(conn, ("ok", result))
                elif op == "del":
                    with self._cond:
                        self._data.pop(req[1], None)
                    _send_frame(conn, ("ok", None))
                elif op == "list":
                    prefix = req[1]
                    with self._cond:
                        items = {k: v for k, v in self._data.items()
                                 if k.startswith(prefix)}
                    _send_frame(conn, ("ok", items))
                elif op == "watch":
                    prefix = req[1]
                    lock = threading.Lock()
                    with self._cond:
                        # Replay existing matching keys first so a late
                        # watcher misses nothing (skein event-stream parity).
                        existing = [(k, v) for k, v in self._data.items()
                                    if k.startswith(prefix)]
                        self._watchers.append((prefix, conn, lock))
                    _send_frame(conn, ("ok", None))
                    with lock:
                        for k, v in existing:
                            _send_frame(conn, ("event", k, v))
                    # Connection now belongs to the push stream: hand it
                    # over to _put / stop without closing it here.
                    keep_open = True
                    return
                else:
                    _send_frame(conn, ("err", f"unknown op {op!r}"))
        except (ConnectionError, EOFError, OSError):
            pass
        finally:
            if not keep_open:
                try:
                    conn.close()
                except OSError:
                    pass

    def _put(self, key: str, value: bytes) -> None:
        with self._cond:
            self._data[key] = value
            self._cond.notify_all()
        self._notify_watchers(key, value)

    def _notify_watchers(self, key: str, value: bytes) -> None:
        with self._cond:
            watchers = list(self._watchers)
        dead = []
        for prefix, wsock, wlock in watchers:
            if key.startswith(prefix):
                try:
                    with wlock:
                        _send_frame(wsock, ("event", key, value))
                except OSError:
                    dead.append((prefix, wsock, wlock))
        if dead:
            with self._cond:
                for w in dead:
                    if w in self._watchers:
                        self._watchers.remove(w)

    # -- lifecycle ----------------------------------------------------------

    def items(self) -> Dict[str, bytes]:
        with self._cond:
            return dict(self._data)

    def stop(self) -> None:
        self._running = False
        with self._cond:
            self._cond.notify_all()
            watchers = list(self._watchers)
        try:
            self._sock.close()
        except OSError:
            pass
        for _, wsock, wlock in watchers:
            try:
                with wlock:
                    _send_frame(wsock, ("closed",))
                wsock.close()
            except OSError:
                pass


class KVClient:
    """Client for :class:`KVServer`.  Thread-safe; one socket per client."""

    def __init__(self, address: str):
        self.address = address
        host, port = address.rsplit(":", 1)
        self._host, self._port = host, int(port)
        self._lock = threading.Lock()
        self._sock = self._connect()

    def _connect(self) -> socket.socket:
        sock = socket.create_connection((self._host, self._port), timeout=60)
        sock.settimeout(None)
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        return sock

    def _call(self, *req):
        with self._lock:
            _send_frame(self._sock, req)
            status, value = _recv_frame(self._sock)
        if status == "timeout":
            raise TimeoutError(f"kv wait timed out: {req[1]!r}")
        if status == "err":
            raise RuntimeError(value)
        return value

    def put(self, key: str, value: bytes) -> None:
        if not isinstance(value, (bytes, bytearray)):
            raise TypeError(f"kv values are bytes, got {type(value)}")
        self._call("put", key, bytes(value))

    def get(self, key: str) -> Optional[bytes]:
        return self._call("get", key)

    def wait(self, key: str, timeout: Optional[float] = None) -> bytes:
        """Block until *key* exists (server-side wait, no polling)."""
        # Dedicated socket: a long wait must not serialize other calls.
        sock = self._connect()
        try:
            _send_frame(sock, ("wait", key, timeout))
            status, value = _recv_frame(sock)
        finally:
            sock.close()
        if status == "timeout":
            raise TimeoutError(f"kv wait timed out after {timeout}s: {key!r}")
        return value

    def add(self, key: str, amount: int) -> int:
        """Atomically add to a decimal counter; returns the new value."""
        return self._call("add", key, amount)

    def compare_set(self, key: str, expected: bytes,
                    desired: bytes) -> bytes:
        """Set *key* to *desired* iff its value equals *expected* (or the
        key is absent and expected is empty); returns the resulting value."""
        return self._call("cas", key, bytes(expected), bytes(desired))

    def delete(self, key: str) -> None:
        self._call("del", key)

    def list(self, prefix: str = "") -> Dict[str, bytes]:
        return self._call("list", prefix)

    def events(self, prefix: str = "") -> Generator[Tuple[str, bytes], None, None]:
        """Yield (key, value) for every PUT matching *prefix*.

        Existing keys are replayed first.  The generator ends when the server
        shuts down.  Runs on its own socket.
        """
        sock = self._connect()
        try:
            _send_frame(sock, ("watch", prefix))
            status, _ = _recv_frame(sock)
            assert status == "ok"
            while True:
                msg = _recv_frame(sock)
                if msg[0] == "closed":
                    return
                yield msg[1], msg[2]
        except (ConnectionError, OSError):
            return
        finally:
            try:
                sock.close()
            except OSError:
                pass

    def close(self) -> None:
        try:
            self._sock.close()
        except OSError:
            pass
