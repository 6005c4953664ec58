"""Control-plane key-value store.

Single-node replacement for the skein application KV store the reference
routes every coordination primitive through (reference
``tf_yarn/event.py:13-79``, ``client.py:633-657``): barriers, master
election, cluster-spec exchange, exception propagation and lifecycle
timing all become ``put`` / blocking ``wait`` / prefix ``watch`` against
this server.

Two interchangeable servers speak one language-neutral binary protocol
(documented in ``csrc_kv/kv_server.cpp``):

* the native C++ server (``tf_yarn_amd._kv_native``, built by setup.py) —
  the skein-ApplicationMaster-equivalent runtime component, used when the
  extension is available;
* a pure-Python threaded fallback (:class:`PyKVServer`) with identical
  semantics, used in GPU-less/dev environments and as the reference
  implementation for tests.

``KVServer()`` picks the native one automatically.  Values are opaque
``bytes`` (the skein KV contract); blocking waits are served server-side;
``watch`` upgrades a connection into a push stream of PUT events.
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
from typing import Dict, Generator, List, Optional, Tuple

logger = logging.getLogger(__name__)

# ops
OP_PUT, OP_GET, OP_WAIT, OP_DEL, OP_LIST, OP_WATCH, OP_ADD, OP_CAS = \
    range(1, 9)
# response status
ST_OK, ST_OK_EMPTY, ST_NOT_FOUND, ST_TIMEOUT, ST_ERROR = range(5)
EV_PUT, EV_CLOSED = 10, 11

MAX_FRAME = 1 << 30


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("kv connection closed")
        buf += chunk
    return bytes(buf)


def _request(op: int, key: str, payload: bytes = b"") -> bytes:
    kb = key.encode()
    frame = 1 + 2 + len(kb) + 8 + len(payload)
    return (struct.pack("<IBH", frame, op, len(kb)) + kb
            + struct.pack("<Q", len(payload)) + payload)


def _read_response(sock: socket.socket) -> Tuple[int, bytes]:
    (frame,) = struct.unpack("<I", _recv_exact(sock, 4))
    if frame > MAX_FRAME:
        raise ValueError(f"kv frame too large: {frame}")
    body = _recv_exact(sock, frame)
    status = body[0]
    (plen,) = struct.unpack("<Q", body[1:9])
    return status, body[9:9 + plen]


# ---------------------------------------------------------------------------
# Pure-Python fallback server (same wire protocol as the C++ server)
# ---------------------------------------------------------------------------

class PyKVServer:
    """Threaded TCP KV server with blocking wait and prefix watch."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self._data: Dict[str, bytes] = {}
        self._cond = threading.Condition()
        self._watchers: List[Tuple[str, socket.socket, threading.Lock]] = []
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind((host, port))
        self._sock.listen(128)
        self._addr = f"{host}:{self._sock.getsockname()[1]}"
        self._running = True
        threading.Thread(target=self._accept_loop, name="kv-accept",
                         daemon=True).start()

    @property
    def address(self) -> str:
        return self._addr

    def _accept_loop(self) -> None:
        while self._running:
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    @staticmethod
    def _respond(conn, lock, status: int, payload: bytes = b"") -> None:
        frame = 1 + 8 + len(payload)
        data = (struct.pack("<IB", frame, status)
                + struct.pack("<Q", len(payload)) + payload)
        if lock:
            with lock:
                conn.sendall(data)
        else:
            conn.sendall(data)

    @staticmethod
    def _event_frame(etype: int, key: str = "", value: bytes = b"") -> bytes:
        if etype == EV_CLOSED:
            return struct.pack("<IB", 1, etype)
        kb = key.encode()
        frame = 1 + 2 + len(kb) + 8 + len(value)
        return (struct.pack("<IBH", frame, etype, len(kb)) + kb
                + struct.pack("<Q", len(value)) + value)

    def _serve_conn(self, conn: socket.socket) -> None:
        send_lock = threading.Lock()
        keep_open = False
        try:
            while True:
                (frame,) = struct.unpack("<I", _recv_exact(conn, 4))
                body = _recv_exact(conn, frame)
                op = body[0]
                (klen,) = struct.unpack("<H", body[1:3])
                key = body[3:3 + klen].decode()
                (plen,) = struct.unpack("<Q", body[3 + klen:11 + klen])
                payload = body[11 + klen:11 + klen + plen]

                if op == OP_PUT:
                    self._put(key, payload)
                    self._respond(conn, send_lock, ST_OK_EMPTY)
                elif op == OP_GET:
                    with self._cond:
                        v = self._data.get(key)
                    if v is None:
                        self._respond(conn, send_lock, ST_NOT_FOUND)
                    else:
                        self._respond(conn, send_lock, ST_OK, v)
                elif op == OP_WAIT:
                    (timeout_ms,) = struct.unpack("<Q", payload[:8])
                    timeout = timeout_ms / 1000.0 if timeout_ms else None
                    with self._cond:
                        ok = self._cond.wait_for(
                            lambda: key in self._data or not self._running,
                            timeout=timeout)
                        v = self._data.get(key) if ok else None
                    if v is not None:
                        self._respond(conn, send_lock, ST_OK, v)
                    else:
                        self._respond(conn, send_lock, ST_TIMEOUT)
                elif op == OP_DEL:
                    with self._cond:
                        self._data.pop(key, None)
                    self._respond(conn, send_lock, ST_OK_EMPTY)
                elif op == OP_LIST:
                    out = bytearray()
                    with self._cond:
                        for k, v in self._data.items():
                            if not k.startswith(key):
                                continue
                            kb = k.encode()
                            out += struct.pack("<H", len(kb)) + kb
                            out += struct.pack("<Q", len(v)) + v
                    self._respond(conn, send_lock, ST_OK, bytes(out))
                elif op == OP_WATCH:
                    with self._cond:
                        existing = [(k, v) for k, v in self._data.items()
                                    if k.startswith(key)]
                        self._watchers.append((key, conn, send_lock))
                    self._respond(conn, send_lock, ST_OK_EMPTY)
                    with send_lock:
                        for k, v in existing:
                            conn.sendall(self._event_frame(EV_PUT, k, v))
                    keep_open = True
                    return
                elif op == OP_ADD:
                    (amount,) = struct.unpack("<q", payload[:8])
                    with self._cond:
                        cur = int(self._data.get(key, b"0") or b"0")
                        cur += amount
                        sval = str(cur).encode()
                        self._data[key] = sval
                        self._cond.notify_all()
                    self._notify(key, sval)
                    self._respond(conn, send_lock, ST_OK, sval)
                elif op == OP_CAS:
                    (elen,) = struct.unpack("<Q", payload[:8])
                    expected = payload[8:8 + elen]
                    desired = payload[8 + elen:]
                    with self._cond:
                        cur = self._data.get(key)
                        if cur == expected or (cur is None
                                               and expected == b""):
                            self._data[key] = desired
                            result, changed = desired, True
                            self._cond.notify_all()
                        else:
                            result = cur if cur is not None else expected
                            changed = False
                    if changed:
                        self._notify(key, desired)
                    self._respond(conn, send_lock, ST_OK, result)
                else:
                    self._respond(conn, send_lock, ST_ERROR,
                                  b"unknown op")
        except (ConnectionError, OSError, struct.error):
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
        self._notify(key, value)

    def _notify(self, key: str, value: bytes) -> None:
        with self._cond:
            watchers = list(self._watchers)
        dead = []
        for prefix, wsock, wlock in watchers:
            if key.startswith(prefix):
                try:
                    with wlock:
                        wsock.sendall(
                            self._event_frame(EV_PUT, key, value))
                except OSError:
                    dead.append((prefix, wsock, wlock))
        if dead:
            with self._cond:
                for w in dead:
                    if w in self._watchers:
                        self._watchers.remove(w)

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
                    wsock.sendall(self._event_frame(EV_CLOSED))
                wsock.close()
            except OSError:
                pass


class NativeKVServer:
    """Wrapper around the C++ server (tf_yarn_amd._kv_native)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        from tf_yarn_amd import _kv_native
        self._impl = _kv_native.KvServer()
        actual = self._impl.start(host, port)
        self._addr = f"{host}:{actual}"

    @property
    def address(self) -> str:
        return self._addr

    def items(self) -> Dict[str, bytes]:
        return KVClient(self._addr).list("")

    def stop(self) -> None:
        self._impl.stop()


def _native_available() -> bool:
    try:
        from tf_yarn_amd import _kv_native  # noqa: F401
        return True
    except ImportError:
        return False


def KVServer(host: str = "127.0.0.1", port: int = 0):
    """Factory: native C++ server when built, Python fallback otherwise."""
    if _native_available():
        return NativeKVServer(host, port)
    return PyKVServer(host, port)


# ---------------------------------------------------------------------------
# Client
# ---------------------------------------------------------------------------

class KVClient:
    """Client for the KV server (either implementation).  Thread-safe."""

    def __init__(self, address: str):
        self.address = address
        host, port = address.rsplit(":", 1)
        self._host, self._port = host, int(port)
        self._lock = threading.Lock()
        self._sock = self._connect()

    def _connect(self) -> socket.socket:
        sock = socket.create_connection((self._host, self._port),
                                        timeout=60)
        sock.settimeout(None)
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        return sock

    def _call(self, op: int, key: str,
              payload: bytes = b"") -> Tuple[int, bytes]:
        with self._lock:
            self._sock.sendall(_request(op, key, payload))
            return _read_response(self._sock)

    def put(self, key: str, value: bytes) -> None:
        if not isinstance(value, (bytes, bytearray)):
            raise TypeError(f"kv values are bytes, got {type(value)}")
        status, _ = self._call(OP_PUT, key, bytes(value))
        if status == ST_ERROR:
            raise RuntimeError("kv put failed")

    def get(self, key: str) -> Optional[bytes]:
        status, payload = self._call(OP_GET, key)
        return payload if status == ST_OK else None

    def wait(self, key: str, timeout: Optional[float] = None) -> bytes:
        """Block until *key* exists (server-side wait, no polling)."""
        timeout_ms = int(timeout * 1000) if timeout else 0
        if timeout is not None and timeout_ms == 0:
            timeout_ms = 1
        sock = self._connect()  # long waits must not serialize other calls
        try:
            sock.sendall(_request(OP_WAIT, key,
                                  struct.pack("<Q", timeout_ms)))
            status, payload = _read_response(sock)
        finally:
            sock.close()
        if status != ST_OK:
            raise TimeoutError(
                f"kv wait timed out after {timeout}s: {key!r}")
        return payload

    def add(self, key: str, amount: int) -> int:
        status, payload = self._call(OP_ADD, key,
                                     struct.pack("<q", amount))
        if status != ST_OK:
            raise RuntimeError("kv add failed")
        return int(payload)

    def compare_set(self, key: str, expected: bytes,
                    desired: bytes) -> bytes:
        payload = (struct.pack("<Q", len(expected)) + bytes(expected)
                   + bytes(desired))
        status, result = self._call(OP_CAS, key, payload)
        if status != ST_OK:
            raise RuntimeError("kv cas failed")
        return result

    def delete(self, key: str) -> None:
        self._call(OP_DEL, key)

    def list(self, prefix: str = "") -> Dict[str, bytes]:
        status, payload = self._call(OP_LIST, prefix)
        out: Dict[str, bytes] = {}
        o = 0
        while o < len(payload):
            (klen,) = struct.unpack_from("<H", payload, o)
            o += 2
            k = payload[o:o + klen].decode()
            o += klen
            (vlen,) = struct.unpack_from("<Q", payload, o)
            o += 8
            out[k] = payload[o:o + vlen]
            o += vlen
        return out

    def events(self, prefix: str = ""
               ) -> Generator[Tuple[str, bytes], None, None]:
        """Yield (key, value) for every PUT matching *prefix*; existing
        keys replay first; ends on server shutdown.  Own socket."""
        sock = self._connect()
        try:
            sock.sendall(_request(OP_WATCH, prefix))
            status, _ = _read_response(sock)
            assert status == ST_OK_EMPTY
            while True:
                (frame,) = struct.unpack("<I", _recv_exact(sock, 4))
                body = _recv_exact(sock, frame)
                etype = body[0]
                if etype == EV_CLOSED:
                    return
                (klen,) = struct.unpack("<H", body[1:3])
                key = body[3:3 + klen].decode()
                (vlen,) = struct.unpack("<Q", body[3 + klen:11 + klen])
                yield key, body[11 + klen:11 + klen + vlen]
        except (ConnectionError, OSError, struct.error):
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
