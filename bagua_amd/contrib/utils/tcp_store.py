"""TCP key-value store — the redis replacement.

The reference spawned one redis-server per node and connected a sharded
ClusterStore over them (bagua/torch_api/contrib/utils/redis_store.py:46-99).
redis isn't in this image, so this build ships its own node-local KV
server: a stdlib ThreadingTCPServer speaking a tiny length-prefixed
msgpack protocol. Functionally equivalent for the cache-loader workload
(set/get/mset/mget/num_keys/clear) and dependency-free.

Wire format: 4-byte big-endian length + msgpack [op, args...].
"""

import socket
import socketserver
import struct
import threading
from typing import Dict, List, Union

import msgpack

from .store import ClusterStore, Store

__all__ = ["TcpStore", "TcpStoreServer", "ClusterTcpStore",
           "start_store_server"]


def _recv_exact(sock, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("store connection closed")
        buf += chunk
    return buf


def _send_msg(sock, obj):
    payload = msgpack.packb(obj, use_bin_type=True)
    sock.sendall(struct.pack(">I", len(payload)) + payload)


def _recv_msg(sock):
    (length,) = struct.unpack(">I", _recv_exact(sock, 4))
    return msgpack.unpackb(_recv_exact(sock, length), raw=False)


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        data: Dict[str, bytes] = self.server.kv  # type: ignore
        lock: threading.Lock = self.server.kv_lock  # type: ignore
        try:
            while True:
                msg = _recv_msg(self.request)
                op = msg[0]
                if op == "set":
                    with lock:
                        data[msg[1]] = msg[2]
                    _send_msg(self.request, ["ok"])
                elif op == "get":
                    with lock:
                        _send_msg(self.request, ["ok", data.get(msg[1])])
                elif op == "mset":
                    with lock:
                        data.update(msg[1])
                    _send_msg(self.request, ["ok"])
                elif op == "mget":
                    with lock:
                        _send_msg(self.request,
                                  ["ok", [data.get(k) for k in msg[1]]])
                elif op == "num_keys":
                    with lock:
                        _send_msg(self.request, ["ok", len(data)])
                elif op == "clear":
                    with lock:
                        data.clear()
                    _send_msg(self.request, ["ok"])
                elif op == "ping":
                    _send_msg(self.request, ["ok"])
                else:
                    _send_msg(self.request, ["err", "unknown op"])
        except (ConnectionError, OSError):
            pass


class TcpStoreServer:
    """Node-local KV server; one per node, like the reference's managed
    redis-server instance."""

    def __init__(self, host: str = "0.0.0.0", port: int = 0):
        self.server = socketserver.ThreadingTCPServer(
            (host, port), _Handler, bind_and_activate=True)
        self.server.daemon_threads = True
        self.server.kv = {}
        self.server.kv_lock = threading.Lock()
        self.port = self.server.server_address[1]
        self.thread = threading.Thread(target=self.server.serve_forever,
                                       daemon=True)
        self.thread.start()

    def shutdown(self):
        self.server.shutdown()
        self.thread.join(timeout=5)


class TcpStore(Store):
    """Client for one TcpStoreServer."""

    def __init__(self, host: str, port: int, bootstrap_server: bool = False):
        self._server = None
        if bootstrap_server:
            self._server = TcpStoreServer(port=port if port else 0)
            host, port = "127.0.0.1", self._server.port
        self.host, self.port = host, port
        self._sock = None
        self._lock = threading.Lock()

    def _conn(self):
        if self._sock is None:
            self._sock = socket.create_connection((self.host, self.port),
                                                  timeout=30)
        return self._sock

    def _call(self, *msg):
        with self._lock:
            try:
                s = self._conn()
                _send_msg(s, list(msg))
                rsp = _recv_msg(s)
            except (ConnectionError, OSError):
                self._sock = None
                s = self._conn()
                _send_msg(s, list(msg))
                rsp = _recv_msg(s)
        if rsp[0] != "ok":
            raise RuntimeError("store error: %r" % rsp)
        return rsp[1] if len(rsp) > 1 else None

    @staticmethod
    def _b(v: Union[str, bytes]) -> bytes:
        return v.encode() if isinstance(v, str) else bytes(v)

    def set(self, key, value):
        self._call("set", key, self._b(value))

    def get(self, key):
        return self._call("get", key)

    def mset(self, dictionary):
        self._call("mset", {k: self._b(v) for k, v in dictionary.items()})

    def mget(self, keys):
        return self._call("mget", list(keys))

    def num_keys(self):
        return self._call("num_keys")

    def clear(self):
        self._call("clear")

    def status(self):
        try:
            self._call("ping")
            return True
        except Exception:  # noqa: BLE001
            return False

    def shutdown(self):
        if self._sock is not None:
            self._sock.close()
            self._sock = None
        if self._server is not None:
            self._server.shutdown()
            self._server = None


class ClusterTcpStore(ClusterStore):
    """Sharded store over several (host, port) servers
    (reference: RedisStore cluster mode, redis_store.py:46-99)."""

    def __init__(self, hosts: List[Dict], bootstrap_local: bool = False):
        stores = []
        for h in hosts:
            stores.append(TcpStore(h["host"], h["port"]))
        super().__init__(stores)


def start_store_server(port: int = 0) -> TcpStoreServer:
    return TcpStoreServer(port=port)
