"""Sample cache with batched write-behind
(reference: bagua/torch_api/contrib/cache_loader.py:40-140).

``CacheLoader.get(key, load_fn)`` serves from the KV store when present,
otherwise computes via ``load_fn`` and queues the value for a batched
``mset`` (``writer_buffer_size`` entries per flush). Backends:

* ``"inmemory"`` — per-process dict (single-worker);
* ``"tcp"`` — the node-local TCP store (redis replacement; pass
  ``hosts=[{"host":..., "port":...}]`` or ``bootstrap_server=True``).
"""

import pickle
from collections import OrderedDict
from typing import Callable

from .utils.store import InMemoryStore
from .utils.tcp_store import ClusterTcpStore, TcpStore

__all__ = ["CacheLoader"]


def serialize(obj) -> bytes:
    return pickle.dumps(obj)


def deserialize(data: bytes):
    return pickle.loads(data)


class BatchFetcher:
    """Write-buffering front of a store
    (reference: cache_loader.py:97-140)."""

    def __init__(self, store, read_buffer_size: int,
                 writer_buffer_size: int):
        self.store = store
        self.writer_buffer_size = max(1, writer_buffer_size)
        self.write_buf = OrderedDict()
        self.write_counter = 0
        self.read_counter = 0

    def read(self, key: str):
        self.read_counter += 1
        if key in self.write_buf:
            return deserialize(self.write_buf[key])
        data = self.store.get(key)
        return deserialize(data) if data is not None else None

    def write(self, key: str, value):
        self.write_counter += 1
        self.write_buf[key] = serialize(value)
        if len(self.write_buf) >= self.writer_buffer_size:
            self.flush_write_buffer()

    def flush_write_buffer(self):
        if self.write_buf:
            self.store.mset(dict(self.write_buf))
            self.write_buf.clear()


class CacheLoader:
    def __init__(self, backend: str = "tcp", dataset_name: str = "",
                 writer_buffer_size: int = 20, **kwargs):
        self.backend = backend
        self.dataset_name = dataset_name
        if backend == "inmemory":
            self.store = InMemoryStore()
        elif backend == "tcp":
            hosts = kwargs.get("hosts")
            if hosts:
                self.store = ClusterTcpStore(hosts)
            else:
                self.store = TcpStore(
                    kwargs.get("host", "127.0.0.1"),
                    kwargs.get("port", 0),
                    bootstrap_server=kwargs.get("bootstrap_server", True))
        else:
            raise ValueError("unknown cache backend %r (use 'inmemory' or "
                             "'tcp')" % backend)
        self.fetcher = BatchFetcher(self.store, 1, writer_buffer_size)

    def get(self, key, load_fn: Callable):
        cache_key = "{}_{}".format(self.dataset_name, key)
        value = self.fetcher.read(cache_key)
        if value is None:
            value = load_fn(key)
            self.fetcher.write(cache_key, value)
        return value

    def num_keys(self) -> int:
        self.fetcher.flush_write_buffer()
        return self.store.num_keys()

    def cleanup(self):
        self.store.shutdown()
