"""Key-value store abstraction
(reference: bagua/torch_api/contrib/utils/store.py).

``Store`` is the ABC; ``ClusterStore`` shards keys across store instances
by xxhash64. Concrete backends: :class:`InMemoryStore` (per-process) and
:class:`bagua_amd.contrib.utils.tcp_store.TcpStore` (per-node server —
the redis replacement; this image has no redis)."""

from typing import Dict, List, Optional, Union

__all__ = ["Store", "ClusterStore", "InMemoryStore"]


class Store:
    """Base class for key-value stores."""

    def set(self, key: str, value: Union[str, bytes]):
        raise NotImplementedError

    def get(self, key: str) -> Optional[Union[str, bytes]]:
        raise NotImplementedError

    def num_keys(self) -> int:
        raise NotImplementedError

    def clear(self):
        raise NotImplementedError

    def mset(self, dictionary: Dict[str, Union[str, bytes]]):
        raise NotImplementedError

    def mget(self, keys: List[str]) -> List[Optional[Union[str, bytes]]]:
        raise NotImplementedError

    def status(self) -> bool:
        raise NotImplementedError

    def shutdown(self):
        pass


class InMemoryStore(Store):
    """Single-process dict-backed store (tests, single-node cache)."""

    def __init__(self):
        self._d: Dict[str, bytes] = {}

    @staticmethod
    def _to_bytes(v) -> bytes:
        return v.encode() if isinstance(v, str) else bytes(v)

    def set(self, key, value):
        self._d[key] = self._to_bytes(value)

    def get(self, key):
        return self._d.get(key)

    def num_keys(self):
        return len(self._d)

    def clear(self):
        self._d.clear()

    def mset(self, dictionary):
        for k, v in dictionary.items():
            self.set(k, v)

    def mget(self, keys):
        return [self._d.get(k) for k in keys]

    def status(self):
        return True


class ClusterStore(Store):
    """Shards entries across stores by xxhash64 of the key
    (reference: utils/store.py:56-120)."""

    def __init__(self, stores: List[Store]):
        self.stores = stores
        self.num_stores = len(stores)
        import xxhash

        self.hash_fn = lambda x: xxhash.xxh64(x).intdigest()

    def _route(self, key: str) -> Store:
        if self.num_stores == 1:
            return self.stores[0]
        return self.stores[self.hash_fn(key.encode()) % self.num_stores]

    def set(self, key, value):
        self._route(key).set(key, value)

    def get(self, key):
        return self._route(key).get(key)

    def num_keys(self):
        return sum(s.num_keys() for s in self.stores)

    def clear(self):
        for s in self.stores:
            s.clear()

    def mset(self, dictionary):
        if self.num_stores == 1:
            return self.stores[0].mset(dictionary)
        buckets: Dict[int, Dict] = {}
        for k, v in dictionary.items():
            idx = self.hash_fn(k.encode()) % self.num_stores
            buckets.setdefault(idx, {})[k] = v
        for idx, d in buckets.items():
            self.stores[idx].mset(d)

    def mget(self, keys):
        if self.num_stores == 1:
            return self.stores[0].mget(keys)
        by_store: Dict[int, List[str]] = {}
        for k in keys:
            by_store.setdefault(
                self.hash_fn(k.encode()) % self.num_stores, []).append(k)
        results: Dict[str, Optional[bytes]] = {}
        for idx, ks in by_store.items():
            for k, v in zip(ks, self.stores[idx].mget(ks)):
                results[k] = v
        return [results[k] for k in keys]

    def status(self):
        return all(s.status() for s in self.stores)

    def shutdown(self):
        for s in self.stores:
            s.shutdown()
