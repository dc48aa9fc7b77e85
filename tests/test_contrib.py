"""Contrib: samplers, cache loader, TCP store, sync BN."""

import numpy as np
import pytest
import torch
import torch.nn as nn

from tests.internal.multi_process import run_multi_process


# ---------------------------------------------------------------------------
# load-balancing sampler
# ---------------------------------------------------------------------------


class _VarLenDataset(torch.utils.data.Dataset):
    def __init__(self, n=97, seed=0):
        rng = np.random.RandomState(seed)
        self.lengths = rng.randint(1, 100, size=n).tolist()

    def __getitem__(self, i):
        return self.lengths[i]

    def __len__(self):
        return len(self.lengths)


def test_load_balancing_sampler_balances():
    from bagua_amd.contrib import LoadBalancingDistributedSampler

    ds = _VarLenDataset(101)
    per_rank = []
    for rank in range(4):
        s = LoadBalancingDistributedSampler(
            ds, complexity_fn=lambda x: x, num_replicas=4, rank=rank,
            shuffle=True, seed=7)
        idx = list(iter(s))
        assert len(idx) == len(s)
        per_rank.append(sum(ds[i] for i in idx))
    # complexity totals should be near-equal across ranks (<10% spread)
    assert max(per_rank) - min(per_rank) < 0.1 * max(per_rank)


def test_load_balancing_sampler_epoch_changes_order():
    from bagua_amd.contrib import LoadBalancingDistributedSampler

    ds = _VarLenDataset(64)
    s = LoadBalancingDistributedSampler(
        ds, complexity_fn=lambda x: x, num_replicas=2, rank=0, seed=3)
    a = list(iter(s))
    s.set_epoch(1)
    b = list(iter(s))
    assert a != b


def test_batch_sampler_pads_counts():
    from bagua_amd.contrib import (
        LoadBalancingDistributedBatchSampler,
        LoadBalancingDistributedSampler,
    )

    ds = _VarLenDataset(50)

    def batch_fn(indices):
        # token-budget batching: ragged batch counts across ranks
        batches, cur, budget = [], [], 0
        for i in indices:
            if cur and budget + ds[i] > 150:
                batches.append(cur)
                cur, budget = [], 0
            cur.append(i)
            budget += ds[i]
        if cur:
            batches.append(cur)
        return batches

    lens = set()
    for rank in range(2):
        s = LoadBalancingDistributedSampler(
            ds, complexity_fn=lambda x: x, num_replicas=2, rank=rank,
            seed=5)
        bs = LoadBalancingDistributedBatchSampler(s, batch_fn=batch_fn)
        lens.add(len(list(iter(bs))))
    assert len(lens) == 1, "ranks must run equal batch counts"


# ---------------------------------------------------------------------------
# KV store + cache
# ---------------------------------------------------------------------------


def test_tcp_store_roundtrip():
    from bagua_amd.contrib.utils import TcpStore

    store = TcpStore("", 0, bootstrap_server=True)
    try:
        store.set("a", b"1")
        store.mset({"b": b"2", "c": b"3"})
        assert store.get("a") == b"1"
        assert store.mget(["b", "c", "missing"]) == [b"2", b"3", None]
        assert store.num_keys() == 3
        assert store.status()
        store.clear()
        assert store.num_keys() == 0
    finally:
        store.shutdown()


def test_cluster_store_sharding():
    from bagua_amd.contrib.utils import ClusterStore, InMemoryStore

    backends = [InMemoryStore(), InMemoryStore(), InMemoryStore()]
    cs = ClusterStore(list(backends))
    kv = {"k%d" % i: str(i).encode() for i in range(50)}
    cs.mset(kv)
    assert cs.num_keys() == 50
    assert cs.mget(list(kv)) == list(kv.values())
    assert all(b.num_keys() > 0 for b in backends), "sharding degenerate"


def test_cached_dataset():
    from bagua_amd.contrib import CachedDataset

    calls = []

    class DS(torch.utils.data.Dataset):
        def __getitem__(self, i):
            calls.append(i)
            return np.ones(4) * i

        def __len__(self):
            return 10

    cds = CachedDataset(DS(), backend="inmemory", dataset_name="t",
                        writer_buffer_size=1)
    for _ in range(3):
        for i in range(10):
            v = cds[i]
            assert v[0] == i
    assert len(calls) == 10, "cache never hit"
    assert cds.cache_loader.num_keys() == 10


# ---------------------------------------------------------------------------
# sync BN
# ---------------------------------------------------------------------------


def _worker_syncbn(rank, nprocs):
    import bagua_amd
    from bagua_amd.contrib import SyncBatchNorm

    bagua_amd.init_process_group()
    torch.manual_seed(11)
    bn = SyncBatchNorm(3)
    # each rank sees half the global batch
    torch.manual_seed(100)
    full = torch.randn(8, 3, 5, 5)
    local = full.chunk(nprocs)[rank].requires_grad_(True)
    out = bn(local)
    out.pow(2).mean().backward()
    return (out.detach(), bn.running_mean.clone(), bn.running_var.clone(),
            local.grad.clone())


def test_sync_batchnorm_matches_full_batch():
    nprocs = 2
    res = run_multi_process(nprocs, _worker_syncbn)
    # reference: plain BN over the FULL batch
    torch.manual_seed(11)
    bn = nn.BatchNorm2d(3)
    torch.manual_seed(100)
    full = torch.randn(8, 3, 5, 5).requires_grad_(True)
    out = bn(full)
    # distributed loss = mean over ranks of per-rank mean == global mean
    out.pow(2).mean().backward()

    got = torch.cat([res[0][0], res[1][0]])
    assert torch.allclose(got, out.detach(), atol=1e-5)
    assert torch.allclose(res[0][1], bn.running_mean, atol=1e-5)
    assert torch.allclose(res[0][2], bn.running_var, atol=1e-5)
    grads = torch.cat([res[0][3], res[1][3]])
    # per-rank loss normalizes by local count; full-batch loss by global
    # count -> scale grads by nprocs for comparison
    assert torch.allclose(grads / nprocs, full.grad, atol=1e-5)


def test_convert_sync_batchnorm():
    from bagua_amd.contrib import SyncBatchNorm

    m = nn.Sequential(nn.Conv2d(3, 8, 3), nn.BatchNorm2d(8), nn.ReLU(),
                      nn.Sequential(nn.BatchNorm1d(4)))
    conv = SyncBatchNorm.convert_sync_batchnorm(m)
    assert isinstance(conv[1], SyncBatchNorm)
    assert isinstance(conv[3][0], SyncBatchNorm)


def test_tcp_store_concurrent_clients():
    import threading

    from bagua_amd.contrib.utils import TcpStore
    from bagua_amd.contrib.utils.tcp_store import TcpStoreServer

    server = TcpStoreServer(port=0)
    try:
        errors = []

        def client(cid):
            try:
                st = TcpStore("127.0.0.1", server.port)
                for i in range(50):
                    st.set("c%d_k%d" % (cid, i), b"v%d" % i)
                got = st.mget(["c%d_k%d" % (cid, i) for i in range(50)])
                assert got == [b"v%d" % i for i in range(50)]
                st.shutdown()
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        threads = [threading.Thread(target=client, args=(c,))
                   for c in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
        assert not errors, errors
        probe = TcpStore("127.0.0.1", server.port)
        assert probe.num_keys() == 8 * 50
        probe.shutdown()
    finally:
        server.shutdown()
