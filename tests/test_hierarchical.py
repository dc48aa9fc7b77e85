"""Hierarchical communication tested with pseudo-nodes on one machine
(subsets of local ranks act as nodes by overriding the
node-rank env per process)."""

import os

import torch
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


def _worker_hier(rank, nprocs, algo_name):
    # each rank pretends to be its own node -> intra = {self}, inter = all
    os.environ["NODE_RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = "0"

    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    pg = bagua_amd.communication._get_default_group()
    assert pg._get_intra_ranks() == [rank]
    assert pg._get_inter_ranks() == list(range(nprocs))

    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    algorithm = GlobalAlgorithmRegistry.get(algo_name)(hierarchical=True)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)
    for step in range(6):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_hierarchical_gradient_allreduce_consensus():
    out = run_multi_process(2, _worker_hier, args=("gradient_allreduce",))
    assert torch.equal(out[0], out[1]), "hierarchical allreduce diverged"


def test_hierarchical_bytegrad_consensus():
    out = run_multi_process(2, _worker_hier, args=("bytegrad",))
    assert torch.equal(out[0], out[1]), "hierarchical bytegrad diverged"


def _worker_hier_matches_flat(rank, nprocs):
    """Pseudo-node hierarchical allreduce must equal the flat result."""
    os.environ["NODE_RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = "0"

    import bagua_amd
    from bagua_amd.bucket import BaguaBucket, CentralizedSyncOp
    from bagua_amd.executor import execute_ops
    from bagua_amd.tensor import ensure_bagua_tensor

    bagua_amd.init_process_group()
    group = bagua_amd.communication._get_default_group()

    torch.manual_seed(50 + rank)
    t = torch.randn(64)
    expect = None
    # flat average across ranks computed out-of-band
    probe = t.clone()
    bagua_amd.allreduce_inplace(probe)
    expect = probe

    bt = ensure_bagua_tensor(t, "x")
    bucket = BaguaBucket([bt], "b0", flatten=True)
    bucket.ops = [CentralizedSyncOp(hierarchical=True, average=True)]
    execute_ops(bucket, group)
    got = bucket.comm_tensor()
    bagua_amd.deinit_process_group()
    return got, expect


def test_hierarchical_equals_flat():
    out = run_multi_process(2, _worker_hier_matches_flat)
    for got, expect in out:
        assert torch.allclose(got, expect, atol=1e-6)


def test_hierarchical_qadam_consensus():
    out = run_multi_process(2, _worker_hier_qadam)
    assert torch.equal(out[0], out[1]), "hierarchical qadam diverged"


def _worker_hier_qadam(rank, nprocs):
    os.environ["NODE_RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = "0"

    import bagua_amd
    from bagua_amd.parallel.algorithms.q_adam import (
        QAdamAlgorithm,
        QAdamOptimizer,
    )
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = QAdamOptimizer(model.parameters(), lr=1e-3, warmup_steps=4)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=QAdamAlgorithm(optimizer, hierarchical=True))
    for step in range(8):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat
