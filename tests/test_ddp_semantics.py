"""torch-DDP behavioral equivalence beyond gradient parity
(reference: tests/torch_api/data_parallel/test_c10d_common.py — the
vendored 1,066-LoC c10d comparison suite; this file covers the behavior
classes relevant to the bagua wrapper: buffer broadcasting, comm-hook
rejection semantics, sparse-parameter rejection, no_sync).
"""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


class BNNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc = nn.Linear(8, 6)
        # momentum=0: forward leaves running stats untouched, so the
        # value observed after forward is exactly what the pre-forward
        # broadcast (or its absence) left there
        self.bn = nn.BatchNorm1d(6, momentum=0.0)

    def forward(self, x):
        return self.bn(self.fc(x))


def _worker_buffers(rank, nprocs, broadcast_buffers):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(1)
    model = BNNet()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.01)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm(),
        broadcast_buffers=broadcast_buffers)

    # drift the buffers apart AFTER construction (init broadcast already
    # synced them)
    with torch.no_grad():
        model.bn.running_mean.fill_(float(rank + 1))

    torch.manual_seed(50 + rank)
    data = torch.randn(16, 8)
    optimizer.zero_grad()
    out = ddp(data)
    # capture the buffer AS SEEN going into BN this forward: with
    # broadcast_buffers the pre-forward sync overwrote the drift
    seen_mean = model.bn.running_mean.detach().clone()
    F.mse_loss(out, torch.zeros_like(out)).backward()
    optimizer.step()
    bagua_amd.deinit_process_group()
    return seen_mean


def test_broadcast_buffers_resyncs_each_forward():
    out = run_multi_process(2, _worker_buffers, args=(True,))
    assert torch.equal(out[0], out[1]), (
        "broadcast_buffers=True must re-sync buffers before forward")


def test_no_broadcast_buffers_keeps_local():
    out = run_multi_process(2, _worker_buffers, args=(False,))
    assert not torch.equal(out[0], out[1]), (
        "broadcast_buffers=False must leave buffers rank-local")


def _worker_comm_hook(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    model = nn.Linear(4, 4)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[], algorithm=GradientAllReduceAlgorithm())
    try:
        ddp.register_comm_hook(None, lambda state, bucket: None)
        raised = False
    except NotImplementedError as e:
        raised = "Algorithm" in str(e)
    try:
        ddp._register_builtin_comm_hook(object())
        raised2 = False
    except NotImplementedError:
        raised2 = True
    bagua_amd.deinit_process_group()
    return raised and raised2


def test_comm_hook_rejected_with_guidance():
    out = run_multi_process(2, _worker_comm_hook)
    assert all(out)


def _worker_sparse(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()

    class SparseNet(nn.Module):
        def __init__(self):
            super().__init__()
            self.w = nn.Parameter(torch.randn(4, 4).to_sparse())

    model = SparseNet()
    msg = ""
    try:
        bagua_amd.DistributedDataParallel(
            model, optimizers=[], algorithm=GradientAllReduceAlgorithm())
    except ValueError as e:
        msg = str(e)
    bagua_amd.deinit_process_group()
    return msg


def test_sparse_parameters_rejected_with_message():
    out = run_multi_process(2, _worker_sparse)
    for msg in out:
        assert "sparse" in msg.lower(), (
            "sparse rejection must be explicit, got: %r" % msg)


def _worker_no_sync_alternating(rank, nprocs):
    """no_sync accumulation then sync step — torch DDP's documented
    pattern; grads under no_sync stay local, next sync step averages the
    ACCUMULATED gradient."""
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(1)
    model = nn.Linear(4, 2)
    optimizer = torch.optim.SGD(model.parameters(), lr=0.1)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    torch.manual_seed(10 + rank)
    d1 = torch.randn(8, 4)
    d2 = torch.randn(8, 4)
    optimizer.zero_grad()
    with ddp.no_sync():
        ddp(d1).sum().backward()
    local_after_nosync = model.weight.grad.detach().clone()
    ddp(d2).sum().backward()
    synced = model.weight.grad.detach().clone()
    bagua_amd.deinit_process_group()
    return local_after_nosync, synced


def test_no_sync_then_sync_accumulates():
    out = run_multi_process(2, _worker_no_sync_alternating)
    # under no_sync grads differ per rank (local data)
    assert not torch.equal(out[0][0], out[1][0])
    # after the sync step, the accumulated grads agree across ranks
    assert torch.allclose(out[0][1], out[1][1], atol=1e-6)
