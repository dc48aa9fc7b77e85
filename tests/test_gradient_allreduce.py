"""End-to-end GradientAllReduce: consensus + torch-DDP parity, 2-proc gloo
(reference pattern: tests/torch_api/test_gradient_allreduce.py:75-131)."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(8, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, 4)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        return self.fc3(x)


def _train(rank, model, optimizer, steps=10, seed_offset=0):
    torch.manual_seed(1000 + rank + seed_offset)
    losses = []
    for _ in range(steps):
        data = torch.randn(4, 8)
        target = torch.randn(4, 4)
        optimizer.zero_grad()
        loss = F.mse_loss(model(data), target)
        loss.backward()
        optimizer.step()
        losses.append(loss.item())
    return losses


def _worker_bagua(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)  # different init; broadcast must fix it
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    _train(rank, ddp, optimizer)
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def _worker_torch_ddp(rank, nprocs):
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=nprocs)
    torch.manual_seed(13)  # torch DDP broadcasts rank0 weights anyway
    model = Net()
    ddp = torch.nn.parallel.DistributedDataParallel(model)
    optimizer = torch.optim.SGD(ddp.parameters(), lr=0.05)
    _train(rank, ddp, optimizer)
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    dist.destroy_process_group()
    return flat


def test_consensus_and_ddp_parity():
    """After N steps all bagua ranks agree bitwise AND match torch DDP."""
    nprocs = 2
    bagua_out = run_multi_process(nprocs, _worker_bagua)
    assert torch.equal(bagua_out[0], bagua_out[1]), "ranks diverged"
    torch_out = run_multi_process(nprocs, _worker_torch_ddp)
    assert torch.equal(torch_out[0], torch_out[1])
    assert torch.allclose(bagua_out[0], torch_out[0], atol=1e-6), (
        "bagua result differs from torch DDP")


def _worker_no_sync(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    torch.manual_seed(50 + rank)
    data = torch.randn(4, 8)
    target = torch.randn(4, 4)
    with ddp.no_sync():
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
    grad = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return grad


def test_no_sync_keeps_local_grads():
    out = run_multi_process(2, _worker_no_sync)
    assert not torch.equal(out[0], out[1]), (
        "grads were synced inside no_sync()")


def _worker_no_flatten(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm(),
        gradient_as_bucket_view=False)
    _train(rank, ddp, optimizer)
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_no_flatten_matches_flatten():
    """do_flatten=False (gather-copy comm path, used with the fused
    optimizer) must produce the same result as the fused-view path."""
    out = run_multi_process(2, _worker_no_flatten)
    assert torch.equal(out[0], out[1])
    ref = run_multi_process(2, _worker_bagua)
    assert torch.allclose(out[0], ref[0], atol=1e-6)


def _worker_grad_accumulation(rank, nprocs):
    """no_sync accumulation then synced step must match torch DDP."""
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    for step in range(3):
        optimizer.zero_grad()
        with ddp.no_sync():  # micro-batch 1: local accumulation
            torch.manual_seed(600 + rank * 17 + step * 2)
            loss = F.mse_loss(ddp(torch.randn(4, 8)), torch.randn(4, 4))
            loss.backward()
        # micro-batch 2: synced
        torch.manual_seed(601 + rank * 17 + step * 2)
        loss = F.mse_loss(ddp(torch.randn(4, 8)), torch.randn(4, 4))
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def _worker_grad_accumulation_torch(rank, nprocs):
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=nprocs)
    torch.manual_seed(13)
    model = Net()
    ddp = torch.nn.parallel.DistributedDataParallel(model)
    optimizer = torch.optim.SGD(ddp.parameters(), lr=0.05)
    for step in range(3):
        optimizer.zero_grad()
        with ddp.no_sync():
            torch.manual_seed(600 + rank * 17 + step * 2)
            loss = F.mse_loss(ddp(torch.randn(4, 8)), torch.randn(4, 4))
            loss.backward()
        torch.manual_seed(601 + rank * 17 + step * 2)
        loss = F.mse_loss(ddp(torch.randn(4, 8)), torch.randn(4, 4))
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    dist.destroy_process_group()
    return flat


def test_gradient_accumulation_matches_torch_ddp():
    ours = run_multi_process(2, _worker_grad_accumulation)
    ref = run_multi_process(2, _worker_grad_accumulation_torch)
    assert torch.equal(ours[0], ours[1])
    assert torch.allclose(ours[0], ref[0], atol=1e-6), (
        "grad accumulation deviates from torch DDP (max %g)"
        % (ours[0] - ref[0]).abs().max())
