"""Algorithm zoo semantics on CPU/gloo, 2 processes.

Tiers (reference test strategy):
* consensus — after N steps all ranks' flattened weights agree;
* golden-model — decentralized rules re-simulated in pure Python and
  compared against the framework trajectory
  (reference: tests/torch_api/test_decentralized.py,
  test_low_precision_decentralized.py).
"""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(8, 16)
        self.fc2 = nn.Linear(16, 4)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


def _make_data(rank, step):
    torch.manual_seed(7000 + rank * 131 + step)
    return torch.randn(4, 8), torch.randn(4, 4)


def _flat_params(model):
    return torch.cat([p.detach().reshape(-1) for p in model.parameters()])


def _worker_algorithm(rank, nprocs, algo_name, steps, kwargs):
    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()

    if algo_name == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        optimizer = QAdamOptimizer(model.parameters(), lr=1e-3,
                                   warmup_steps=kwargs.pop("warmup_steps", 6))
        algorithm = GlobalAlgorithmRegistry.get(algo_name)(optimizer,
                                                           **kwargs)
    else:
        optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
        algorithm = GlobalAlgorithmRegistry.get(algo_name)(**kwargs)

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)

    losses = []
    for step in range(steps):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
        losses.append(loss.item())

    if algo_name == "async":
        ddp.inner.bagua_algorithm.abort(ddp)

    flat = _flat_params(model)
    bagua_amd.deinit_process_group()
    return flat, losses


@pytest.mark.parametrize("algo,kwargs", [
    ("bytegrad", {}),
    ("decentralized", {"peer_selection_mode": "all"}),
    ("decentralized", {"peer_selection_mode": "shift_one"}),
    ("low_precision_decentralized", {}),
    ("qadam", {}),
])
def test_consensus(algo, kwargs):
    """All ranks end identical (decentralized modes reach consensus after
    the final communication because peer averaging with 2 ranks is exact)."""
    nprocs = 2
    out = run_multi_process(nprocs, _worker_algorithm,
                            args=(algo, 12, dict(kwargs)))
    flat0, losses0 = out[0]
    flat1, losses1 = out[1]
    assert torch.isfinite(flat0).all() and torch.isfinite(flat1).all()
    if algo in ("bytegrad", "qadam"):
        # centralized algorithms keep ranks bitwise identical every step
        assert torch.equal(flat0, flat1), "ranks diverged"
    else:
        # decentralized: last optimizer step applies local grads after the
        # last averaging, so ranks differ by one local step; they must
        # still be close and the averaged trajectories consistent
        assert torch.allclose(flat0, flat1, atol=0.5)


def test_async_lifecycle():
    out = run_multi_process(2, _worker_algorithm,
                            args=("async", 6,
                                  {"sync_interval_ms": 10,
                                   "warmup_steps": 2}))
    for flat, losses in out:
        assert torch.isfinite(flat).all()
        assert len(losses) == 6


# ---------------------------------------------------------------------------
# golden-model: decentralized "all" exactly matches a pure-python simulation
# ---------------------------------------------------------------------------


def _worker_decentralized_golden(rank, nprocs, steps):
    import bagua_amd
    from bagua_amd.parallel.algorithms.decentralized import (
        DecentralizedAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=DecentralizedAlgorithm(peer_selection_mode="all"))
    for step in range(steps):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = _flat_params(model)
    bagua_amd.deinit_process_group()
    return flat


def _simulate_decentralized(nprocs, steps, lr=0.05):
    """Pure-python re-implementation of decentralized-all:
    per step: average weights across ranks (communicated at forward-pre,
    installed post-backward), grads computed on pre-average weights,
    optimizer applies local grads to the averaged weights."""
    torch.manual_seed(13)  # rank0 init broadcast
    models = [Net() for _ in range(nprocs)]
    with torch.no_grad():
        for m in models[1:]:
            for p, p0 in zip(m.parameters(), models[0].parameters()):
                p.copy_(p0)
    for step in range(steps):
        # grads on current (pre-average) weights
        grads = []
        for rank, m in enumerate(models):
            data, target = _make_data(rank, step)
            m.zero_grad()
            loss = F.mse_loss(m(data), target)
            loss.backward()
            grads.append([p.grad.clone() for p in m.parameters()])
        # average weights
        with torch.no_grad():
            avgs = [torch.stack([list(m.parameters())[i].detach()
                                 for m in models]).mean(0)
                    for i in range(len(list(models[0].parameters())))]
            for m, g in zip(models, grads):
                for p, a, gr in zip(m.parameters(), avgs, g):
                    p.copy_(a - lr * gr)
    return [_flat_params(m) for m in models]


def test_decentralized_golden_model():
    nprocs, steps = 2, 4
    out = run_multi_process(nprocs, _worker_decentralized_golden,
                            args=(steps,))
    golden = _simulate_decentralized(nprocs, steps)
    for rank in range(nprocs):
        assert torch.allclose(out[rank], golden[rank], atol=1e-5), (
            "rank %d deviates from golden decentralized trajectory" % rank)


def _worker_shift_one_golden(rank, nprocs, steps):
    import bagua_amd
    from bagua_amd.parallel.algorithms.decentralized import (
        DecentralizedAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=DecentralizedAlgorithm(peer_selection_mode="shift_one"))
    for step in range(steps):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = _flat_params(model)
    bagua_amd.deinit_process_group()
    return flat


def _simulate_shift_one(nprocs, steps, lr=0.05):
    """Golden: per step, each rank pairs via the half-ring formula and
    averages weights with its peer, then applies local grads."""
    torch.manual_seed(13)
    models = [Net() for _ in range(nprocs)]
    with torch.no_grad():
        for m in models[1:]:
            for p, p0 in zip(m.parameters(), models[0].parameters()):
                p.copy_(p0)
    for step in range(steps):
        grads = []
        for rank, m in enumerate(models):
            data, target = _make_data(rank, step)
            m.zero_grad()
            F.mse_loss(m(data), target).backward()
            grads.append([p.grad.clone() for p in m.parameters()])
        # pairing (reference formula,
        # decentralized_full_precision_synchronous.rs:79-85)
        n = nprocs
        peers = {}
        for rank in range(n):
            if rank < n // 2:
                peers[rank] = ((step + rank) % ((n + 1) // 2)) + n // 2
            else:
                peers[rank] = (rank - n // 2 - step) % (n // 2)
        with torch.no_grad():
            olds = [[p.detach().clone() for p in m.parameters()]
                    for m in models]
            for rank, m in enumerate(models):
                peer = peers[rank]
                for p, mine, theirs, g in zip(m.parameters(), olds[rank],
                                              olds[peer], grads[rank]):
                    p.copy_((mine + theirs) / 2 - lr * g)
    return [_flat_params(m) for m in models]


def test_shift_one_golden_4ranks():
    nprocs, steps = 4, 3
    out = run_multi_process(nprocs, _worker_shift_one_golden, args=(steps,))
    golden = _simulate_shift_one(nprocs, steps)
    for rank in range(nprocs):
        assert torch.allclose(out[rank], golden[rank], atol=1e-5), (
            "rank %d deviates from golden shift_one trajectory" % rank)
