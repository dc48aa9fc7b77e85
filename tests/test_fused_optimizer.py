"""Fused optimizer equivalence vs plain optimizers
(reference pattern: tests/contrib/test_fused_optimizer.py)."""

import copy

import pytest
import torch
import torch.nn as nn

from bagua_amd.contrib import fuse_optimizer, is_fused_optimizer


def _model():
    torch.manual_seed(5)
    return nn.Sequential(
        nn.Linear(13, 27), nn.ReLU(), nn.Linear(27, 11), nn.ReLU(),
        nn.Linear(11, 3))


def _run(model, optimizer, fused, steps=6):
    torch.manual_seed(9)
    for _ in range(steps):
        x = torch.randn(8, 13)
        y = torch.randn(8, 3)
        loss = ((model(x) - y) ** 2).mean()
        optimizer.zero_grad()
        loss.backward()
        if fused:
            optimizer.fuse_step()
        else:
            optimizer.step()
    return torch.cat([p.detach().reshape(-1) for p in model.parameters()])


@pytest.mark.parametrize("opt_cls,kwargs", [
    (torch.optim.SGD, {"lr": 0.05, "momentum": 0.9, "weight_decay": 1e-4}),
    (torch.optim.Adam, {"lr": 1e-2}),
    (torch.optim.AdamW, {"lr": 1e-2}),
    (torch.optim.Adadelta, {"lr": 0.5}),
])
def test_fused_matches_plain(opt_cls, kwargs):
    m1 = _model()
    m2 = copy.deepcopy(m1)

    o1 = opt_cls(m1.parameters(), **kwargs)
    ref = _run(m1, o1, fused=False)

    o2 = fuse_optimizer(opt_cls(m2.parameters(), **kwargs))
    assert is_fused_optimizer(o2)
    out = _run(m2, o2, fused=True)

    assert torch.allclose(ref, out, atol=1e-6), (
        "fused %s deviates from plain" % opt_cls.__name__)
    assert o2._bagua_fused_count > 0, "fusion never engaged"


def test_double_fuse_raises():
    m = _model()
    o = fuse_optimizer(torch.optim.SGD(m.parameters(), lr=0.1))
    with pytest.raises(RuntimeError):
        fuse_optimizer(o)


def test_fuse_then_plain_step_still_works():
    m = _model()
    o = fuse_optimizer(torch.optim.SGD(m.parameters(), lr=0.1))
    x = torch.randn(4, 13)
    loss = m(x).sum()
    o.zero_grad()
    loss.backward()
    o.step()  # plain step must keep working
    assert all(torch.isfinite(p).all() for p in m.parameters())


def _worker_fused_with_ddp(rank, nprocs):
    """Reference-documented combo: fuse_optimizer + with_bagua
    do_flatten=False (fuse/optimizer.py docstring)."""
    import bagua_amd
    from bagua_amd.contrib import fuse_optimizer
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    torch.manual_seed(13 + rank)
    bagua_amd.init_process_group()
    model = _model()
    optimizer = fuse_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9))
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm(),
        gradient_as_bucket_view=False)
    torch.manual_seed(80 + rank)
    for _ in range(5):
        x, y = torch.randn(6, 13), torch.randn(6, 3)
        optimizer.zero_grad()
        ((ddp(x) - y) ** 2).mean().backward()
        optimizer.fuse_step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_fused_optimizer_with_ddp_no_flatten():
    from tests.internal.multi_process import run_multi_process

    out = run_multi_process(2, _worker_fused_with_ddp)
    assert torch.equal(out[0], out[1]), "ranks diverged"
    assert torch.isfinite(out[0]).all()
