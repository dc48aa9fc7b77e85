"""GPU golden-model algorithm tests at world 1.

The quantized wire math (HIP minmax/quantize/dequantize kernels), the
chunk reduce and the elementwise kernels are device-executed even at one
rank — these tests run the FULL algorithm on the MI355X and compare the
trajectory against the same pure-python simulators the CPU/gloo tier
uses (tests/test_low_precision_golden.py, tests/test_qadam_golden.py),
so a kernel regression shows up as a trajectory deviation rather than a
loss that merely stays finite.

Tolerances: the simulators run on CPU; fp32 linear layers this small
drift O(1e-7) per step across devices, amplified at most ~1 quant level
by the compressor — 5e-4 catches any real kernel defect (which shows as
O(1e-1)+ deviation or NaN).
"""

import os

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


def _setup_env():
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29593")


def _flat_reversed(model):
    ps = list(model.named_parameters())
    return torch.cat([p.detach().reshape(-1).cpu()
                      for _, p in reversed(ps)])


@requires_gpu
def test_low_precision_decentralized_matches_golden_on_gpu():
    from tests.test_low_precision_golden import TinyNet, _data, _simulate

    _setup_env()
    import bagua_amd
    from bagua_amd.ops import native
    from bagua_amd.parallel.algorithms.decentralized import (
        LowPrecisionDecentralizedAlgorithm,
    )

    assert native.available(), "HIP extension must be loaded on GPU"
    torch.cuda.set_device(0)
    bagua_amd.init_process_group()

    steps = 3
    torch.manual_seed(13)
    model = TinyNet().cuda()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=LowPrecisionDecentralizedAlgorithm())
    for step in range(steps):
        data, target = _data(0, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data.cuda()), target.cuda())
        loss.backward()
        optimizer.step()
    torch.cuda.synchronize()

    golden = _simulate(1, steps)[0]
    got = _flat_reversed(model)
    assert torch.allclose(got, golden, atol=5e-4), (
        "GPU low-precision trajectory deviates from the python-oracle "
        "golden (max diff %g)" % (got - golden).abs().max().item())


@requires_gpu
def test_qadam_matches_golden_on_gpu():
    from tests import test_qadam_golden as G

    _setup_env()
    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry
    from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

    torch.cuda.set_device(0)
    bagua_amd.init_process_group()

    torch.manual_seed(13)
    model = G.TinyNet().cuda()
    optimizer = QAdamOptimizer(model.parameters(), lr=G.LR,
                               warmup_steps=G.WARMUP)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GlobalAlgorithmRegistry.get("qadam")(optimizer))
    for step in range(G.STEPS):
        data, target = G._data(0, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data.cuda()), target.cuda())
        loss.backward()
        optimizer.step()
    torch.cuda.synchronize()

    golden = _simulate_qadam_rank0()
    got = torch.cat([p.detach().reshape(-1).cpu()
                     for p in model.parameters()])
    assert torch.allclose(got, golden, atol=5e-4), (
        "GPU QAdam trajectory deviates from golden (max diff %g)"
        % (got - golden).abs().max().item())


def _simulate_qadam_rank0():
    from tests import test_qadam_golden as G

    return G._simulate(1)[0]


@requires_gpu
def test_decentralized_all_matches_local_sgd_on_gpu():
    """World 1: peer averaging is the identity, so the trajectory must be
    EXACTLY local SGD — proves the peer-weight copy/copy-back kernels do
    not corrupt the weights."""
    from tests.test_low_precision_golden import TinyNet, _data

    _setup_env()
    import bagua_amd
    from bagua_amd.parallel.algorithms.decentralized import (
        DecentralizedAlgorithm,
    )

    torch.cuda.set_device(0)
    bagua_amd.init_process_group()

    steps = 4
    torch.manual_seed(13)
    model = TinyNet().cuda()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=DecentralizedAlgorithm(peer_selection_mode="all"))
    for step in range(steps):
        data, target = _data(0, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data.cuda()), target.cuda())
        loss.backward()
        optimizer.step()
    torch.cuda.synchronize()

    # golden: identical math, no wrapper, same device (bitwise target)
    torch.manual_seed(13)
    ref = TinyNet().cuda()
    opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    for step in range(steps):
        data, target = _data(0, step)
        opt.zero_grad()
        F.mse_loss(ref(data.cuda()), target.cuda()).backward()
        opt.step()
    torch.cuda.synchronize()

    a = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    b = torch.cat([p.detach().reshape(-1) for p in ref.parameters()])
    assert torch.allclose(a, b, atol=1e-6), (
        "decentralized(all) at world 1 deviates from local SGD "
        "(max diff %g)" % (a - b).abs().max().item())
