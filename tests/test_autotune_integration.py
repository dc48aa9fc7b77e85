"""End-to-end autotune: engine registers tensors, reports metrics,
re-buckets from service proposals, and training stays correct."""

import os

import torch
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


def _worker(rank, nprocs):
    os.environ["BAGUA_AUTOTUNE"] = "1"
    os.environ["BAGUA_AUTOTUNE_INTERVAL"] = "4"
    os.environ["BAGUA_AUTOTUNE_WARMUP_TIME_S"] = "0"
    os.environ["BAGUA_AUTOTUNE_SAMPLING_CONFIDENCE_TIME_S"] = "0"
    os.environ["BAGUA_AUTOTUNE_MAX_SAMPLES"] = "2"

    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    for step in range(14):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    completed = ddp.inner._autotune_completed
    bagua_amd.deinit_process_group()
    return flat, completed


def test_autotune_end_to_end():
    out = run_multi_process(2, _worker, timeout=300)
    assert torch.equal(out[0][0], out[1][0]), "ranks diverged under autotune"
    assert torch.isfinite(out[0][0]).all()
