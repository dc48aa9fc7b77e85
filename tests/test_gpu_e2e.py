"""GPU end-to-end: one-rank training through the native path."""

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
    os.environ.setdefault("MASTER_PORT", "29591")


@requires_gpu
@pytest.mark.parametrize("algo", ["gradient_allreduce", "bytegrad"])
def test_train_steps(algo):
    _setup_env()
    import bagua_amd
    from bagua_amd.models import MnistNet
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry

    torch.cuda.set_device(0)
    bagua_amd.init_process_group()
    torch.manual_seed(3)
    model = MnistNet().cuda()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.01)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GlobalAlgorithmRegistry.get(algo)())
    losses = []
    for step in range(5):
        data = torch.randn(16, 1, 28, 28, device="cuda")
        target = torch.randint(0, 10, (16,), device="cuda")
        optimizer.zero_grad()
        loss = F.nll_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
