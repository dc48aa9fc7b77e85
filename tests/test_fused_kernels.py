"""FusedSGD / FusedAdam vs torch reference optimizers (CPU math here,
GPU kernel path under -m gpu)."""

import copy

import pytest
import torch
import torch.nn as nn

from bagua_amd.contrib import FusedAdam, FusedAdamW, FusedSGD


def _model():
    torch.manual_seed(21)
    return nn.Sequential(nn.Linear(10, 20), nn.ReLU(), nn.Linear(20, 5))


def _train(model, opt, steps=5):
    torch.manual_seed(33)
    device = next(model.parameters()).device
    for _ in range(steps):
        x = torch.randn(6, 10, device=device)
        y = torch.randn(6, 5, device=device)
        opt.zero_grad()
        ((model(x) - y) ** 2).mean().backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in model.parameters()])


@pytest.mark.parametrize("fused_cls,ref_cls,kwargs", [
    (FusedSGD, torch.optim.SGD,
     {"lr": 0.05, "momentum": 0.9, "weight_decay": 1e-4}),
    (FusedSGD, torch.optim.SGD,
     {"lr": 0.05, "momentum": 0.9, "nesterov": True, "weight_decay": 1e-4}),
    (FusedSGD, torch.optim.SGD, {"lr": 0.05}),
    (FusedAdam, torch.optim.Adam, {"lr": 1e-2, "weight_decay": 1e-3}),
    (FusedAdamW, torch.optim.AdamW, {"lr": 1e-2, "weight_decay": 1e-2}),
])
def test_matches_torch_cpu(fused_cls, ref_cls, kwargs):
    m1, m2 = _model(), None
    m2 = copy.deepcopy(m1)
    ref = _train(m1, ref_cls(m1.parameters(), **kwargs))
    out = _train(m2, fused_cls(m2.parameters(), **kwargs))
    assert torch.allclose(ref, out, atol=1e-6), "fused optimizer deviates"


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
@pytest.mark.parametrize("fused_cls,ref_cls,kwargs", [
    (FusedSGD, torch.optim.SGD,
     {"lr": 0.05, "momentum": 0.9, "weight_decay": 1e-4}),
    (FusedAdam, torch.optim.Adam, {"lr": 1e-2, "weight_decay": 1e-3}),
    (FusedAdamW, torch.optim.AdamW, {"lr": 1e-2, "weight_decay": 1e-2}),
])
def test_matches_torch_gpu(fused_cls, ref_cls, kwargs):
    m1 = _model().cuda()
    m2 = copy.deepcopy(m1)
    ref = _train(m1, ref_cls(m1.parameters(), **kwargs))
    out = _train(m2, fused_cls(m2.parameters(), **kwargs))
    torch.cuda.synchronize()
    assert torch.allclose(ref, out, atol=1e-5), "fused kernel deviates"
