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


def test_bf16_master_matches_fp32_cpu():
    """Pure-bf16 FusedSGD (fp32 master) must track plain fp32 SGD within
    bf16 rounding of the weights."""
    torch.manual_seed(44)
    m32 = _model()
    m16 = copy.deepcopy(m32).to(torch.bfloat16)
    o32 = torch.optim.SGD(m32.parameters(), lr=0.05, momentum=0.9)
    o16 = FusedSGD(m16.parameters(), lr=0.05, momentum=0.9)
    torch.manual_seed(55)
    for _ in range(5):
        x = torch.randn(6, 10)
        y = torch.randn(6, 5)
        o32.zero_grad()
        ((m32(x) - y) ** 2).mean().backward()
        o32.step()
        o16.zero_grad()
        ((m16(x.bfloat16()) - y.bfloat16()) ** 2).mean().backward()
        o16.step()
    w32 = torch.cat([p.detach().reshape(-1) for p in m32.parameters()])
    w16 = torch.cat([p.detach().float().reshape(-1)
                     for p in m16.parameters()])
    assert torch.allclose(w32, w16, atol=0.05), (w32 - w16).abs().max()


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
def test_bf16_master_kernel_matches_cpu_math():
    torch.manual_seed(61)
    m_gpu = _model().to(torch.bfloat16).cuda()
    m_cpu = copy.deepcopy(m_gpu).cpu()
    og = FusedSGD(m_gpu.parameters(), lr=0.05, momentum=0.9,
                  weight_decay=1e-4)
    oc = FusedSGD(m_cpu.parameters(), lr=0.05, momentum=0.9,
                  weight_decay=1e-4)
    torch.manual_seed(62)
    for _ in range(4):
        x = torch.randn(6, 10).bfloat16()
        y = torch.randn(6, 5).bfloat16()
        og.zero_grad()
        ((m_gpu(x.cuda()) - y.cuda()) ** 2).mean().backward()
        og.step()
        oc.zero_grad()
        ((m_cpu(x) - y) ** 2).mean().backward()
        oc.step()
    torch.cuda.synchronize()
    wg = torch.cat([p.detach().float().cpu().reshape(-1)
                    for p in m_gpu.parameters()])
    wc = torch.cat([p.detach().float().reshape(-1)
                    for p in m_cpu.parameters()])
    # bf16 fwd/bwd on different devices can differ by rounding; the
    # master-weight update itself is exact
    assert torch.allclose(wg, wc, atol=2e-2), (wg - wc).abs().max()


def test_fused_sgd_state_dict_roundtrip():
    m = _model()
    o = FusedSGD(m.parameters(), lr=0.05, momentum=0.9)
    _train(m, o, steps=3)
    sd = o.state_dict()

    m2 = copy.deepcopy(m)
    o2 = FusedSGD(m2.parameters(), lr=0.05, momentum=0.9)
    o2.load_state_dict(sd)
    a = _train(m, o, steps=2)
    b = _train(m2, o2, steps=2)
    assert torch.equal(a.float(), b.float()), "state_dict roundtrip broke"


def test_adam_bf16_master_matches_fp32_cpu():
    torch.manual_seed(71)
    m32 = _model()
    m16 = copy.deepcopy(m32).to(torch.bfloat16)
    o32 = torch.optim.Adam(m32.parameters(), lr=1e-2)
    o16 = FusedAdam(m16.parameters(), lr=1e-2)
    torch.manual_seed(72)
    for _ in range(5):
        x = torch.randn(6, 10)
        y = torch.randn(6, 5)
        o32.zero_grad()
        ((m32(x) - y) ** 2).mean().backward()
        o32.step()
        o16.zero_grad()
        ((m16(x.bfloat16()) - y.bfloat16()) ** 2).mean().backward()
        o16.step()
    w32 = torch.cat([p.detach().reshape(-1) for p in m32.parameters()])
    w16 = torch.cat([p.detach().float().reshape(-1)
                     for p in m16.parameters()])
    assert torch.allclose(w32, w16, atol=0.06), (w32 - w16).abs().max()
