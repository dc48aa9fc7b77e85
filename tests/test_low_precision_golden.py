"""Low-precision decentralized golden-model test: pure-python simulation
of the exact update rule, quantizer included, vs the framework
(reference: tests/torch_api/test_low_precision_decentralized.py)."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from bagua_amd.ops import quant
from tests.internal.multi_process import run_multi_process


class TinyNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(5, 7)
        self.fc2 = nn.Linear(7, 3)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


def _data(rank, step):
    torch.manual_seed(4000 + rank * 77 + step)
    return torch.randn(4, 5), torch.randn(4, 3)


def _worker(rank, nprocs, steps):
    import bagua_amd
    from bagua_amd.parallel.algorithms.decentralized import (
        LowPrecisionDecentralizedAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = TinyNet()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=LowPrecisionDecentralizedAlgorithm())
    for step in range(steps):
        data, target = _data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    # params in the engine's registration order (reverse param order)
    names = [n for n, _ in reversed(ddp.inner.bagua_build_params())]
    params = dict(model.named_parameters())
    flat = torch.cat([params[n].detach().reshape(-1) for n in names])
    bagua_amd.deinit_process_group()
    return flat


def _simulate(nprocs, steps, lr=0.05):
    """Exact re-implementation over flat vectors:
        x <- optimizer step
        t  = x + L/3 + R/3 - 5W/3
        c  = Q(t) (one chunk, 32-elem padded)
        L += deQ(c_left); R += deQ(c_right); W += deQ(c_own); x = W
    """
    torch.manual_seed(13)  # rank0 init broadcast
    models = [TinyNet() for _ in range(nprocs)]
    with torch.no_grad():
        for m in models[1:]:
            for p, p0 in zip(m.parameters(), models[0].parameters()):
                p.copy_(p0)

    # flat order = reverse of named_parameters
    def get_flat(m):
        ps = list(m.named_parameters())
        return torch.cat([p.detach().reshape(-1)
                          for _, p in reversed(ps)])

    def set_flat(m, flat):
        ps = list(m.named_parameters())
        offset = 0
        with torch.no_grad():
            for _, p in reversed(ps):
                p.copy_(flat[offset:offset + p.numel()].view_as(p))
                offset += p.numel()

    numel = get_flat(models[0]).numel()
    padded = (numel + 31) // 32 * 32

    def pad(v):
        out = torch.zeros(padded)
        out[:numel] = v
        return out

    W = [pad(get_flat(m)) for m in models]
    L = [w.clone() for w in W]
    R = [w.clone() for w in W]

    for step in range(steps):
        xs = []
        for rank, m in enumerate(models):
            data, target = _data(rank, step)
            m.zero_grad()
            F.mse_loss(m(data), target).backward()
            with torch.no_grad():
                for p in m.parameters():
                    p.sub_(p.grad, alpha=lr)
            xs.append(pad(get_flat(m)))

        comp = []
        for rank in range(nprocs):
            t = xs[rank] + L[rank] / 3 + R[rank] / 3 - 5 * W[rank] / 3
            comp.append(quant.compress_chunked(t, 1))
        for rank in range(nprocs):
            left = (rank - 1) % nprocs
            right = (rank + 1) % nprocs
            L[rank] += quant.decompress_chunked(comp[left], 1, padded)
            R[rank] += quant.decompress_chunked(comp[right], 1, padded)
            W[rank] += quant.decompress_chunked(comp[rank], 1, padded)
            set_flat(models[rank], W[rank][:numel])
    return [get_flat(m) for m in models]


def test_low_precision_matches_golden():
    nprocs, steps = 2, 3
    out = run_multi_process(nprocs, _worker, args=(steps,))
    golden = _simulate(nprocs, steps)
    for rank in range(nprocs):
        assert torch.allclose(out[rank], golden[rank], atol=1e-5), (
            "rank %d deviates from the golden low-precision trajectory"
            % rank)
