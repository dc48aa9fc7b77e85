"""QAdam golden-model test: exact pure-python simulation of warmup
(grad allreduce) and compression phase (momentum update on the comm path
+ chunked MinMaxUInt8 sync), compared against the framework
(reference: tests/torch_api/test_qadam.py)."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from bagua_amd.ops import quant
from tests.internal.multi_process import run_multi_process

WARMUP = 4
STEPS = 7
LR = 1e-3


class TinyNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(5, 6)
        self.fc2 = nn.Linear(6, 2)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


def _data(rank, step):
    torch.manual_seed(3000 + rank * 53 + step)
    return torch.randn(4, 5), torch.randn(4, 2)


def _worker(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.q_adam import (
        QAdamAlgorithm,
        QAdamOptimizer,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = TinyNet()
    optimizer = QAdamOptimizer(model.parameters(), lr=LR,
                               warmup_steps=WARMUP)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=QAdamAlgorithm(optimizer))
    for step in range(STEPS):
        data, target = _data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1)
                      for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def _simulate(nprocs):
    from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

    torch.manual_seed(13)  # rank0 broadcast
    models = [TinyNet() for _ in range(nprocs)]
    with torch.no_grad():
        for m in models[1:]:
            for p, p0 in zip(m.parameters(), models[0].parameters()):
                p.copy_(p0)
    opts = [QAdamOptimizer(m.parameters(), lr=LR, warmup_steps=WARMUP)
            for m in models]

    # engine init: optimizer state materialization with zero grads
    # (engine._bagua_broadcast_optimizer_state) — the engine rewinds the
    # step counters afterwards so the fake step is invisible
    for m, o in zip(models, opts):
        for p in m.parameters():
            p.grad = torch.zeros_like(p)
        o.step()
        o.zero_grad()
        for p in m.parameters():
            o.state[p]["step"] = 0

    # registration order: reverse of build-params order, matching the
    # QAdam impl (sorted by _q_adam_idx)
    def reg_params(m):
        return [p for _, p in reversed(list(m.named_parameters()))]

    def step_id(o, m):
        return o.state[next(iter(m.parameters()))]["step"]

    for step in range(STEPS):
        warmup_phase = step_id(opts[0], models[0]) + 1 < WARMUP + 1 \
            and step_id(opts[0], models[0]) < WARMUP
        # phase at comm time = phase chosen at last (re)init:
        # engine re-inits when optimizer_step_id == WARMUP (forward-pre)
        compress_phase = step_id(opts[0], models[0]) >= WARMUP

        for rank, (m, o) in enumerate(zip(models, opts)):
            data, target = _data(rank, step)
            o.zero_grad()
            F.mse_loss(m(data), target).backward()

        if not compress_phase:
            # warmup: allreduce-AVG grads
            with torch.no_grad():
                for ps in zip(*[m.parameters() for m in models]):
                    g = torch.stack([p.grad for p in ps]).mean(0)
                    for p in ps:
                        p.grad.copy_(g)
        else:
            # momentum update + chunked compressed sync of exp_avg
            beta1 = 0.9
            flats = []
            for m, o in zip(models, opts):
                for p in reg_params(m):
                    st = o.state[p]
                    st["exp_avg"].mul_(beta1).add_(p.grad,
                                                   alpha=1 - beta1)
                flat = torch.cat([o.state[p]["exp_avg"].reshape(-1)
                                  for p in reg_params(m)])
                numel = flat.numel()
                align = nprocs * 32
                padded = (numel + align - 1) // align * align
                buf = torch.zeros(padded)
                buf[:numel] = flat
                flats.append(buf)
            n = nprocs
            chunk = flats[0].numel() // n
            # compress all chunks -> alltoall -> reduce own -> compress
            # own -> allgather -> decompress (ByteGrad wire)
            comps = [quant.compress_chunked(f, n) for f in flats]
            stride = quant.compressed_chunk_bytes(chunk)
            for rank in range(n):
                # after alltoall rank holds everyone's rank-th chunk
                dec = torch.stack([
                    quant.decompress_chunked(comps[src], n, chunk,
                                             target_chunk=rank)
                    .view(n, -1)[rank]
                    for src in range(n)])
                red = dec.mean(0)
                flats[rank].view(n, -1)[rank] = red
            # each rank compresses its own reduced chunk; allgather
            own_comp = [quant.compress_chunked(flats[r], n,
                                               target_chunk=r)
                        for r in range(n)]
            gathered = torch.zeros_like(comps[0])
            for r in range(n):
                gathered[r * stride:(r + 1) * stride] = \
                    own_comp[r][r * stride:(r + 1) * stride]
            final = quant.decompress_chunked(gathered, n, chunk)
            for m, o in zip(models, opts):
                offset = 0
                for p in reg_params(m):
                    st = o.state[p]
                    st["exp_avg"].copy_(
                        final[offset:offset + p.numel()].view_as(p))
                    offset += p.numel()

        for m, o in zip(models, opts):
            o.step()

    return [torch.cat([p.detach().reshape(-1)
                       for p in m.parameters()]) for m in models]


def test_qadam_matches_golden():
    nprocs = 2
    out = run_multi_process(nprocs, _worker)
    golden = _simulate(nprocs)
    for rank in range(nprocs):
        assert torch.allclose(out[rank], golden[rank], atol=1e-6), (
            "rank %d deviates from golden QAdam trajectory (max %g)"
            % (rank, (out[rank] - golden[rank]).abs().max()))
