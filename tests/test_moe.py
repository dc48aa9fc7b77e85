"""MoE expert parallelism + MoE-aware checkpointing, 2-proc gloo
(reference patterns: tests/moe/, benchmark_master.sh:114-160)."""

import os
import tempfile

import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


class MoEModel(nn.Module):
    def __init__(self, hidden=16, num_local_experts=2, k=1):
        super().__init__()
        from bagua_amd.parallel.moe import MoE

        self.fc1 = nn.Linear(8, hidden)
        self.moe = MoE(hidden,
                       expert=nn.Sequential(nn.Linear(hidden, 32),
                                            nn.ReLU(),
                                            nn.Linear(32, hidden)),
                       num_local_experts=num_local_experts, k=k)
        self.out = nn.Linear(hidden, 4)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        x, l_aux, _ = self.moe(x)
        return self.out(x), l_aux


def _worker_moe_train(rank, nprocs, k):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from bagua_amd.parallel.moe import is_moe_param

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = MoEModel(k=k)
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    # expert params must be excluded from DP sync
    dp_names = {n for n, _ in ddp.inner.bagua_build_params()}
    assert not any("bagua_experts" in n for n in dp_names)
    n_expert_params = sum(1 for p in model.parameters() if is_moe_param(p))
    assert n_expert_params > 0

    losses = []
    for step in range(8):
        torch.manual_seed(500 + rank * 31 + step)
        data = torch.randn(16, 8)
        target = torch.randn(16, 4)
        optimizer.zero_grad()
        out, l_aux = ddp(data)
        loss = F.mse_loss(out, target) + 0.01 * l_aux
        loss.backward()
        optimizer.step()
        losses.append(loss.item())

    dense = torch.cat([p.detach().reshape(-1)
                       for n, p in sorted(model.named_parameters())
                       if not is_moe_param(p)])
    expert = torch.cat([p.detach().reshape(-1)
                        for n, p in sorted(model.named_parameters())
                        if is_moe_param(p)])
    bagua_amd.deinit_process_group()
    return dense, expert, losses


def test_moe_train_top1():
    out = run_multi_process(2, _worker_moe_train, args=(1,))
    # dense params stay in consensus; expert params differ per rank
    assert torch.equal(out[0][0], out[1][0]), "dense params diverged"
    assert not torch.equal(out[0][1], out[1][1]), (
        "expert params identical across EP ranks — EP not sharding")
    assert all(torch.isfinite(torch.tensor(r[2])).all() for r in out)


def test_moe_train_top2():
    out = run_multi_process(2, _worker_moe_train, args=(2,))
    assert torch.equal(out[0][0], out[1][0])


def _worker_moe_init_broadcast(rank, nprocs):
    """Regression: the init state broadcast must NOT overwrite rank-local
    expert parameters with rank 0's (engine broadcast used the
    full state_dict; reference broadcasts bagua_build_params() which
    excludes MoE params — bagua_distributed.py:172)."""
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from bagua_amd.parallel.moe import is_moe_param

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)  # per-rank expert init
    model = MoEModel(k=1)
    optimizer = torch.optim.Adam(model.parameters(), lr=1e-3)
    # materialize per-rank optimizer state for the experts BEFORE wrapping,
    # so the optimizer-state broadcast path is exercised too
    out, l_aux = model(torch.randn(8, 8))
    (out.sum() + 0.01 * l_aux).backward()
    optimizer.step()
    optimizer.zero_grad()
    expert_state_before = {
        id(p): {k: v.clone() for k, v in optimizer.state[p].items()
                if isinstance(v, torch.Tensor)}
        for p in model.parameters() if is_moe_param(p) and p in
        optimizer.state}

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    del ddp

    dense = torch.cat([p.detach().reshape(-1)
                       for n, p in sorted(model.named_parameters())
                       if not is_moe_param(p)])
    expert = torch.cat([p.detach().reshape(-1)
                        for n, p in sorted(model.named_parameters())
                        if is_moe_param(p)])
    # expert optimizer state must also survive the broadcast
    state_ok = all(
        torch.equal(optimizer.state[p][k], expert_state_before[id(p)][k])
        for p in model.parameters()
        if is_moe_param(p) and id(p) in expert_state_before
        for k in expert_state_before[id(p)])
    bagua_amd.deinit_process_group()
    return dense, expert, state_ok


def test_moe_init_broadcast_preserves_experts():
    out = run_multi_process(2, _worker_moe_init_broadcast)
    assert torch.equal(out[0][0], out[1][0]), "dense params not broadcast"
    assert not torch.equal(out[0][1], out[1][1]), (
        "init broadcast clobbered rank-local expert params")
    assert out[0][2] and out[1][2], (
        "init broadcast clobbered rank-local expert optimizer state")


def _worker_moe_checkpoint(rank, nprocs, path):
    import bagua_amd
    from bagua_amd.checkpoint import load_checkpoint, save_checkpoint

    from bagua_amd.parallel.moe import is_moe_param

    bagua_amd.init_process_group()
    torch.manual_seed(77)  # dense params identical across ranks (like DP)
    model = MoEModel()
    with torch.no_grad():  # expert shards differ per rank
        for p in model.parameters():
            if is_moe_param(p):
                p.add_(0.1 * (rank + 1))
    optimizer = torch.optim.Adam(model.parameters(), lr=1e-3)
    # one step so optimizer state materializes
    out, l_aux = model(torch.randn(8, 8))
    (out.sum() + l_aux).backward()
    optimizer.step()
    # emulate DP consistency: dense params averaged across ranks
    import torch.distributed as dist

    with torch.no_grad():
        for p in model.parameters():
            if not is_moe_param(p):
                dist.all_reduce(p)
                p.div_(nprocs)

    save_checkpoint(3, path, model, optimizer)

    ref = {n: p.detach().clone() for n, p in model.named_parameters()}

    # perturb then reload
    with torch.no_grad():
        for p in model.parameters():
            p.add_(1.0)
    it = load_checkpoint(path, model, optimizer)
    assert it == 3
    for n, p in model.named_parameters():
        assert torch.allclose(p, ref[n]), "param %s not restored" % n

    bagua_amd.deinit_process_group()
    return True


def test_moe_checkpoint_roundtrip():
    with tempfile.TemporaryDirectory() as tmp:
        out = run_multi_process(2, _worker_moe_checkpoint, args=(tmp,))
        assert all(out)


def test_dense_checkpoint_roundtrip():
    from bagua_amd.checkpoint import load_checkpoint, save_checkpoint

    with tempfile.TemporaryDirectory() as tmp:
        torch.manual_seed(1)
        model = nn.Linear(4, 4)
        opt = torch.optim.Adam(model.parameters())
        model(torch.randn(2, 4)).sum().backward()
        opt.step()
        save_checkpoint(7, tmp, model, opt)
        ref = {n: p.detach().clone() for n, p in model.named_parameters()}
        with torch.no_grad():
            for p in model.parameters():
                p.mul_(0)
        assert load_checkpoint(tmp, model, opt) == 7
        for n, p in model.named_parameters():
            assert torch.allclose(p, ref[n])
        assert os.path.exists(
            os.path.join(tmp, "iter_0000007", "mp_rank_00_model_states.pt"))


def test_gate_capacity_and_laux():
    """Top-1 gate respects capacity and emits a finite aux loss."""
    import torch as t

    from bagua_amd.parallel.moe.sharded_moe import top1gating

    t.manual_seed(3)
    logits = t.randn(64, 4)
    l_aux, combine, dispatch, counts = top1gating(
        logits, capacity_factor=1.0, min_capacity=4)
    assert t.isfinite(l_aux)
    cap = combine.shape[2]
    assert cap == max(4, 64 // 4)
    # no expert slot double-booked
    per_slot = dispatch.sum(dim=0)  # (experts, capacity)
    assert int(per_slot.max()) <= 1
    # every kept token has exactly one (expert, slot)
    per_token = dispatch.reshape(64, -1).sum(1)
    assert int(per_token.max()) <= 1


def test_gate_top2_weights_normalized():
    import torch as t

    from bagua_amd.parallel.moe.sharded_moe import top2gating

    t.manual_seed(4)
    logits = t.randn(32, 4)
    l_aux, combine, dispatch, counts = top2gating(
        logits, capacity_factor=1.0, min_capacity=4)
    sums = combine.reshape(32, -1).sum(1)
    kept = sums > 0
    assert t.all(sums[kept] <= 1.0 + 1e-5)
    assert t.isfinite(l_aux)


def _worker_moe_subgroup_ep(rank, nprocs):
    """ep_size < world_size: EP alltoall confined to a 2-rank sub-group
    at world 4 (goes beyond the reference, which hardcoded
    group=dist.group.WORLD — layer.py:84; the MOELayer group parameter is
    the DeepSpeed-lineage interface for this)."""
    import torch.distributed as dist

    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from bagua_amd.parallel.moe import MoE, is_moe_param

    bagua_amd.init_process_group()
    # two EP groups: {0,1} and {2,3}; every rank must create both
    g01 = dist.new_group([0, 1])
    g23 = dist.new_group([2, 3])
    ep_group = g01 if rank < 2 else g23

    torch.manual_seed(13 + rank)
    model = MoEModel(k=1)
    # swap in a sub-group MoE layer
    import torch.nn as nn

    model.moe = MoE(16,
                    expert=nn.Sequential(nn.Linear(16, 32), nn.ReLU(),
                                         nn.Linear(32, 16)),
                    num_local_experts=2, k=1,
                    expert_parallel_group=ep_group)
    assert model.moe.ep_size == 2
    assert model.moe.num_experts == 4  # 2 local x 2 ranks in the EP group

    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    for step in range(6):
        torch.manual_seed(900 + rank * 17 + step)
        data = torch.randn(16, 8)
        target = torch.randn(16, 4)
        optimizer.zero_grad()
        out, l_aux = ddp(data)
        loss = F.mse_loss(out, target) + 0.01 * l_aux
        loss.backward()
        optimizer.step()
        assert torch.isfinite(loss)

    dense = torch.cat([p.detach().reshape(-1)
                       for n, p in sorted(model.named_parameters())
                       if not is_moe_param(p)])
    expert = torch.cat([p.detach().reshape(-1)
                        for n, p in sorted(model.named_parameters())
                        if is_moe_param(p)])
    bagua_amd.deinit_process_group()
    return dense, expert


def test_moe_subgroup_expert_parallel():
    out = run_multi_process(4, _worker_moe_subgroup_ep)
    # dense params: full-world consensus
    for r in range(1, 4):
        assert torch.equal(out[0][0], out[r][0]), "dense diverged"
    # experts shard within each EP group (different per rank)
    assert not torch.equal(out[0][1], out[1][1])
    assert not torch.equal(out[2][1], out[3][1])
