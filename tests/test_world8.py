"""Standing 8-process gloo suite: the full algorithm zoo, the bench
contract and multi-node-shaped hierarchical execution at the driver's
8-GPU world size, runnable on CPU (everything this
framework is for only exists at N>1; the 8-way schedule must be proven
before the driver's one-shot ``bench.py --gpus 8``).

Reference analog: the 2-node Buildkite suite
(.buildkite/scripts/benchmark_master.sh:79-160) ran every algorithm at
world size 8; here the same world size runs under gloo in-process.
"""

import json
import os
import subprocess
import sys

import pytest
import torch
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process

WORLD = 8


def _worker_algo8(rank, nprocs, algo_name, steps, kwargs):
    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()

    if algo_name == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        optimizer = QAdamOptimizer(model.parameters(), lr=1e-3,
                                   warmup_steps=3)
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

    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat, losses


@pytest.mark.parametrize("algo,kwargs", [
    ("gradient_allreduce", {}),
    ("gradient_allreduce", {"hierarchical": True}),
    ("bytegrad", {}),
    ("decentralized", {"peer_selection_mode": "all"}),
    ("decentralized", {"peer_selection_mode": "shift_one"}),
    ("low_precision_decentralized", {}),
    ("qadam", {}),
    ("async", {"sync_interval_ms": 20}),
])
def test_world8_algorithm(algo, kwargs):
    out = run_multi_process(WORLD, _worker_algo8,
                            args=(algo, 6, dict(kwargs)), timeout=420)
    flats = [o[0] for o in out]
    for f in flats:
        assert torch.isfinite(f).all()
    if algo in ("gradient_allreduce", "bytegrad", "qadam"):
        for r in range(1, WORLD):
            assert torch.equal(flats[0], flats[r]), (
                "rank %d diverged under %s at world %d"
                % (r, algo, WORLD))
    elif algo == "decentralized" and kwargs.get(
            "peer_selection_mode") == "all":
        # peer averaging with mode=all is a full allreduce: after the last
        # sync ranks differ only by the final local step
        for r in range(1, WORLD):
            assert torch.allclose(flats[0], flats[r], atol=0.5)
    else:
        # gossip algorithms mix gradually at 8 ranks; trajectories must
        # stay bounded and close in mean
        mean = torch.stack(flats).mean(0)
        for r in range(WORLD):
            assert torch.allclose(flats[r], mean, atol=1.0)


def _worker_hier_2x4(rank, nprocs, algo_name):
    """8 ranks as 2 pseudo-nodes x 4 local ranks: the TRUE multi-node
    hierarchical shape (intra reduce -> inter op on leaders -> intra
    bcast), including per-node intra communicator construction."""
    os.environ["NODE_RANK"] = str(rank // 4)
    os.environ["LOCAL_RANK"] = str(rank % 4)
    os.environ["LOCAL_WORLD_SIZE"] = "4"

    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    pg = bagua_amd.communication._get_default_group()
    my_node = rank // 4
    assert pg._get_intra_ranks() == [my_node * 4 + i for i in range(4)]
    assert pg._get_inter_ranks() == [0, 4]
    # the intra communicator name must be node-local (uid-exchange key
    # collision regression)
    intra = pg.get_intra_node_communicator()
    assert str(my_node) in intra.name

    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    algorithm = GlobalAlgorithmRegistry.get(algo_name)(hierarchical=True)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)
    for step in range(5):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


@pytest.mark.parametrize("algo", ["gradient_allreduce", "bytegrad"])
def test_world8_hierarchical_2x4(algo):
    out = run_multi_process(WORLD, _worker_hier_2x4, args=(algo,),
                            timeout=420)
    for r in range(1, WORLD):
        assert torch.equal(out[0], out[r]), (
            "2x4 hierarchical %s diverged at rank %d" % (algo, r))


def _worker_hier_vs_flat_2x4(rank, nprocs):
    """Hierarchical averaged-allreduce at 2x4 must equal the flat result
    (the intra-AVG -> inter-AVG composition is exact when node sizes are
    equal)."""
    os.environ["NODE_RANK"] = str(rank // 4)
    os.environ["LOCAL_RANK"] = str(rank % 4)
    os.environ["LOCAL_WORLD_SIZE"] = "4"

    import bagua_amd
    from bagua_amd.bucket import BaguaBucket, CentralizedSyncOp
    from bagua_amd.executor import execute_ops
    from bagua_amd.tensor import ensure_bagua_tensor

    bagua_amd.init_process_group()
    pg = bagua_amd.communication._get_default_group()

    torch.manual_seed(100 + rank)
    base = torch.randn(64)
    t_h = ensure_bagua_tensor(base.clone(), "h")
    b_h = BaguaBucket([t_h], "bh", flatten=True)
    b_h.ops.append(CentralizedSyncOp(hierarchical=True, average=True,
                                     scattergather=False, compression=None,
                                     group=None))
    execute_ops(b_h, pg)

    t_f = ensure_bagua_tensor(base.clone(), "f")
    b_f = BaguaBucket([t_f], "bf", flatten=True)
    b_f.ops.append(CentralizedSyncOp(hierarchical=False, average=True,
                                     scattergather=False, compression=None,
                                     group=None))
    execute_ops(b_f, pg)

    hier = t_h.tensor().clone()
    flat = t_f.tensor().clone()
    bagua_amd.deinit_process_group()
    return hier, flat


def test_world8_hierarchical_matches_flat():
    out = run_multi_process(WORLD, _worker_hier_vs_flat_2x4, timeout=420)
    for hier, flat in out:
        assert torch.allclose(hier, flat, atol=1e-5)
    for r in range(1, WORLD):
        assert torch.allclose(out[0][0], out[r][0], atol=1e-5)


def test_world8_bench_contract():
    """The exact launch the driver uses for SCALE, at 8 ranks on CPU."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--standalone", "--local-addr",
         "127.0.0.1", "bench.py", "--gpus", "8", "--steps", "2",
         "--warmup", "1", "--model", "mnist", "--batch-size", "4",
         "--dtype", "fp32"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-3000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 8
    assert rec["value"] > 0


def _worker_moe8(rank, nprocs):
    import torch.nn as nn

    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from bagua_amd.parallel.moe import is_moe_param
    from tests.test_moe import MoEModel

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = MoEModel(k=1, num_local_experts=1)  # 8 experts at world 8
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    assert model.moe.num_experts == nprocs
    for step in range(4):
        torch.manual_seed(700 + rank * 13 + step)
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


def test_world8_moe_expert_parallel():
    """EP over the full 8-rank alltoall: dense consensus, expert shards
    stay rank-local (the driver's world size)."""
    out = run_multi_process(WORLD, _worker_moe8, timeout=420)
    for r in range(1, WORLD):
        assert torch.equal(out[0][0], out[r][0]), "dense diverged"
    for r in range(1, WORLD):
        assert not torch.equal(out[0][1], out[r][1]), (
            "experts identical across EP ranks")


def _worker_qadam_2x4(rank, nprocs):
    """QAdam at 2 pseudo-nodes x 4: warmup -> compressed-momentum
    transition (algorithm re-init) under the hierarchical shape."""
    os.environ["NODE_RANK"] = str(rank // 4)
    os.environ["LOCAL_RANK"] = str(rank % 4)
    os.environ["LOCAL_WORLD_SIZE"] = "4"

    import bagua_amd
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry
    from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = QAdamOptimizer(model.parameters(), lr=1e-3,
                               warmup_steps=4)
    algorithm = GlobalAlgorithmRegistry.get("qadam")(optimizer,
                                                     hierarchical=True)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)
    for step in range(9):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(data), target)
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_world8_qadam_hierarchical_2x4():
    out = run_multi_process(WORLD, _worker_qadam_2x4, timeout=420)
    for r in range(1, WORLD):
        assert torch.equal(out[0], out[r]), (
            "2x4 hierarchical qadam diverged at rank %d" % r)
