"""Reference test-suite parity: multiple models per process, optimizer
state broadcast, process-group conversion, autograd allreduce
(reference: tests/torch_api/test_multi_models.py, test_broadcast_state.py,
test_process_group.py, data_parallel/functional.py)."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(6, 12)
        self.fc2 = nn.Linear(12, 3)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


def _worker_multi_models(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.bytegrad import ByteGradAlgorithm
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    results = []
    for i, algo in enumerate([GradientAllReduceAlgorithm(),
                              ByteGradAlgorithm()]):
        torch.manual_seed(13 + rank + i)
        model = Net()
        optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
        ddp = bagua_amd.DistributedDataParallel(
            model, optimizers=[optimizer], algorithm=algo)
        for step in range(5):
            torch.manual_seed(100 * i + step * 7 + rank)
            optimizer.zero_grad()
            loss = F.mse_loss(ddp(torch.randn(4, 6)), torch.randn(4, 3))
            loss.backward()
            optimizer.step()
        results.append(torch.cat([p.detach().reshape(-1)
                                  for p in model.parameters()]))
    bagua_amd.deinit_process_group()
    return results


def test_multiple_models_per_process():
    """Two DDP-wrapped models with different algorithms coexist; each
    keeps its ranks in consensus (reference: test_multi_models.py)."""
    out = run_multi_process(2, _worker_multi_models)
    assert torch.equal(out[0][0], out[1][0]), "model A diverged"
    assert torch.equal(out[0][1], out[1][1]), "model B diverged"


def _worker_broadcast_state(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.Adam(model.parameters(), lr=1e-2)
    # materialize DIFFERENT optimizer state per rank before wrapping
    for _ in range(2 + rank):
        optimizer.zero_grad()
        F.mse_loss(model(torch.randn(4, 6)), torch.randn(4, 3)).backward()
        optimizer.step()

    bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    # after wrapping, model weights AND optimizer tensor state must match
    # rank 0 (reference: bagua_distributed.py:229-323)
    params = torch.cat([p.detach().reshape(-1)
                        for p in model.parameters()])
    state = torch.cat([
        optimizer.state[p][k].reshape(-1)
        for p in model.parameters()
        for k in ("exp_avg", "exp_avg_sq")])
    bagua_amd.deinit_process_group()
    return params, state


def test_broadcast_module_and_optimizer_state():
    out = run_multi_process(2, _worker_broadcast_state)
    assert torch.equal(out[0][0], out[1][0]), "weights not broadcast"
    assert torch.equal(out[0][1], out[1][1]), "optimizer state not synced"


def _worker_process_group(rank, nprocs):
    import torch.distributed as dist

    import bagua_amd
    from bagua_amd.communication import ReduceOp
    from bagua_amd.data_parallel import to_bagua_process_group

    bagua_amd.init_process_group()
    # conversion from a torch group
    tg = dist.new_group(list(range(nprocs)))
    pg = to_bagua_process_group(tg)
    t = torch.ones(3) * (rank + 1)
    pg.get_global_communicator().allreduce_inplace(t, ReduceOp.SUM)
    # default conversion
    pg2 = to_bagua_process_group(None)
    assert pg2.ranks == list(range(nprocs))
    # intra/inter rank mappings on one node
    assert pg2._get_intra_ranks() == list(range(nprocs))
    assert pg2._get_inter_ranks() == [0]
    bagua_amd.deinit_process_group()
    return t


def test_process_group_conversion():
    out = run_multi_process(2, _worker_process_group)
    for t in out:
        assert torch.allclose(t, torch.ones(3) * 3)


def _worker_functional(rank, nprocs):
    import bagua_amd
    from bagua_amd.data_parallel.functional import all_reduce
    from bagua_amd.communication import ReduceOp

    bagua_amd.init_process_group()
    x = torch.ones(4, requires_grad=True)
    y = all_reduce(x * (rank + 1), op=ReduceOp.SUM)
    y.sum().backward()
    grad = x.grad.clone()
    bagua_amd.deinit_process_group()
    return y.detach(), grad


def test_functional_allreduce_autograd():
    out = run_multi_process(2, _worker_functional)
    for rank, (y, grad) in enumerate(out):
        assert torch.allclose(y, torch.ones(4) * 3)  # 1 + 2
        # backward allreduces the ones-grad -> 2, scaled by (rank+1)
        assert torch.allclose(grad, torch.ones(4) * 2 * (rank + 1))


def _worker_duplicate_detection(rank, nprocs):
    import bagua_amd
    from bagua_amd.backend import BaguaBackend
    from bagua_amd.bucket import BaguaBucket
    from bagua_amd.tensor import ensure_bagua_tensor

    bagua_amd.init_process_group()
    group = bagua_amd.communication._get_default_group()
    backend = BaguaBackend(group)
    t = torch.randn(8)
    b1 = BaguaBucket([ensure_bagua_tensor(t, "x")], "b1", flatten=True)
    t2 = torch.randn(8)
    b2 = BaguaBucket([ensure_bagua_tensor(t2, "x")], "b2", flatten=True)
    try:
        backend.register_ordered_buckets([b1, b2])
        raised = False
    except ValueError:
        raised = True
    bagua_amd.deinit_process_group()
    return raised


def test_duplicate_tensor_name_rejected():
    out = run_multi_process(1, _worker_duplicate_detection)
    assert out[0], "duplicate tensor registration was not rejected"


class PartiallyUsedNet(nn.Module):
    """Second branch unused when flag is off (unused-parameter case)."""

    def __init__(self):
        super().__init__()
        self.a = nn.Linear(6, 3)
        self.b = nn.Linear(6, 3)  # never used

    def forward(self, x):
        return self.a(x)


def _worker_unused_params(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = PartiallyUsedNet()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm(),
        find_unused_parameters=True)
    for step in range(4):
        torch.manual_seed(900 + rank * 7 + step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(torch.randn(4, 6)), torch.randn(4, 3))
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1)
                      for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_find_unused_parameters():
    """Unused parameters must not wedge the in-order bucket queue; used
    parameters stay in consensus."""
    out = run_multi_process(2, _worker_unused_params, timeout=120)
    assert torch.equal(out[0], out[1]), "ranks diverged with unused params"


def _worker_ignore_list(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)  # different init on purpose
    model = Net()
    model._bagua_params_and_buffers_to_ignore = ["fc2.weight", "fc2.bias"]
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    names = {n for n, _ in ddp.inner.bagua_build_params()}
    assert "fc2.weight" not in names
    for step in range(3):
        torch.manual_seed(700 + rank + step)
        optimizer.zero_grad()
        loss = F.mse_loss(ddp(torch.randn(4, 6)), torch.randn(4, 3))
        loss.backward()
        optimizer.step()
    fc1 = model.fc1.weight.detach().reshape(-1)
    fc2 = model.fc2.weight.detach().reshape(-1)
    bagua_amd.deinit_process_group()
    return fc1, fc2


def test_params_to_ignore_stay_local():
    out = run_multi_process(2, _worker_ignore_list)
    assert torch.equal(out[0][0], out[1][0]), "synced param diverged"
    assert not torch.equal(out[0][1], out[1][1]), (
        "ignored param was synced")


def test_c_abi_symbols_present():
    """The C ABI for non-Python hosts must export its surface
    (reference: bagua-core-c)."""
    import ctypes

    from bagua_amd import _C

    lib = ctypes.CDLL(_C.__file__)
    for sym in ("bagua_comm_create", "bagua_comm_destroy",
                "bagua_comm_rank", "bagua_comm_nranks", "bagua_comm_abort",
                "bagua_comm_allreduce_inplace", "bagua_comm_broadcast",
                "bagua_comm_allgather_inplace", "bagua_nccl_unique_id",
                # full collective mirror (reference:
                # bagua-core-c/src/lib.rs:23-347)
                "bagua_comm_reduce_inplace",
                "bagua_comm_reduce_scatter_inplace",
                "bagua_comm_alltoall", "bagua_comm_send", "bagua_comm_recv",
                "bagua_comm_gather", "bagua_comm_scatter",
                "bagua_comm_group_start", "bagua_comm_group_end",
                "bagua_comm_barrier"):
        assert hasattr(lib, sym), "missing C ABI symbol %s" % sym


def _worker_with_bagua(rank, nprocs):
    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )

    bagua_amd.init_process_group()
    torch.manual_seed(13 + rank)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    # legacy API: function form + opt-in method patch
    bagua_amd.patch_torch_module()
    model = model.with_bagua([optimizer], GradientAllReduceAlgorithm())
    assert hasattr(model, "bagua_ddp")
    assert model.bagua_optimizers == [optimizer]
    assert len(model.bagua_buckets) > 0
    for step in range(4):
        torch.manual_seed(300 + rank + step)
        optimizer.zero_grad()
        loss = F.mse_loss(model(torch.randn(4, 6)), torch.randn(4, 3))
        loss.backward()
        optimizer.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


def test_with_bagua_legacy_api():
    out = run_multi_process(2, _worker_with_bagua)
    assert torch.equal(out[0], out[1]), "with_bagua ranks diverged"
