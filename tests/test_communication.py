"""Communication primitives vs expected math, 2-process gloo
(reference pattern: tests/comm/test_communicator.py)."""

import pytest
import torch

from tests.internal.multi_process import run_multi_process


def _worker_collectives(rank, nprocs):
    import bagua_amd
    from bagua_amd.communication import ReduceOp

    bagua_amd.init_process_group()
    results = {}

    # allreduce AVG
    t = torch.ones(16) * (rank + 1)
    bagua_amd.allreduce_inplace(t, op=ReduceOp.AVG)
    results["allreduce_avg"] = t.clone()

    # allreduce SUM
    t = torch.ones(8) * (rank + 1)
    bagua_amd.allreduce_inplace(t, op=ReduceOp.SUM)
    results["allreduce_sum"] = t.clone()

    # broadcast
    t = torch.arange(4.0) if rank == 0 else torch.zeros(4)
    bagua_amd.broadcast(t, src=0)
    results["broadcast"] = t.clone()

    # allgather
    send = torch.ones(4) * rank
    recv = torch.zeros(4 * nprocs)
    bagua_amd.allgather(send, recv)
    results["allgather"] = recv.clone()

    # reduce to 0
    t = torch.ones(4) * (rank + 1)
    bagua_amd.reduce_inplace(t, dst=0, op=ReduceOp.SUM)
    results["reduce"] = t.clone()

    # reduce_scatter
    send = torch.arange(float(2 * nprocs)) + rank
    recv = torch.zeros(2)
    bagua_amd.reduce_scatter(send, recv, op=ReduceOp.SUM)
    results["reduce_scatter"] = recv.clone()

    # alltoall
    send = torch.ones(nprocs * 2) * rank
    recv = torch.zeros(nprocs * 2)
    bagua_amd.alltoall(send, recv)
    results["alltoall"] = recv.clone()

    # scatter / gather
    if rank == 0:
        send = torch.arange(float(nprocs * 3))
    else:
        send = torch.zeros(nprocs * 3)
    recv = torch.zeros(3)
    bagua_amd.scatter(send, recv, src=0)
    results["scatter"] = recv.clone()

    grecv = torch.zeros(nprocs * 3)
    bagua_amd.gather(recv, grecv, dst=0)
    results["gather"] = grecv.clone()

    # send/recv ring
    t = torch.ones(4) * rank
    out = torch.zeros(4)
    peer_up = (rank + 1) % nprocs
    peer_down = (rank - 1) % nprocs
    c = bagua_amd.communication._get_default_group().get_global_communicator()
    c.group_start()
    c.send(t, peer_up)
    c.recv(out, peer_down)
    c.group_end()
    results["sendrecv"] = out.clone()

    # broadcast_object
    obj = {"a": rank, "b": [1, 2, 3]} if rank == 0 else None
    obj = bagua_amd.broadcast_object(obj, src=0)
    results["broadcast_object"] = obj

    bagua_amd.barrier()
    bagua_amd.deinit_process_group()
    return results


def test_collectives_world2():
    nprocs = 2
    results = run_multi_process(nprocs, _worker_collectives)
    for rank, res in enumerate(results):
        assert torch.allclose(res["allreduce_avg"], torch.ones(16) * 1.5)
        assert torch.allclose(res["allreduce_sum"], torch.ones(8) * 3)
        assert torch.allclose(res["broadcast"], torch.arange(4.0))
        expect_ag = torch.cat([torch.ones(4) * r for r in range(nprocs)])
        assert torch.allclose(res["allgather"], expect_ag)
        if rank == 0:
            assert torch.allclose(res["reduce"], torch.ones(4) * 3)
        # reduce_scatter: sum over ranks of (arange(4)+rank), rank r gets
        # chunk r
        total = (torch.arange(4.0) + 0) + (torch.arange(4.0) + 1)
        assert torch.allclose(res["reduce_scatter"],
                              total.view(nprocs, 2)[rank])
        expect_a2a = torch.cat(
            [torch.ones(2) * r for r in range(nprocs)])
        assert torch.allclose(res["alltoall"], expect_a2a)
        assert torch.allclose(res["scatter"],
                              torch.arange(float(nprocs * 3)).view(
                                  nprocs, 3)[rank])
        if rank == 0:
            assert torch.allclose(res["gather"],
                                  torch.arange(float(nprocs * 3)))
        assert torch.allclose(res["sendrecv"],
                              torch.ones(4) * ((rank - 1) % nprocs))
        assert res["broadcast_object"] == {"a": 0, "b": [1, 2, 3]}


def _worker_subgroup(rank, nprocs):
    import bagua_amd
    from bagua_amd.communication import ReduceOp

    bagua_amd.init_process_group()
    group = bagua_amd.new_group([0, 1])
    comm = group.get_global_communicator()
    t = torch.ones(4) * (rank + 1)
    comm.allreduce_inplace(t, ReduceOp.SUM)
    bagua_amd.deinit_process_group()
    return t


@pytest.mark.parametrize("nprocs", [2])
def test_subgroup(nprocs):
    results = run_multi_process(nprocs, _worker_subgroup)
    for res in results:
        assert torch.allclose(res, torch.ones(4) * 3)


def _worker_alltoall_v(rank, nprocs):
    import bagua_amd

    bagua_amd.init_process_group()
    # rank r sends (p+1) elements of value r to each peer p
    send_counts = [p + 1 for p in range(nprocs)]
    send_displs = [sum(send_counts[:p]) for p in range(nprocs)]
    send = torch.cat([torch.ones(c) * rank for c in send_counts])
    recv_counts = [rank + 1] * nprocs
    recv_displs = [sum(recv_counts[:p]) for p in range(nprocs)]
    recv = torch.zeros(sum(recv_counts))
    bagua_amd.alltoall_v(send, send_counts, send_displs,
                         recv, recv_counts, recv_displs)
    bagua_amd.deinit_process_group()
    return recv


def test_alltoall_v():
    nprocs = 2
    out = run_multi_process(nprocs, _worker_alltoall_v)
    for rank, recv in enumerate(out):
        expect = torch.cat([torch.ones(rank + 1) * p
                            for p in range(nprocs)])
        assert torch.allclose(recv, expect)


def _worker_reinit_cycle(rank, nprocs):
    """deinit -> re-init -> train again in ONE process: epoch guards must
    give the re-created communicators fresh uid/handle store keys and
    the caches must not leak stale state (multi-model long-lived
    processes)."""
    import torch
    import torch.nn.functional as F

    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from tests.test_algorithms import Net, _make_data

    flats = []
    for cycle in range(2):
        bagua_amd.init_process_group()
        torch.manual_seed(13 + rank + 100 * cycle)
        model = Net()
        optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
        ddp = bagua_amd.DistributedDataParallel(
            model, optimizers=[optimizer],
            algorithm=GradientAllReduceAlgorithm())
        for step in range(3):
            data, target = _make_data(rank, step)
            optimizer.zero_grad()
            F.mse_loss(ddp(data), target).backward()
            optimizer.step()
        flats.append(torch.cat([p.detach().reshape(-1)
                                for p in model.parameters()]))
        bagua_amd.deinit_process_group()
    return flats


def test_deinit_reinit_cycle():
    out = run_multi_process(2, _worker_reinit_cycle)
    for cycle in range(2):
        assert torch.equal(out[0][cycle], out[1][cycle]), (
            "cycle %d diverged after re-init" % cycle)
