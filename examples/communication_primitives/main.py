#!/usr/bin/env python3
"""Communication primitives demo / integration check
(reference: examples/communication_primitives/main.py:23-70).

Exercises every module-level collective and asserts the math.

Launch:
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/communication_primitives/main.py
"""

import torch

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd
from bagua_amd import ReduceOp, env


def main():
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()
    rank, world = env.get_rank(), env.get_world_size()
    device = "cuda" if use_cuda else "cpu"

    def check(name, got, expect):
        if use_cuda:
            torch.cuda.synchronize()
        assert torch.allclose(got.cpu(), expect), (
            "%s mismatch on rank %d: %s vs %s" % (name, rank, got, expect))
        if rank == 0:
            print("%-22s ok" % name)

    t = torch.ones(8, device=device) * (rank + 1)
    bagua_amd.allreduce_inplace(t, op=ReduceOp.SUM)
    check("allreduce SUM", t, torch.ones(8) * world * (world + 1) / 2)

    t = torch.ones(8, device=device) * (rank + 1)
    bagua_amd.allreduce_inplace(t, op=ReduceOp.AVG)
    check("allreduce AVG", t, torch.ones(8) * (world + 1) / 2)

    t = torch.arange(4.0, device=device) if rank == 0 \
        else torch.zeros(4, device=device)
    bagua_amd.broadcast(t, src=0)
    check("broadcast", t, torch.arange(4.0))

    send = torch.ones(4, device=device) * rank
    recv = torch.zeros(4 * world, device=device)
    bagua_amd.allgather(send, recv)
    check("allgather", recv,
          torch.cat([torch.ones(4) * r for r in range(world)]))

    send = torch.ones(2 * world, device=device) * rank
    recv = torch.zeros(2 * world, device=device)
    bagua_amd.alltoall(send, recv)
    check("alltoall", recv,
          torch.cat([torch.ones(2) * r for r in range(world)]))

    send = torch.arange(float(2 * world), device=device)
    recv = torch.zeros(2, device=device)
    bagua_amd.reduce_scatter(send, recv, op=ReduceOp.SUM)
    check("reduce_scatter", recv,
          torch.arange(float(2 * world)).view(world, 2)[rank] * world)

    obj = {"msg": "hello from rank 0"} if rank == 0 else None
    obj = bagua_amd.broadcast_object(obj, src=0)
    assert obj["msg"] == "hello from rank 0"

    bagua_amd.barrier()
    if rank == 0:
        print("all communication primitives verified on %d rank(s)"
              % world)


if __name__ == "__main__":
    main()
