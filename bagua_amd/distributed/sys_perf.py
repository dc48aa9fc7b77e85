"""Communication micro-benchmark CLI — ``python -m bagua_amd.distributed.sys_perf``.

MI355X equivalent of the reference's ``bagua_sys_perf`` tool: sweeps
message sizes over the core collectives (allreduce / allgather /
reduce_scatter / alltoall) on the current process group and prints
algorithm bandwidth per size. Launch with torchrun/bagua_amd.distributed.run,
one rank per GPU; on one node this measures RCCL over xGMI directly.

busbw factors follow the standard nccl-tests conventions:
    allreduce      2(n-1)/n x size / time
    allgather      (n-1)/n x size / time   (size = full buffer)
    reduce_scatter (n-1)/n x size / time
    alltoall       (n-1)/n x size / time
"""

import argparse
import json
import time

import torch


def parse_args(args=None):
    p = argparse.ArgumentParser()
    p.add_argument("--min-bytes", type=int, default=1 << 20)
    p.add_argument("--max-bytes", type=int, default=1 << 28)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--collectives", type=str,
                   default="allreduce,allgather,reduce_scatter,alltoall")
    return p.parse_args(args)


def main(args=None):
    args = parse_args(args)
    import bagua_amd
    from bagua_amd import env

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()
    comm = bagua_amd.communication._get_default_group() \
        .get_global_communicator()
    n = comm.nranks()
    device = "cuda" if use_cuda else "cpu"

    def run_op(name, buf):
        if name == "allreduce":
            comm.allreduce_inplace(buf, bagua_amd.ReduceOp.SUM)
        elif name == "allgather":
            comm.allgather_inplace(buf)
        elif name == "reduce_scatter":
            comm.reduce_scatter_inplace(buf)
        elif name == "alltoall":
            comm.alltoall_inplace(buf)

    busbw_factor = {
        "allreduce": lambda s: 2 * (n - 1) / n * s,
        "allgather": lambda s: (n - 1) / n * s,
        "reduce_scatter": lambda s: (n - 1) / n * s,
        "alltoall": lambda s: (n - 1) / n * s,
    }

    results = []
    size = args.min_bytes
    while size <= args.max_bytes:
        numel = size // 4
        numel = (numel // max(n, 1)) * max(n, 1)
        buf = torch.ones(numel, device=device)
        for name in args.collectives.split(","):
            for _ in range(args.warmup):
                run_op(name, buf)
            bagua_amd.barrier()
            if use_cuda:
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                run_op(name, buf)
            bagua_amd.barrier()
            if use_cuda:
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.iters
            busbw = busbw_factor[name](numel * 4) / dt / 1e9
            results.append({"collective": name, "bytes": numel * 4,
                            "time_us": dt * 1e6, "busbw_GBs": busbw})
            if env.get_rank() == 0:
                print("%-14s %12d B  %10.1f us  busbw %8.2f GB/s"
                      % (name, numel * 4, dt * 1e6, busbw))
        size *= 4

    if env.get_rank() == 0:
        print(json.dumps({"world_size": n, "results": results}))


if __name__ == "__main__":
    main()
