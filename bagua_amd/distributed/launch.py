"""Legacy single-use launcher — ``python -m bagua_amd.distributed.launch``.

Subprocess-per-local-rank with hand-set RANK/LOCAL_RANK/MASTER_* env
(reference: bagua/distributed/launch.py:1-343, itself a fork of the old
torch.distributed.launch). Prefer ``bagua_amd.distributed.run``.
"""

import argparse
import os
import signal
import subprocess
import sys


def parse_args(args=None):
    parser = argparse.ArgumentParser(
        description="bagua_amd legacy launcher")
    parser.add_argument("--nnodes", type=int, default=1)
    parser.add_argument("--node_rank", type=int, default=0)
    parser.add_argument("--nproc_per_node", type=int, default=1)
    parser.add_argument("--master_addr", default="127.0.0.1", type=str)
    parser.add_argument("--master_port", default=29500, type=int)
    parser.add_argument("-m", "--module", default=False,
                        action="store_true")
    parser.add_argument("--no_python", default=False, action="store_true")
    parser.add_argument("--default_bucket_size", type=int,
                        default=32 * 1024 * 1024)
    parser.add_argument("--autotune_level", type=int, default=0)
    parser.add_argument("training_script", type=str)
    parser.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return parser.parse_args(args)


def main(args=None):
    args = parse_args(args)
    world_size = args.nnodes * args.nproc_per_node

    current_env = os.environ.copy()
    # keep in-tree bagua_amd importable from worker scripts in other dirs
    current_env["PYTHONPATH"] = os.pathsep.join(
        [os.getcwd()] + [p for p in
                         current_env.get("PYTHONPATH", "").split(os.pathsep)
                         if p])
    current_env["MASTER_ADDR"] = args.master_addr
    current_env["MASTER_PORT"] = str(args.master_port)
    current_env["WORLD_SIZE"] = str(world_size)
    current_env["LOCAL_WORLD_SIZE"] = str(args.nproc_per_node)
    current_env["NODE_RANK"] = str(args.node_rank)
    current_env["BAGUA_DEFAULT_BUCKET_SIZE"] = str(
        args.default_bucket_size)
    current_env["BAGUA_AUTOTUNE"] = str(args.autotune_level)

    processes = []
    for local_rank in range(args.nproc_per_node):
        env = current_env.copy()
        env["RANK"] = str(args.nproc_per_node * args.node_rank
                          + local_rank)
        env["LOCAL_RANK"] = str(local_rank)

        cmd = []
        if not args.no_python:
            cmd = [sys.executable, "-u"]
            if args.module:
                cmd.append("-m")
        cmd.append(args.training_script)
        cmd.extend(args.training_script_args)
        processes.append(subprocess.Popen(cmd, env=env))

    try:
        alive = list(processes)
        while alive:
            finished = []
            for p in alive:
                ret = p.poll()
                if ret is not None:
                    if ret != 0:
                        for q in processes:
                            if q.poll() is None:
                                q.send_signal(signal.SIGTERM)
                        raise subprocess.CalledProcessError(
                            ret, p.args)
                    finished.append(p)
            alive = [p for p in alive if p not in finished]
            if alive:
                import time

                time.sleep(0.2)
    except KeyboardInterrupt:
        for p in processes:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        raise


if __name__ == "__main__":
    main()
