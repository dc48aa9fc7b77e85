"""Multi-host fan-out launcher — ``baguarun``.

Replaces the reference's pssh-based baguarun
(bagua/script/baguarun.py:36-100) with plain ``ssh`` subprocesses (pssh
is not in this image). Each host runs ``bagua_amd.distributed.launch``
with the proper node_rank / master env. Single-node xGMI jobs don't need
this; it exists for multi-node parity.
"""

import argparse
import shlex
import subprocess
import sys


def parse_args(args=None):
    parser = argparse.ArgumentParser(description="bagua_amd ssh fan-out")
    parser.add_argument("--host_list", type=str, required=True,
                        help="comma separated host[:ssh_port] list; the "
                             "first host is the master")
    parser.add_argument("--nproc_per_node", type=int, default=8)
    parser.add_argument("--master_port", type=int, default=29500)
    parser.add_argument("--ssh_port", type=int, default=22)
    parser.add_argument("--default_bucket_size", type=int,
                        default=32 * 1024 * 1024)
    parser.add_argument("--autotune_level", type=int, default=0)
    parser.add_argument("training_script", type=str)
    parser.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return parser.parse_args(args)


def build_remote_command(args, node_rank: int, master_addr: str) -> str:
    cmd = [
        sys.executable, "-m", "bagua_amd.distributed.launch",
        "--nnodes", str(len(args.host_list.split(","))),
        "--node_rank", str(node_rank),
        "--nproc_per_node", str(args.nproc_per_node),
        "--master_addr", master_addr,
        "--master_port", str(args.master_port),
        "--default_bucket_size", str(args.default_bucket_size),
        "--autotune_level", str(args.autotune_level),
        args.training_script,
    ] + args.training_script_args
    return " ".join(shlex.quote(c) for c in cmd)


def main(args=None):
    args = parse_args(args)
    hosts = [h.strip() for h in args.host_list.split(",") if h.strip()]
    master_addr = hosts[0].split(":")[0]

    procs = []
    for node_rank, host in enumerate(hosts):
        hostname, _, port = host.partition(":")
        remote = build_remote_command(args, node_rank, master_addr)
        ssh_cmd = ["ssh", "-o", "StrictHostKeyChecking=no",
                   "-p", port or str(args.ssh_port), hostname, remote]
        procs.append(subprocess.Popen(ssh_cmd))

    rc = 0
    for p in procs:
        p.wait()
        rc = rc or p.returncode
    sys.exit(rc)


if __name__ == "__main__":
    main()
