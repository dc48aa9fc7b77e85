"""Spawn-N-process single-node test harness.

Mirrors the reference's test pattern (tests/internal/multi_process.py:9-53):
each worker gets a hand-rolled env (WORLD_SIZE/RANK/LOCAL_RANK/MASTER_*)
on a fresh port, runs the target function, and ships results back through
a multiprocessing queue. On the CPU CI this exercises the gloo path; the
same harness with cuda devices runs under ``-m gpu`` on the MI355X box.
"""

import multiprocessing as mp
import os
import pickle
import socket
import traceback


def find_free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(fn, rank, nprocs, port, args, queue):
    os.environ["WORLD_SIZE"] = str(nprocs)
    os.environ["LOCAL_WORLD_SIZE"] = str(nprocs)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["NODE_RANK"] = "0"
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        # N procs x default-N intra-op threads thrashes the box at world 8
        import torch

        torch.set_num_threads(max(1, (os.cpu_count() or 1) // nprocs))
    except Exception:
        pass
    try:
        result = fn(rank, nprocs, *args)
        queue.put((rank, "ok", pickle.dumps(result)))
    except Exception:
        queue.put((rank, "error", traceback.format_exc()))


_PORT_RACE_MARKERS = ("Address already in use", "Connection refused",
                      "timed out", "EADDRINUSE", "connectFullMesh")


def run_multi_process(nprocs, fn, args=(), timeout=180, retries=2):
    """Run ``fn(rank, nprocs, *args)`` in ``nprocs`` spawned processes.
    Returns results ordered by rank. Raises on any worker failure.
    Rendezvous-port races (another process grabbed the ephemeral port
    between probe and bind) are retried with a fresh port."""
    last = None
    for _ in range(retries + 1):
        try:
            return _run_once(nprocs, fn, args, timeout)
        except RuntimeError as e:
            msg = str(e)
            if not any(m in msg for m in _PORT_RACE_MARKERS):
                raise
            last = e
    raise last


def _run_once(nprocs, fn, args, timeout):
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    port = find_free_port()
    procs = []
    for rank in range(nprocs):
        p = ctx.Process(target=_worker,
                        args=(fn, rank, nprocs, port, args, queue))
        p.start()
        procs.append(p)

    results = {}
    errors = []
    for _ in range(nprocs):
        try:
            rank, status, payload = queue.get(timeout=timeout)
        except Exception:
            for p in procs:
                p.terminate()
            raise TimeoutError("worker result timeout after %ss" % timeout)
        if status == "ok":
            results[rank] = pickle.loads(payload)
        else:
            errors.append("rank %d:\n%s" % (rank, payload))

    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()

    if errors:
        raise RuntimeError("worker failures:\n" + "\n".join(errors))
    return [results[r] for r in range(nprocs)]
