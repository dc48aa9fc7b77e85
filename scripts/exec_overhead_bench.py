#!/usr/bin/env python3
"""Python-executor vs native-executor dispatch overhead on GPU
(VERDICT r1 item 8 evidence).

Measures, at world 1 on one MI355X:
  1. VGG16 GradientAllReduce e2e img/s with the C++ BucketExecutor
     (BAGUA_NATIVE_SCHEDULER=1) vs the Python executor (=0) — the
     end-to-end cost of Python bucket dispatch on the multi-bucket
     centralized path;
  2. host ns per bucket dispatch (backend.exec_host_ns) for the
     algorithms whose ops run ONLY through the Python executor today
     (decentralized, low_precision_decentralized) — these pack ALL
     parameters into ONE bucket per step, so per-step Python overhead is
     a single dispatch.

Usage (GPU box): python scripts/exec_overhead_bench.py
"""

import json
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run(algo, native, steps=25, warmup=8):
    os.environ["BAGUA_NATIVE_SCHEDULER"] = "1" if native else "0"
    import bagua_amd
    from bagua_amd.models import create_model
    from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry

    torch.manual_seed(3)
    model = create_model("vgg16").cuda().to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    if algo in ("gradient_allreduce", "bytegrad"):
        from bagua_amd.contrib import FusedSGD

        optimizer = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    else:  # low-prec decentralized refuses fused optimizers
        optimizer = torch.optim.SGD(model.parameters(), lr=0.01,
                                    momentum=0.9)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GlobalAlgorithmRegistry.get(algo)())
    backend = ddp.inner.bagua_backend

    data = torch.randn(32, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    target = torch.randint(0, 1000, (32,), device="cuda")

    def step():
        optimizer.zero_grad()
        loss = F.cross_entropy(ddp(data), target)
        loss.backward()
        optimizer.step()

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    backend.exec_host_ns = 0
    backend.exec_dispatches = 0
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    n_buckets = len(ddp.inner.bagua_buckets)
    native_engaged = any(getattr(b, "_native_idx", None) is not None
                         for b in ddp.inner.bagua_buckets)
    rec = {
        "algo": algo,
        "native_scheduler": native,
        "native_engaged": native_engaged,
        "img_per_s": 32 * steps / elapsed,
        "ms_per_step": elapsed / steps * 1e3,
        "n_buckets": n_buckets,
        "dispatches": backend.exec_dispatches,
        "host_us_per_dispatch": (backend.exec_host_ns / 1e3
                                 / max(1, backend.exec_dispatches)),
        "host_dispatch_us_per_step": (backend.exec_host_ns / 1e3 / steps),
    }
    print(json.dumps(rec), flush=True)
    return rec


def main():
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29597")
    torch.cuda.set_device(0)
    torch.backends.cudnn.benchmark = True

    import bagua_amd

    bagua_amd.init_process_group()
    for algo, native in [
        ("gradient_allreduce", True),
        ("gradient_allreduce", False),
        ("bytegrad", True),
        ("bytegrad", False),
        ("decentralized", False),
        ("low_precision_decentralized", False),
    ]:
        run(algo, native)
    bagua_amd.deinit_process_group()


if __name__ == "__main__":
    main()
