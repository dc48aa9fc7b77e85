#!/usr/bin/env python3
"""Autotune end-to-end on GPU: does the tuner converge and does the
converged bucket size hold throughput?

Runs VGG16 GradientAllReduce at world 1 with aggressive tuner gates,
logs img/s per 25-step window together with the bucket size in force,
prints a JSON summary at the end.
"""

import json
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29599")
    os.environ["BAGUA_AUTOTUNE"] = "1"
    os.environ["BAGUA_AUTOTUNE_WARMUP_TIME_S"] = "3"
    os.environ["BAGUA_AUTOTUNE_SAMPLING_CONFIDENCE_TIME_S"] = "0.2"
    os.environ["BAGUA_AUTOTUNE_MAX_SAMPLES"] = "8"
    os.environ["BAGUA_AUTOTUNE_INTERVAL"] = "25"

    torch.cuda.set_device(0)
    torch.backends.cudnn.benchmark = True

    import bagua_amd
    from bagua_amd.models import create_model
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from bagua_amd.contrib import FusedSGD

    bagua_amd.init_process_group()
    torch.manual_seed(3)
    model = create_model("vgg16").cuda().to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    optimizer = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    data = torch.randn(32, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    target = torch.randint(0, 1000, (32,), device="cuda")

    def step():
        optimizer.zero_grad()
        loss = F.cross_entropy(ddp(data), target)
        loss.backward()
        optimizer.step()

    for _ in range(30):  # MIOpen find etc.
        step()
    torch.cuda.synchronize()

    windows = []
    total_steps = 700
    win = 25
    for w in range(total_steps // win):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(win):
            step()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        hp = ddp.inner._current_hp
        windows.append({
            "window": w,
            "img_per_s": 32 * win / dt,
            "bucket_size": hp.bucket_size,
            "n_buckets": len(ddp.inner.bagua_buckets),
            "completed": ddp.inner._autotune_completed,
        })
        print(json.dumps(windows[-1]), flush=True)

    first = windows[0]["img_per_s"]
    done = [x for x in windows if x["completed"]]
    summary = {
        "first_window_img_s": first,
        "converged": bool(done),
        "final_bucket_size": windows[-1]["bucket_size"],
        "final_n_buckets": windows[-1]["n_buckets"],
        "post_convergence_img_s": (sum(x["img_per_s"] for x in done[-4:])
                                   / max(1, len(done[-4:]))),
    }
    print("SUMMARY " + json.dumps(summary), flush=True)
    bagua_amd.deinit_process_group()


if __name__ == "__main__":
    main()
