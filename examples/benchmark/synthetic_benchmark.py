#!/usr/bin/env python3
"""Synthetic data-parallel benchmark (Horovod-derived protocol).

MI355X port of the reference benchmark
(reference: examples/benchmark/synthetic_benchmark.py:1-120): trains a
model on a fixed random batch and prints "Img/sec per GPU" over
``--num-iters`` iterations of ``--num-batches-per-iter`` batches, for any
of the six algorithms. ``--deterministic`` seeds everything and prints the
final loss so CI can assert it exactly
(reference: .buildkite/scripts/benchmark_master.sh:79-110).

Launch:
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/benchmark/synthetic_benchmark.py --algorithm bytegrad
"""

import argparse
import timeit

import numpy as np
import torch
import torch.nn.functional as F

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd
from bagua_amd import env
from bagua_amd.models import create_model
from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--model", type=str, default="vgg16")
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--num-warmup-batches", type=int, default=10)
    p.add_argument("--num-batches-per-iter", type=int, default=10)
    p.add_argument("--num-iters", type=int, default=10)
    p.add_argument("--algorithm", type=str, default="gradient_allreduce",
                   choices=GlobalAlgorithmRegistry.names())
    p.add_argument("--deterministic", action="store_true")
    p.add_argument("--fuse-optimizer", action="store_true")
    p.add_argument("--amp", action="store_true",
                   help="bf16 autocast (MI355X-native mixed precision)")
    p.add_argument("--async-sync-interval-ms", type=int, default=500)
    p.add_argument("--async-warmup-steps", type=int, default=100)
    return p.parse_args()


def main():
    args = parse_args()
    use_cuda = torch.cuda.is_available()
    if args.deterministic:
        torch.manual_seed(env.get_rank())
        np.random.seed(env.get_rank())
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False
    else:
        torch.backends.cudnn.benchmark = True

    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()

    device = torch.device("cuda" if use_cuda else "cpu")
    model = create_model(args.model).to(device)
    if use_cuda and not args.deterministic:
        model = model.to(memory_format=torch.channels_last)

    if args.algorithm == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        optimizer = QAdamOptimizer(model.parameters(), lr=0.01 * 0.1,
                                   warmup_steps=100)
        algorithm = GlobalAlgorithmRegistry.get("qadam")(optimizer)
    else:
        optimizer = torch.optim.SGD(model.parameters(), lr=0.01)
        if args.algorithm == "async":
            algorithm = GlobalAlgorithmRegistry.get("async")(
                sync_interval_ms=args.async_sync_interval_ms,
                warmup_steps=args.async_warmup_steps)
        else:
            algorithm = GlobalAlgorithmRegistry.get(args.algorithm)()

    if args.fuse_optimizer:
        from bagua_amd.contrib import fuse_optimizer

        optimizer = fuse_optimizer(optimizer, do_flatten=True)

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm,
        gradient_as_bucket_view=not args.fuse_optimizer)

    data = torch.randn(args.batch_size, 3, 224, 224, device=device)
    if use_cuda and not args.deterministic:
        data = data.to(memory_format=torch.channels_last)
    target = torch.randint(0, 1000, (args.batch_size,), device=device)
    last_loss = [None]

    def benchmark_step():
        optimizer.zero_grad()
        with torch.autocast(device.type, dtype=torch.bfloat16,
                            enabled=args.amp):
            loss = F.cross_entropy(ddp(data), target)
        loss.backward()
        if args.fuse_optimizer:
            optimizer.fuse_step()
        else:
            optimizer.step()
        last_loss[0] = loss

    def log(s):
        if env.get_rank() == 0:
            print(s, flush=True)

    log("Model: %s, batch size %d, %d GPUs" %
        (args.model, args.batch_size, env.get_world_size()))
    timeit.timeit(benchmark_step, number=args.num_warmup_batches)

    img_secs = []
    for _ in range(args.num_iters):
        t = timeit.timeit(benchmark_step,
                          number=args.num_batches_per_iter)
        img_sec = args.batch_size * args.num_batches_per_iter / t
        log("Iter #%d: %.1f img/sec per GPU" % (len(img_secs), img_sec))
        img_secs.append(img_sec)

    img_sec_mean = np.mean(img_secs)
    img_sec_conf = 1.96 * np.std(img_secs)
    log("Img/sec per GPU: %.1f +-%.1f" % (img_sec_mean, img_sec_conf))
    log("Total img/sec on %d GPU(s): %.1f +-%.1f" %
        (env.get_world_size(), env.get_world_size() * img_sec_mean,
         env.get_world_size() * img_sec_conf))
    if args.deterministic:
        if use_cuda:
            torch.cuda.synchronize()
        log("Final loss: %.6f" % last_loss[0].item())

    if args.algorithm == "async":
        ddp.inner.bagua_algorithm.abort(ddp)


if __name__ == "__main__":
    main()
