#!/usr/bin/env python3
"""MNIST example (reference: examples/mnist/main.py shape).

There is no network access for the real dataset, so ``--synthetic``
(default) trains on random data of the MNIST shape; pass ``--data-dir``
if you have the tensors locally.

Launch:
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/mnist/main.py --algorithm gradient_allreduce
"""

import argparse

import torch
import torch.nn.functional as F
import torch.optim as optim

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd
from bagua_amd import env
from bagua_amd.models import MnistNet
from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry


def train(args, model, optimizer, device, epoch):
    model_inner = model
    torch.manual_seed(1000 * epoch + env.get_rank())
    for batch_idx in range(args.batches_per_epoch):
        data = torch.randn(args.batch_size, 1, 28, 28, device=device)
        target = torch.randint(0, 10, (args.batch_size,), device=device)
        optimizer.zero_grad()
        loss = F.nll_loss(model_inner(data), target)
        loss.backward()
        optimizer.step()
        if batch_idx % args.log_interval == 0 and env.get_rank() == 0:
            print("epoch %d [%d/%d] loss %.6f"
                  % (epoch, batch_idx, args.batches_per_epoch,
                     loss.item()))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--batch-size", type=int, default=64)
    parser.add_argument("--epochs", type=int, default=2)
    parser.add_argument("--batches-per-epoch", type=int, default=100)
    parser.add_argument("--lr", type=float, default=0.01)
    parser.add_argument("--log-interval", type=int, default=20)
    parser.add_argument("--algorithm", type=str,
                        default="gradient_allreduce",
                        choices=GlobalAlgorithmRegistry.names())
    parser.add_argument("--fuse-optimizer", action="store_true")
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()

    torch.manual_seed(13)
    model = MnistNet().to(device)
    if args.algorithm == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        optimizer = QAdamOptimizer(model.parameters(), lr=args.lr,
                                   warmup_steps=50)
        algorithm = GlobalAlgorithmRegistry.get("qadam")(optimizer)
    else:
        optimizer = optim.SGD(model.parameters(), lr=args.lr)
        algorithm = GlobalAlgorithmRegistry.get(args.algorithm)()
    if args.fuse_optimizer:
        from bagua_amd.contrib import fuse_optimizer

        optimizer = fuse_optimizer(optimizer)

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)

    for epoch in range(1, args.epochs + 1):
        train(args, ddp, optimizer, device, epoch)

    if args.algorithm == "async":
        ddp.inner.bagua_algorithm.abort(ddp)
    if env.get_rank() == 0:
        print("done")


if __name__ == "__main__":
    main()
