#!/usr/bin/env python3
"""Elastic training example (reference: examples/elastic_training/main.py).

Workers self-checkpoint every epoch; on a torchelastic restart (worker
failure or membership change) training resumes from the last checkpoint.

Launch:
    python -m bagua_amd.distributed.run --nnodes=1:1 --nproc-per-node 8 \
        --max-restarts 3 examples/elastic_training/main.py \
        --ckpt-dir /tmp/elastic_ckpt
"""

import argparse
import os

import torch
import torch.nn.functional as F

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd
from bagua_amd import env
from bagua_amd.checkpoint import load_checkpoint, save_checkpoint
from bagua_amd.models import MnistNet
from bagua_amd.parallel.algorithms.gradient_allreduce import (
    GradientAllReduceAlgorithm,
)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--ckpt-dir", type=str, default="/tmp/bagua_elastic")
    parser.add_argument("--epochs", type=int, default=5)
    parser.add_argument("--batches-per-epoch", type=int, default=50)
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()

    torch.manual_seed(13)
    model = MnistNet().to(device)
    optimizer = torch.optim.SGD(model.parameters(), lr=0.01)

    start_epoch = 0
    if os.path.isdir(args.ckpt_dir):
        start_epoch = load_checkpoint(args.ckpt_dir, model, optimizer)
        if env.get_rank() == 0 and start_epoch:
            print("resumed from epoch %d" % start_epoch)

    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    for epoch in range(start_epoch, args.epochs):
        for batch in range(args.batches_per_epoch):
            torch.manual_seed(9000 + epoch * 997 + batch * 31
                              + env.get_rank())
            data = torch.randn(64, 1, 28, 28, device=device)
            target = torch.randint(0, 10, (64,), device=device)
            optimizer.zero_grad()
            loss = F.nll_loss(ddp(data), target)
            loss.backward()
            optimizer.step()
        save_checkpoint(epoch + 1, args.ckpt_dir, model, optimizer)
        if env.get_rank() == 0:
            print("epoch %d done, loss %.6f (checkpointed)"
                  % (epoch, loss.item()))


if __name__ == "__main__":
    main()
