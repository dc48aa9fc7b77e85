#!/usr/bin/env python3
"""MoE training + MoE-aware checkpoint example
(reference: examples/moe/, .buildkite/scripts/benchmark_master.sh:114-160).

Trains an MNIST-shaped ConvNet with an MoE FFN block (num-local-experts
per rank, expert-parallel over all ranks), saves an expert-sharded
checkpoint and reloads it.

Launch:
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/moe/main.py --num-local-experts 2
"""

import argparse
import tempfile

import torch
import torch.nn as nn
import torch.nn.functional as F

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd
from bagua_amd import env
from bagua_amd.checkpoint import load_checkpoint, save_checkpoint
from bagua_amd.parallel.algorithms.gradient_allreduce import (
    GradientAllReduceAlgorithm,
)
from bagua_amd.parallel.moe import MoE


class MoEMnistNet(nn.Module):
    def __init__(self, num_local_experts: int):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, 3, 1)
        self.conv2 = nn.Conv2d(32, 64, 3, 1)
        self.fc1 = nn.Linear(9216, 128)
        self.moe = MoE(
            hidden_size=128,
            expert=nn.Sequential(nn.Linear(128, 512), nn.ReLU(),
                                 nn.Linear(512, 128)),
            num_local_experts=num_local_experts, k=1)
        self.fc2 = nn.Linear(128, 10)

    def forward(self, x):
        x = F.relu(self.conv1(x))
        x = F.max_pool2d(F.relu(self.conv2(x)), 2).flatten(1)
        x = F.relu(self.fc1(x))
        x, l_aux, _ = self.moe(x)
        return F.log_softmax(self.fc2(x), dim=1), l_aux


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-local-experts", type=int, default=2)
    parser.add_argument("--steps", type=int, default=50)
    parser.add_argument("--batch-size", type=int, default=64)
    parser.add_argument("--ckpt-dir", type=str, default="")
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(env.get_local_rank())
    bagua_amd.init_process_group()

    torch.manual_seed(13)
    model = MoEMnistNet(args.num_local_experts).to(device)
    optimizer = torch.optim.Adam(model.parameters(), lr=1e-3)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())

    loss = None
    for step in range(args.steps):
        torch.manual_seed(7000 + env.get_rank() * 31 + step)
        data = torch.randn(args.batch_size, 1, 28, 28, device=device)
        target = torch.randint(0, 10, (args.batch_size,), device=device)
        optimizer.zero_grad()
        out, l_aux = ddp(data)
        loss = F.nll_loss(out, target) + 0.01 * l_aux
        loss.backward()
        optimizer.step()
        if step % 10 == 0 and env.get_rank() == 0:
            print("step %d loss %.6f" % (step, loss.item()))

    ckpt_dir = args.ckpt_dir or tempfile.mkdtemp(prefix="bagua_moe_ckpt_")
    save_checkpoint(args.steps, ckpt_dir, model, optimizer)
    it = load_checkpoint(ckpt_dir, model, optimizer)
    if env.get_rank() == 0:
        print("checkpoint saved+restored at iteration %d in %s"
              % (it, ckpt_dir))
        print("Final loss: %.6f" % loss.item())


if __name__ == "__main__":
    main()
