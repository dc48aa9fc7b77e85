#!/usr/bin/env python3
"""ImageNet training example (reference: examples/imagenet/main.py,
re-designed for MI355X rather than translated).

Differences from the reference's torchvision-derived script, by design:

* pure-bf16 training by default on GPU (bf16 weights + fp32-master
  FusedSGD; MI355X's native training dtype) with NHWC (channels-last)
  layout — the MIOpen fast path;
* ``--synthetic`` (default: on when --data is absent) trains on random
  data of the ImageNet shape, since this environment has no dataset
  downloads; point ``--data`` at an ImageFolder tree for real runs;
* any of the six bagua algorithms via ``--algorithm``.

Launch (one process per GPU):
    python -m bagua_amd.distributed.run --nproc-per-node 8 \
        examples/imagenet/main.py --arch resnet50 --algorithm bytegrad
"""

import argparse
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "..", "..")))

import bagua_amd  # noqa: E402
from bagua_amd import env  # noqa: E402
from bagua_amd.models import create_model  # noqa: E402
from bagua_amd.parallel.algorithms import GlobalAlgorithmRegistry  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--arch", default="resnet50",
                   choices=["resnet50", "vgg16"])
    p.add_argument("--algorithm", default="gradient_allreduce")
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--batch-size", type=int, default=64,
                   help="per-GPU batch size")
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--data", default=None,
                   help="ImageFolder root (train/ + val/); synthetic "
                        "data when absent")
    p.add_argument("--batches-per-epoch", type=int, default=50,
                   help="synthetic mode only")
    p.add_argument("--no-pure-bf16", action="store_true")
    p.add_argument("--print-freq", type=int, default=10)
    return p.parse_args()


def make_loader(args, device):
    if args.data is None:
        def synthetic():
            torch.manual_seed(1234 + env.get_rank())
            for _ in range(args.batches_per_epoch):
                yield (torch.randn(args.batch_size, 3, 224, 224,
                                   device=device),
                       torch.randint(0, 1000, (args.batch_size,),
                                     device=device))

        return synthetic, args.batches_per_epoch
    import torchvision.datasets as datasets
    import torchvision.transforms as transforms
    from torch.utils.data import DataLoader
    from torch.utils.data.distributed import DistributedSampler

    ds = datasets.ImageFolder(
        os.path.join(args.data, "train"),
        transforms.Compose([
            transforms.RandomResizedCrop(224),
            transforms.RandomHorizontalFlip(),
            transforms.ToTensor(),
            transforms.Normalize((0.485, 0.456, 0.406),
                                 (0.229, 0.224, 0.225)),
        ]))
    sampler = DistributedSampler(ds)
    loader = DataLoader(ds, batch_size=args.batch_size, sampler=sampler,
                        num_workers=8, pin_memory=True, drop_last=True)

    def real():
        for data, target in loader:
            yield data.to(device, non_blocking=True), \
                target.to(device, non_blocking=True)

    return real, len(loader)


def main():
    args = parse_args()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", env.get_local_rank()) if use_cuda \
        else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True
    bagua_amd.init_process_group()

    torch.manual_seed(42)
    model = create_model(args.arch).to(device)
    pure_bf16 = (use_cuda and not args.no_pure_bf16
                 and args.arch == "vgg16")  # BN-free archs only
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    if pure_bf16:
        model = model.to(torch.bfloat16)
        from bagua_amd.contrib import FusedSGD

        optimizer = FusedSGD(model.parameters(), lr=args.lr,
                             momentum=args.momentum,
                             weight_decay=args.weight_decay)
    else:
        optimizer = torch.optim.SGD(model.parameters(), lr=args.lr,
                                    momentum=args.momentum,
                                    weight_decay=args.weight_decay)

    algorithm = GlobalAlgorithmRegistry.get(args.algorithm)()
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer], algorithm=algorithm)

    amp = torch.autocast(device.type, dtype=torch.bfloat16,
                         enabled=use_cuda and not pure_bf16)
    loader_fn, nbatches = make_loader(args, device)
    for epoch in range(args.epochs):
        t0 = time.time()
        seen = 0
        for i, (data, target) in enumerate(loader_fn()):
            if use_cuda:
                data = data.to(memory_format=torch.channels_last)
            if pure_bf16:
                data = data.to(torch.bfloat16)
            optimizer.zero_grad()
            with amp:
                loss = F.cross_entropy(ddp(data), target)
            loss.backward()
            optimizer.step()
            seen += data.size(0)
            if env.get_rank() == 0 and i % args.print_freq == 0:
                print("epoch %d [%d/%d] loss %.4f" %
                      (epoch, i, nbatches, loss.item()), flush=True)
        if use_cuda:
            torch.cuda.synchronize()
        dt = time.time() - t0
        if env.get_rank() == 0:
            world = env.get_world_size()
            print("epoch %d done: %.1f img/s (whole job)"
                  % (epoch, seen * world / dt), flush=True)

    if args.algorithm == "async":
        ddp.inner.bagua_algorithm.abort(ddp)
    bagua_amd.deinit_process_group()


if __name__ == "__main__":
    main()
