#!/usr/bin/env python3
"""CIFAR-10/100 ResNet/VGG/WRN training with distributed K-FAC.

Reference analog: examples/pytorch_cifar10_resnet.py -- dual
SPEED/convergence modes, per-phase timers, K-FAC param scheduler, warmup
+ step LR.  Data is synthetic CIFAR-shaped by default (no network in
this environment); pass --data-npz with 'x'/'y' arrays for real data.

Launch (one process per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_cifar.py --model resnet32
"""

import argparse
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples import common  # noqa: E402


def get_model(name, num_classes):
    from kfac_pytorch_amd.models import (get_cifar_model, vgg16, vgg19,
                                         wrn28_10, wrn28_20)
    table = {"vgg16": vgg16, "vgg19": vgg19, "wrn28-10": wrn28_10,
             "wrn28-20": wrn28_20}
    if name in table:
        return table[name](num_classes=num_classes)
    return get_cifar_model(name, num_classes=num_classes)


def synthetic_batches(args, world, rank, epoch):
    g = torch.Generator().manual_seed(args.seed + epoch * 1000 + rank)
    batches = []
    for _ in range(args.iters_per_epoch):
        x = torch.randn(args.batch_size, 3, 32, 32, generator=g)
        y = torch.randint(0, args.num_classes, (args.batch_size,),
                          generator=g)
        batches.append((x, y))
    return batches


def npz_batches(args, world, rank, epoch):
    import numpy as np
    data = np.load(args.data_npz)
    x = torch.from_numpy(data["x"]).float()
    y = torch.from_numpy(data["y"]).long()
    n = x.size(0) // world
    x, y = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    perm = torch.randperm(n, generator=torch.Generator().manual_seed(
        args.seed + epoch))
    bs = args.batch_size
    return [(x[perm[i:i + bs]], y[perm[i:i + bs]])
            for i in range(0, n - bs + 1, bs)]


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--model", default="resnet32")
    p.add_argument("--num-classes", type=int, default=10)
    p.add_argument("--data-npz", default=None)
    common.add_common_args(p)
    args = p.parse_args()

    rank, world, local_rank, use_cuda = common.initialize_distributed()
    torch.manual_seed(args.seed)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    model = get_model(args.model, args.num_classes).to(device)
    ddp_model = model
    if world > 1:
        ddp_model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None)

    optimizer = torch.optim.SGD(model.parameters(),
                                lr=args.base_lr * world,
                                momentum=args.momentum,
                                weight_decay=args.weight_decay)
    precond, kfac_sched = common.build_kfac(model, args, world)

    from kfac_pytorch_amd.utils import create_lr_schedule
    lrs = [torch.optim.lr_scheduler.LambdaLR(
        optimizer, create_lr_schedule(world, args.warmup_epochs,
                                      args.lr_decay))]
    if precond is not None:
        lrs.append(torch.optim.lr_scheduler.LambdaLR(
            precond, create_lr_schedule(world, args.warmup_epochs,
                                        args.lr_decay)))
        lrs.append(kfac_sched)

    criterion = F.cross_entropy
    make_batches = npz_batches if args.data_npz else synthetic_batches

    start_epoch = 0
    if args.resume_from and os.path.exists(args.resume_from):
        from kfac_pytorch_amd.utils import load_checkpoint
        start_epoch = load_checkpoint(model, optimizer, args.resume_from)

    for epoch in range(start_epoch, args.epochs):
        batches = make_batches(args, world, rank, epoch)
        loss, iter_times = common.train_loop(
            model, ddp_model, optimizer, precond, criterion, batches,
            args, rank, use_cuda, lr_schedulers=lrs)
        if rank == 0 and not args.speed:
            common.logger.info("epoch %d loss %.4f", epoch, loss)
        if args.checkpoint_format:
            from kfac_pytorch_amd.utils import save_checkpoint
            save_checkpoint(model, optimizer, args.checkpoint_format,
                            epoch, preconditioner=precond)
        if args.speed:
            common.report_speed(iter_times, args.batch_size, world, rank)


if __name__ == "__main__":
    main()
