#!/usr/bin/env python3
"""ImageNet-scale ResNet training with distributed K-FAC.

Reference analog: examples/pytorch_imagenet_resnet.py -- headline config
ResNet-50 bs 32/GPU, eigen_dp, damping 0.002, update freqs 1
(train_imagenet.sh:4-23).  Adds gradient accumulation
(--batches-per-allreduce, reference :62-65), checkpoint resume and the
KFACParamScheduler wiring.  Synthetic ImageNet-shaped data (no network).
"""

import argparse
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples import common  # noqa: E402


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--batches-per-allreduce", type=int, default=1)
    common.add_common_args(p)
    p.set_defaults(damping=0.002, base_lr=0.0125, lr_decay=[25, 35, 40],
                   warmup_epochs=5, batch_size=32)
    args = p.parse_args()

    rank, world, local_rank, use_cuda = common.initialize_distributed()
    torch.manual_seed(args.seed)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    from kfac_pytorch_amd.models import get_imagenet_model
    model = get_imagenet_model(args.model,
                               num_classes=args.num_classes).to(device)
    ddp_model = model
    if world > 1:
        ddp_model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None)

    scaled_lr = args.base_lr * world * args.batches_per_allreduce
    optimizer = torch.optim.SGD(model.parameters(), lr=scaled_lr,
                                momentum=args.momentum,
                                weight_decay=args.weight_decay)
    precond, kfac_sched = common.build_kfac(model, args, world)

    from kfac_pytorch_amd.utils import create_lr_schedule
    lr_fn = create_lr_schedule(world, args.warmup_epochs, args.lr_decay)
    lrs = [torch.optim.lr_scheduler.LambdaLR(optimizer, lr_fn)]
    if precond is not None:
        lrs += [torch.optim.lr_scheduler.LambdaLR(precond, lr_fn),
                kfac_sched]

    start_epoch = 0
    if args.resume_from and os.path.exists(args.resume_from):
        from kfac_pytorch_amd.utils import load_checkpoint
        start_epoch = load_checkpoint(model, optimizer, args.resume_from)
        if kfac_sched is not None:
            kfac_sched.step(start_epoch)

    autocast = args.dtype == "bf16" and use_cuda
    g = torch.Generator().manual_seed(args.seed + rank)

    for epoch in range(start_epoch, args.epochs):
        model.train()
        iter_times = []
        import time
        last = time.perf_counter()
        for it in range(args.iters_per_epoch):
            optimizer.zero_grad(set_to_none=False)
            # gradient accumulation over sub-batches (reference :355-367)
            for sub in range(args.batches_per_allreduce):
                x = torch.randn(args.batch_size, 3, args.image_size,
                                args.image_size, generator=g).to(device)
                y = torch.randint(0, args.num_classes, (args.batch_size,),
                                  generator=g).to(device)
                sync_now = sub == args.batches_per_allreduce - 1
                ctx = (ddp_model.no_sync()
                       if (world > 1 and not sync_now)
                       else torch.enable_grad())
                with ctx:
                    if autocast:
                        with torch.autocast("cuda", dtype=torch.bfloat16):
                            loss = F.cross_entropy(ddp_model(x), y)
                    else:
                        loss = F.cross_entropy(ddp_model(x), y)
                    (loss / args.batches_per_allreduce).backward()
            if precond is not None:
                precond.step()
            optimizer.step()
            now = time.perf_counter()
            iter_times.append(now - last)
            last = now
            if rank == 0 and (it + 1) % args.display == 0:
                common.logger.info("epoch %d iter %d/%d %.3fs/iter",
                                   epoch, it + 1, args.iters_per_epoch,
                                   iter_times[-1])
        for s in lrs:
            s.step()
        if args.checkpoint_format:
            from kfac_pytorch_amd.utils import save_checkpoint
            save_checkpoint(model, optimizer, args.checkpoint_format,
                            epoch, preconditioner=precond)
        if args.speed:
            common.report_speed(
                iter_times,
                args.batch_size * args.batches_per_allreduce, world, rank)


if __name__ == "__main__":
    main()
