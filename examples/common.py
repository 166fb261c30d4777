"""Shared harness for the example trainers.

Mirrors the reference examples' structure (reference:
examples/pytorch_cifar10_resnet.py): distributed init, SPEED vs
convergence mode, per-phase timers (IO / FW+BW / COMM / KFAC / UPDATE),
K-FAC + SGD wiring and rank-0 logging -- MI355X-first: one process per
GPU over RCCL, DDP for gradient averaging, bf16 autocast compute.

There is no network in this environment, so the data loaders serve
synthetic batches of the real datasets' shapes by default; a local
dataset file can be supplied where available.
"""

from __future__ import annotations

import argparse
import logging
import os
import sys
import time

import torch
import torch.distributed as dist

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

logger = logging.getLogger("kfac_examples")


def add_common_args(p: argparse.ArgumentParser):
    g = p.add_argument_group("training")
    g.add_argument("--epochs", type=int, default=1)
    g.add_argument("--iters-per-epoch", type=int, default=60,
                   help="synthetic iterations per epoch")
    g.add_argument("--batch-size", type=int, default=32)
    g.add_argument("--base-lr", type=float, default=0.1)
    g.add_argument("--momentum", type=float, default=0.9)
    g.add_argument("--weight-decay", type=float, default=5e-4)
    g.add_argument("--warmup-epochs", type=float, default=5)
    g.add_argument("--lr-decay", type=int, nargs="*", default=[35, 75, 90])
    g.add_argument("--checkpoint-format", default=None,
                   help="e.g. ckpt-{epoch}.pth.tar (rank-0 save per epoch)")
    g.add_argument("--resume-from", default=None)
    g.add_argument("--speed", action="store_true",
                   help="throughput mode: timed iterations, no eval")
    g.add_argument("--seed", type=int, default=42)

    k = p.add_argument_group("kfac")
    k.add_argument("--kfac-name", default="eigen_dp",
                   choices=["inverse", "eigen", "inverse_dp", "eigen_dp",
                            "none"])
    k.add_argument("--damping", type=float, default=0.003)
    k.add_argument("--kfac-update-freq", type=int, default=1)
    k.add_argument("--fac-update-freq", type=int, default=1)
    k.add_argument("--kl-clip", type=float, default=0.001)
    k.add_argument("--factor-decay", type=float, default=0.95)
    k.add_argument("--exclude-parts", default="")
    k.add_argument("--damping-alpha", type=float, default=1.0)
    k.add_argument("--damping-schedule", type=int, nargs="*", default=None)
    k.add_argument("--kfac-update-freq-alpha", type=float, default=1.0)
    k.add_argument("--kfac-update-freq-schedule", type=int, nargs="*",
                   default=None)

    d = p.add_argument_group("system")
    d.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    d.add_argument("--display", type=int, default=20)
    return p


def initialize_distributed():
    """One process per GPU: torch.distributed over RCCL (GPU) / gloo (CPU)
    (reference init: examples/pytorch_cifar10_resnet.py:116-134)."""
    if "RANK" not in os.environ:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29772")
        os.environ["RANK"] = "0"
        os.environ["WORLD_SIZE"] = "1"
        os.environ["LOCAL_RANK"] = "0"
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
        # MIOpen find-mode conv autotuning (reference trainers set
        # cudnn.benchmark=True, examples/pytorch_cifar10_resnet.py:134)
        torch.backends.cudnn.benchmark = (
            os.environ.get("KFAC_CONV_BENCHMARK", "1") != "0")
    dist.init_process_group("nccl" if use_cuda else "gloo",
                            init_method="env://", rank=rank,
                            world_size=world)
    import kfac_pytorch_amd.backend as backend
    backend.init("Torch")
    logging.basicConfig(
        level=logging.INFO if rank == 0 else logging.WARNING,
        format="%(asctime)s %(message)s")
    return rank, world, local_rank, use_cuda


def build_kfac(model, args, world: int):
    import kfac_pytorch_amd as kfac
    if args.kfac_name == "none" or args.kfac_update_freq == 0:
        return None, None
    KFAC = kfac.get_kfac_module(args.kfac_name)
    pre = KFAC(model, lr=args.base_lr * world, damping=args.damping,
               fac_update_freq=args.fac_update_freq,
               kfac_update_freq=args.kfac_update_freq,
               kl_clip=args.kl_clip, factor_decay=args.factor_decay,
               exclude_parts=args.exclude_parts)
    sched = kfac.KFACParamScheduler(
        pre, damping_alpha=args.damping_alpha,
        damping_schedule=args.damping_schedule,
        update_freq_alpha=args.kfac_update_freq_alpha,
        update_freq_schedule=args.kfac_update_freq_schedule)
    return pre, sched


def train_loop(model, ddp_model, optimizer, preconditioner, criterion,
               batches, args, rank: int, use_cuda: bool,
               lr_schedulers=()):
    """One epoch over `batches` (iterable of (data, target)), with the
    reference's phase-timer breakdown in SPEED mode."""
    from kfac_pytorch_amd.utils import Metric, PhaseTimers
    model.train()
    timers = PhaseTimers(cuda=use_cuda)
    loss_metric = Metric("train_loss")
    autocast = (args.dtype == "bf16" and use_cuda)
    iter_times = []
    last = time.perf_counter()

    for it, (data, target) in enumerate(batches):
        timers.start("io")
        if use_cuda:
            data = data.cuda(non_blocking=True)
            target = target.cuda(non_blocking=True)
        timers.stop()

        timers.start("fwbw")
        optimizer.zero_grad(set_to_none=False)
        if autocast:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = criterion(ddp_model(data), target)
        else:
            loss = criterion(ddp_model(data), target)
        loss.backward()  # DDP allreduce overlaps here (COMM phase)
        timers.stop()

        if preconditioner is not None:
            timers.start("kfac")
            preconditioner.step()
            timers.stop()

        timers.start("update")
        optimizer.step()
        timers.stop()

        now = time.perf_counter()
        iter_times.append(now - last)
        last = now
        if not args.speed:
            loss_metric.update(loss.detach())
        if rank == 0 and (it + 1) % args.display == 0:
            logger.info("iter %d/%d %s", it + 1, len(batches),
                        timers.format())
            timers.reset()
    for s in lr_schedulers:
        s.step()
    return loss_metric.avg, iter_times


def report_speed(iter_times, batch_size: int, world: int, rank: int,
                 skip: int = 10):
    import numpy as np
    t = np.array(iter_times[skip:]) if len(iter_times) > skip else \
        np.array(iter_times)
    if rank == 0 and len(t):
        logger.info("iteration time: %.4f +- %.4f s, %.1f images/s "
                    "(whole job)", t.mean(), t.std(),
                    batch_size * world / t.mean())
