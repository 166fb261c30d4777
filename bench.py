#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 distributed K-FAC training step.

Measures the BASELINE.json metric -- images/sec for ResNet-50 K-FAC
(eigen_dp, damping 0.002, factor/inverse update freq 1, batch 32/GPU --
the reference's headline ImageNet config, train_imagenet.sh:4-23 /
batch.sh:27-29) on synthetic ImageNet-shape data with random-init
weights, bf16 forward/backward (fp32 factors/eigensolves), one process
per GPU over RCCL.

Usage:
    python bench.py [--gpus N] [--steps K] [--warmup W]
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate images/sec.
"""

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

# reference per-iteration budget measured on the authors' cluster
# (scripts/time_breakdown.py:24-27): K-FAC 1 GPU 0.487 s/iter at bs 32;
# distributed MPD K-FAC 0.882 s/iter at bs 32/GPU.
BASELINE_ITER_1GPU = 0.487
BASELINE_ITER_DIST = 0.882


def parse_args():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=32,
                   help="per-GPU batch size (headline config: 32)")
    p.add_argument("--model", default="resnet50")
    p.add_argument("--kfac-name", default="eigen_dp",
                   choices=["inverse", "eigen", "inverse_dp", "eigen_dp",
                            "none"])
    p.add_argument("--damping", type=float, default=0.002)
    p.add_argument("--fac-update-freq", type=int, default=1)
    p.add_argument("--kfac-update-freq", type=int, default=1)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--image-size", type=int, default=224)
    return p.parse_args()


def init_dist(args):
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        rank = int(os.environ["RANK"])
        world = int(os.environ["WORLD_SIZE"])
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
    else:
        rank, world, local_rank = 0, 1, 0
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29771")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("LOCAL_RANK", "0")
    use_cuda = torch.cuda.is_available()
    backend = "nccl" if use_cuda else "gloo"
    if use_cuda:
        torch.cuda.set_device(local_rank)
        # MIOpen find-mode autotuning for the conv fw/bw (the
        # reference sets cudnn.benchmark=True,
        # examples/pytorch_cifar10_resnet.py:134); KFAC_CONV_BENCHMARK=0
        # disables
        torch.backends.cudnn.benchmark = (
            os.environ.get("KFAC_CONV_BENCHMARK", "1") != "0")
    dist.init_process_group(backend=backend, init_method="env://",
                            world_size=world, rank=rank)
    return rank, world, local_rank, use_cuda


def main():
    args = parse_args()
    rank, world, local_rank, use_cuda = init_dist(args)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.backend as backend
    from kfac_pytorch_amd.models import get_imagenet_model
    backend.init("Torch")

    torch.manual_seed(42)
    model = get_imagenet_model(args.model).to(device)
    if world > 1:
        ddp_model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None,
            bucket_cap_mb=64)
    else:
        ddp_model = model

    optimizer = torch.optim.SGD(model.parameters(), lr=0.0125 * world,
                                momentum=0.9, weight_decay=5e-5)
    precond = None
    if args.kfac_name != "none":
        KFAC = kfac.get_kfac_module(args.kfac_name)
        precond = KFAC(model, lr=0.0125 * world, damping=args.damping,
                       fac_update_freq=args.fac_update_freq,
                       kfac_update_freq=args.kfac_update_freq)

    bs = args.batch_size
    autocast_dtype = torch.bfloat16 if args.dtype == "bf16" else None
    data = torch.randn(bs, 3, args.image_size, args.image_size,
                       device=device)
    target = torch.randint(0, 1000, (bs,), device=device)

    kfac_time = 0.0

    def one_step():
        nonlocal kfac_time
        optimizer.zero_grad(set_to_none=False)
        if autocast_dtype is not None and use_cuda:
            with torch.autocast("cuda", dtype=autocast_dtype):
                out = ddp_model(data)
                loss = F.cross_entropy(out, target)
        else:
            out = ddp_model(data)
            loss = F.cross_entropy(out, target)
        loss.backward()
        if precond is not None:
            t0 = time.perf_counter()
            precond.step()
            if use_cuda:
                torch.cuda.synchronize()
            kfac_time += time.perf_counter() - t0
        optimizer.step()

    # warmup (untimed)
    for _ in range(args.warmup):
        one_step()

    # timed region: barrier + sync on both sides, max over ranks
    if use_cuda:
        torch.cuda.synchronize()
    dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    kfac_time = 0.0
    t_start = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if use_cuda:
        torch.cuda.synchronize()
    dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start

    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if use_cuda else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    images_per_sec = bs * world * args.steps / elapsed
    baseline_iter = BASELINE_ITER_1GPU if world == 1 else BASELINE_ITER_DIST
    baseline_imgs = 32.0 * world / baseline_iter
    vs_baseline = (images_per_sec / baseline_imgs
                   if args.kfac_name != "none" and bs == 32 else None)

    if rank == 0 and precond is not None and \
            hasattr(precond, "phase_times"):
        # *_total entries are cumulative counts, the rest are seconds
        per_step = {k: (round(v, 0) if k.endswith("_total") else
                        round(v / (args.steps + args.warmup) * 1000.0, 2))
                    for k, v in precond.phase_times.items()}
        print("KFAC_PHASES(ms/step):", json.dumps(per_step),
              file=sys.stderr, flush=True)

    if rank == 0:
        result = {
            "metric": "images/sec ResNet-50 K-FAC",
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(vs_baseline, 3) if vs_baseline else None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": bs * world,
                "seq_len": None,
                "parallelism": f"dp{world}",
                "kfac": args.kfac_name,
                "damping": args.damping,
                "fac_update_freq": args.fac_update_freq,
                "kfac_update_freq": args.kfac_update_freq,
                "image_size": args.image_size,
                "precondition_ms_per_step": round(
                    kfac_time / args.steps * 1000.0, 2),
            },
        }
        print(json.dumps(result), flush=True)

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
