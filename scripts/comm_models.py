#!/usr/bin/env python3
"""Fit alpha-beta (latency + per-byte) communication cost models from
measured collective timings, and measure them on the current system.

Reference analog: scripts/comm_models.py (fits NCCL broadcast/allreduce
logs).  Here the measurement pass runs live over torch.distributed
(RCCL over xGMI on a GPU node; gloo on CPU) and fits t = alpha + beta*n.

Launch:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 scripts/comm_models.py
"""

import argparse
import os
import sys
import time

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def fit_alpha_beta(sizes, times):
    """Least-squares fit t = alpha + beta * size."""
    A = np.vstack([np.ones(len(sizes)), np.array(sizes)]).T
    (alpha, beta), *_ = np.linalg.lstsq(A, np.array(times), rcond=None)
    return alpha, beta


def measure(op, sizes, device, iters=20, warmup=5):
    times = []
    for n in sizes:
        t = torch.randn(n, device=device)
        for _ in range(warmup):
            op(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            op(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        times.append((time.perf_counter() - t0) / iters)
    return times


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--max-log2", type=int, default=24)
    args = p.parse_args()

    if "RANK" not in os.environ:
        os.environ.update(RANK="0", WORLD_SIZE="1", LOCAL_RANK="0",
                          MASTER_ADDR="127.0.0.1", MASTER_PORT="29773")
    use_cuda = torch.cuda.is_available()
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if use_cuda:
        torch.cuda.set_device(local_rank)
    dist.init_process_group("nccl" if use_cuda else "gloo",
                            init_method="env://")
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    sizes = [2 ** k for k in range(10, args.max_log2, 2)]
    t_ar = measure(lambda t: dist.all_reduce(t), sizes, device)
    t_bc = measure(lambda t: dist.broadcast(t, src=0), sizes, device)

    if rank == 0:
        for name, times in [("allreduce", t_ar), ("broadcast", t_bc)]:
            alpha, beta = fit_alpha_beta([s * 4 for s in sizes], times)
            print(f"{name}: alpha={alpha * 1e6:.1f} us, "
                  f"beta={beta * 1e9:.3f} ns/B "
                  f"({1.0 / beta / 1e9:.1f} GB/s)")
            for s, t in zip(sizes, times):
                print(f"  {s * 4:>12d} B  {t * 1e6:10.1f} us")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
