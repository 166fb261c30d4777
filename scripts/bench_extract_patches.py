#!/usr/bin/env python3
"""Benchmark im2col patch extraction over real ResNet-50 conv shapes.

Reference analog: scripts/bench_extract_patches.py (times
``_extract_patches`` over conv shapes replayed from a training log).
Here the shape list is inlined (the same ResNet-50 layer shapes at
batch 32) and both the HIP im2col kernel and the torch unfold oracle
are timed; on CPU only the oracle runs.

    python scripts/bench_extract_patches.py [--batch 32]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# (in_c, H, W, kh, stride, pad) per distinct ResNet-50 conv at 224x224
RESNET50_CONVS = [
    (3, 224, 224, 7, 2, 3),
    (64, 56, 56, 1, 1, 0), (64, 56, 56, 3, 1, 1), (256, 56, 56, 1, 1, 0),
    (128, 56, 56, 3, 2, 1), (256, 56, 56, 1, 2, 0),
    (128, 28, 28, 3, 1, 1), (512, 28, 28, 1, 1, 0),
    (256, 28, 28, 3, 2, 1), (512, 28, 28, 1, 2, 0),
    (256, 14, 14, 3, 1, 1), (1024, 14, 14, 1, 1, 0),
    (512, 14, 14, 3, 2, 1), (1024, 14, 14, 1, 2, 0),
    (512, 7, 7, 3, 1, 1), (2048, 7, 7, 1, 1, 0),
]


def timeit(fn, warmup=3, iters=10, cuda=False):
    for _ in range(warmup):
        fn()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if cuda:
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=32)
    args = ap.parse_args()
    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"

    from kfac_pytorch_amd.ops.factors import extract_patches
    import torch.nn.functional as F

    total_hip = total_ref = 0.0
    print(f"{'shape':>28} {'rows x cols':>16} {'hip_ms':>8} {'unfold_ms':>10}")
    for (c, h, w, k, s, p) in RESNET50_CONVS:
        x = torch.randn(args.batch, c, h, w, device=device)
        t_hip = float("nan")
        if use_cuda:
            t_hip = timeit(lambda: extract_patches(
                x, (k, k), (s, s), (p, p)), cuda=True) * 1e3
            total_hip += t_hip

        def oracle():
            cols = F.unfold(x, kernel_size=(k, k), stride=(s, s),
                            padding=(p, p))
            return cols.transpose(1, 2).reshape(-1, cols.size(1))

        t_ref = timeit(oracle, cuda=use_cuda) * 1e3
        total_ref += t_ref
        out_h = (h + 2 * p - k) // s + 1
        print(f"{str((c, h, w, k, s, p)):>28} "
              f"{args.batch * out_h * out_h:>9}x{c * k * k:<6} "
              f"{t_hip:8.3f} {t_ref:10.3f}")
    print(f"\ntotal: hip={total_hip:.2f} ms  unfold={total_ref:.2f} ms")


if __name__ == "__main__":
    main()
