#!/usr/bin/env python3
"""Measure SBR stage-1 band reduction (ops/sbr.py) on GPU at the
flagship eigensolve bucket (4608 x 3, the ResNet-50 factor that binds
the freq-1 eigen step -- docs/SYTRD_DESIGN.md floor study), plus a
GPU correctness residual at 1152.  One short run; evidence lands in
profiles/ for the round-3 budget table in docs/SBR_STAGE2_NOTES.md."""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from kfac_pytorch_amd.ops.sbr import (apply_q_batched,  # noqa: E402
                                      band_reduce_batched)


def spd(N, n, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(N, n, 2 * n, generator=g)
    return (x @ x.mT / (2 * n)).cuda()


def main():
    dev = torch.cuda.get_device_name(0)
    print(f"device: {dev}")

    # correctness on GPU fp32 at 1152 (a real ResNet-50 bucket dim)
    A = spd(2, 1152, seed=1)
    B, panels = band_reduce_batched(A, 64)
    eye = torch.eye(1152, device="cuda").expand(2, -1, -1).contiguous()
    Q = apply_q_batched(panels, eye)
    orth = float((Q @ Q.mT - eye).abs().max())
    resid = float(((Q @ B @ Q.mT - A).norm() / A.norm()))
    band = float(B.triu(65).abs().max())
    print(f"n=1152x2 b=64 fp32: orth={orth:.2e} resid={resid:.2e} "
          f"band={band:.2e}")

    # timing at the flagship bucket
    for n, N, b in [(1152, 4, 64), (2304, 4, 64), (4608, 3, 64)]:
        A = spd(N, n, seed=n)
        for _ in range(2):
            band_reduce_batched(A, b)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 3
        for _ in range(iters):
            band_reduce_batched(A, b)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / iters * 1e3
        print(f"stage1 band_reduce n={n}x{N} b={b}: {ms:.1f} ms")


if __name__ == "__main__":
    main()
