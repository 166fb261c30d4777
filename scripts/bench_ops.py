#!/usr/bin/env python3
"""Op-level microbenchmarks for the K-FAC hot ops on MI355X.

Reference analog: scripts/bench_ops.py -- symeig/GEMM over the factor
sizes K-FAC actually produces, plus correctness checks of the HIP
kernels against torch.  Run on a GPU box:

    python scripts/bench_ops.py [--op all|eig|syrk|precond|im2col]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# ResNet-50 K-FAC factor dims (reference: scripts/inverse_model.py:19-20)
RESNET50_A_DIMS = [27, 64, 65, 128, 129, 147, 256, 257, 512, 513, 576,
                   1024, 1025, 1152, 2048, 2049, 2304, 4608]
RESNET50_G_DIMS = [64, 128, 256, 512, 1000, 1024, 2048]


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_eig():
    from kfac_pytorch_amd.ops.linalg import mat_eig, mat_eig_multi
    from kfac_pytorch_amd.ops import _ext
    print("== symmetric eigensolve ==")
    for m in RESNET50_A_DIMS:
        a = torch.randn(m, m, device="cuda")
        a = a @ a.t() / m + 0.1 * torch.eye(m, device="cuda")
        t_eigh = timeit(lambda: torch.linalg.eigh(a))
        line = f"m={m:5d}  eigh {t_eigh * 1e3:8.2f} ms"
        if _ext.has_jacobi_eigh(m):
            t_jac = timeit(lambda: _ext.jacobi_eigh_batched([a]))
            line += f"  jacobi {t_jac * 1e3:8.2f} ms"
        print(line)
    # batched small dims (the real K-FAC pattern: many at once)
    small = [d for d in RESNET50_A_DIMS + RESNET50_G_DIMS if d <= 128]
    mats = []
    for m in small * 3:
        a = torch.randn(m, m, device="cuda")
        mats.append(a @ a.t() / m + 0.1 * torch.eye(m, device="cuda"))
    t_serial = timeit(lambda: [torch.linalg.eigh(a) for a in mats])
    t_batch = timeit(lambda: _ext.jacobi_eigh_batched(mats))
    print(f"batch of {len(mats)} small factors: serial eigh "
          f"{t_serial * 1e3:.2f} ms vs batched jacobi "
          f"{t_batch * 1e3:.2f} ms")


def bench_syrk():
    from kfac_pytorch_amd.ops import _ext
    print("== factor SYRK (bf16 MFMA) vs torch mm (fp32/bf16) ==")
    shapes = [(401408, 27), (200704, 64), (50176, 576), (12544, 1152),
              (12544, 2304), (3136, 4608), (32, 2048)]
    for rows, d in shapes:
        x32 = torch.randn(rows, d, device="cuda")
        xb = x32.bfloat16()
        out = torch.empty(d + 1, d + 1, device="cuda")
        t_hip = timeit(lambda: _ext.syrk_factor_(xb, out, 1.0,
                                                 float(rows), True, -1.0))
        t_mm32 = timeit(lambda: x32.t() @ x32)
        t_mmbf = timeit(lambda: xb.t() @ xb)
        flops = 2.0 * rows * d * d
        print(f"rows={rows:7d} d={d:5d}  hip {t_hip * 1e3:7.2f} ms "
              f"({flops / t_hip / 1e12:6.1f} TF)  mm.fp32 "
              f"{t_mm32 * 1e3:7.2f} ms  mm.bf16 {t_mmbf * 1e3:7.2f} ms")


def bench_precond():
    from kfac_pytorch_amd.ops.linalg import eigen_precondition
    print("== implicit-eigen preconditioning chain ==")
    for (dg, da) in [(64, 577), (256, 2305), (512, 4608), (1000, 2049)]:
        QA = torch.randn(da, da, device="cuda")
        QG = torch.randn(dg, dg, device="cuda")
        dA = torch.rand(da, device="cuda")
        dG = torch.rand(dg, device="cuda")
        grad = torch.randn(dg, da, device="cuda")
        t = timeit(lambda: eigen_precondition(QA, dA, QG, dG, grad, 0.002))
        print(f"[{dg:5d} x {da:5d}] {t * 1e3:7.3f} ms")


def bench_im2col():
    from kfac_pytorch_amd.ops import _ext
    import torch.nn.functional as F
    print("== im2col (reference: scripts/bench_extract_patches.py) ==")
    shapes = [((32, 64, 56, 56), 3, 1, 1), ((32, 256, 56, 56), 1, 1, 0),
              ((32, 512, 28, 28), 3, 2, 1), ((32, 3, 224, 224), 7, 2, 3)]
    for shape, k, s, p in shapes:
        x = torch.randn(*shape, device="cuda").bfloat16()
        t_hip = timeit(lambda: _ext.im2col(x, k, k, s, s, p, p, 1, 1))
        t_unf = timeit(lambda: F.unfold(x, k, stride=s, padding=p))
        print(f"{str(shape):22s} k={k} s={s}: hip {t_hip * 1e3:7.2f} ms "
              f"unfold {t_unf * 1e3:7.2f} ms")


def bench_sbr():
    """SBR stage-1 band reduction vs whole-solve library time at the
    big buckets (the round-3 eigensolver path; stage-2 chase numbers
    need the HIP kernel -- the torch chase is a correctness oracle,
    not a perf path).  NOT in --op all: opt in explicitly."""
    from kfac_pytorch_amd.ops.sbr import band_reduce_batched
    print("== SBR stage 1 (full->band 64) vs library whole solve ==")
    for n, bsz in [(1152, 4), (2304, 4), (4608, 3)]:
        g = torch.Generator().manual_seed(n)
        x = torch.randn(bsz, n, 2 * n, generator=g)
        A = (x @ x.mT / (2 * n)).cuda()
        t1 = timeit(lambda: band_reduce_batched(A, 64), warmup=2,
                    iters=3)
        t2 = timeit(lambda: torch.linalg.eigh(A), warmup=1, iters=2)
        print(f"n={n:5d}x{bsz}  stage1 {t1 * 1e3:8.1f} ms   "
              f"eigh(whole) {t2 * 1e3:8.1f} ms")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--op", default="all",
                   choices=["all", "eig", "syrk", "precond", "im2col",
                            "sbr"])
    args = p.parse_args()
    assert torch.cuda.is_available(), "bench_ops needs a GPU"
    if args.op in ("all", "eig"):
        bench_eig()
    if args.op in ("all", "syrk"):
        bench_syrk()
    if args.op in ("all", "precond"):
        bench_precond()
    if args.op in ("all", "im2col"):
        bench_im2col()
    if args.op == "sbr":
        bench_sbr()


if __name__ == "__main__":
    main()
