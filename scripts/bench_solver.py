#!/usr/bin/env python3
"""Measure every eigensolve/inverse primitive per factor dim to place
the mat_eig_multi dispatch cutoffs.

Compares (GPU):
  * torch.linalg.eigh (host-synced rocSOLVER syevd -- the naive tier)
  * magma eigh (torch.backends.cuda.preferred_linalg_library) if built
  * _kfac_solver.syevd_pool_   (8 async pool streams, amortized)
  * _kfac_solver.syevdj_batched_ (one strided-batched Jacobi call)
  * hand-written LDS-Jacobi kernel (m <= 128)
  * potri_pool_ vs torch cholesky_inverse

    python scripts/bench_solver.py [--batch 8]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

DIMS = [64, 128, 256, 512, 576, 1024, 1152, 2048, 2304, 4608]


def spd(m, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(m, m, generator=g).cuda()
    return x @ x.t() / m + 0.1 * torch.eye(m, device="cuda")


def timeit(fn, warmup=2, iters=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=8)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    from kfac_pytorch_amd.ops import _ext
    solver = _ext.load_solver()
    b = args.batch

    try:
        has_magma = torch.backends.cuda.is_built() and \
            torch._C._has_magma
    except AttributeError:
        has_magma = False
    print(f"magma available = {has_magma}")

    hdr = (f"{'m':>6} {'eigh':>9} {'magma':>9} {'pool/8':>9} "
           f"{'syevdj/8':>10} {'syevdB/8':>10} {'jac/8':>9} "
           f"{'potri/8':>9} {'chol_inv':>9}")
    print(hdr + "   (ms per matrix)")
    for m in DIMS:
        a = spd(m)
        batch = torch.stack([spd(m, seed=i) for i in range(b)])

        t_eigh = timeit(lambda: torch.linalg.eigh(a))

        t_magma = float("nan")
        if has_magma:
            torch.backends.cuda.preferred_linalg_library("magma")
            try:
                t_magma = timeit(lambda: torch.linalg.eigh(a))
            finally:
                torch.backends.cuda.preferred_linalg_library("default")

        def pool():
            work = [batch[i].clone() for i in range(b)]
            solver.syevd_pool_(work)
        t_pool = timeit(pool) / b

        def sjb():
            w = batch.clone()
            solver.syevdj_batched_(w)
        try:
            t_sjb = timeit(sjb) / b
        except RuntimeError:
            t_sjb = float("nan")

        def sdb():
            w = batch.clone()
            solver.syevd_batched_(w)
        try:
            t_sdb = timeit(sdb) / b
        except RuntimeError:
            t_sdb = float("nan")

        t_jac = float("nan")
        if _ext.has_jacobi_eigh(m):
            def jac():
                _ext.jacobi_eigh_batched([batch[i].contiguous()
                                          for i in range(b)])
            t_jac = timeit(jac) / b

        def potri():
            work = [batch[i].clone() for i in range(b)]
            solver.potri_pool_(work)
        t_potri = timeit(potri) / b
        t_chol = timeit(
            lambda: torch.cholesky_inverse(torch.linalg.cholesky(a)))

        print(f"{m:>6} {t_eigh:9.2f} {t_magma:9.2f} {t_pool:9.2f} "
              f"{t_sjb:10.2f} {t_sdb:10.2f} {t_jac:9.2f} "
              f"{t_potri:9.2f} {t_chol:9.2f}")


if __name__ == "__main__":
    main()
