"""Measure the custom tridiagonalization path against the rocSOLVER
batched-syevd tier on the ResNet-50 factor buckets (the flagship
bench's measured distribution: 4608 x 3, 2304-class x 13, 1152-class
x 19 -- profiles/PERFORMANCE.md).

Usage (on a GPU box):  python scripts/bench_sytrd.py [--reps 3]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def spd(m, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(m, m, generator=g).to("cuda")
    return x @ x.t() / m + 0.1 * torch.eye(m, device="cuda")


def timed(fn, reps):
    fn()  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()
    from kfac_pytorch_amd.ops import _ext
    from kfac_pytorch_amd.ops import linalg
    solver = _ext.load_solver()
    has_custom = hasattr(solver, "sytrd_batched_custom_")
    print(f"custom sytrd available: {has_custom}", flush=True)

    buckets = [(4608, 3), (2304, 13), (1152, 19)]
    for n, b in buckets:
        mats = torch.stack([spd(n, seed=n + i) for i in range(b)])

        def run_syevd():
            w = mats.clone()
            solver.syevd_batched_(w, -1)

        t_lib = timed(run_syevd, args.reps)

        t_sytrd = t_stedc = t_wy = float("nan")
        if has_custom:
            def run_sytrd():
                w = mats.clone()
                solver.sytrd_batched_custom_(w)

            t_sytrd = timed(run_sytrd, args.reps)

            work = mats.clone()
            E, tau, status = solver.sytrd_batched_custom_(work)
            torch.cuda.synchronize()
            st = int(status.abs().sum())
            D = work.diagonal(dim1=1, dim2=2).contiguous()

            def run_stedc():
                Dw = D.clone()
                for k in range(b):
                    solver.stedc_slot_(Dw[k], E[k], k % 8)
                solver.join_pool_()

            t_stedc = timed(run_stedc, args.reps)

            Dw = D.clone()
            Cs = []
            for k in range(b):
                C, _ = solver.stedc_slot_(Dw[k], E[k], k % 8)
                Cs.append(C)
            solver.join_pool_()
            Cstack = torch.stack(Cs)

            def run_wy():
                linalg._wy_backtransform(work, tau, Cstack)

            t_wy = timed(run_wy, args.reps)
            print(f"n={n:5d} b={b:3d} syevd={t_lib:8.1f} ms | custom: "
                  f"sytrd={t_sytrd:8.1f} stedc={t_stedc:7.1f} "
                  f"wy={t_wy:6.1f} sum={t_sytrd + t_stedc + t_wy:8.1f} "
                  f"status={st}", flush=True)
        else:
            print(f"n={n:5d} b={b:3d} syevd={t_lib:8.1f} ms", flush=True)
        del mats
        torch.cuda.empty_cache()

    # full pipelined dispatch over all three buckets at once
    allmats = ([spd(4608, seed=i) for i in range(3)]
               + [spd(2304, seed=50 + i) for i in range(13)]
               + [spd(1152, seed=90 + i) for i in range(19)])

    def run_multi():
        linalg.mat_eig_multi(allmats, need_sorted=False)

    for flag, label in [("1", "custom"), ("0", "library")]:
        os.environ["KFAC_CUSTOM_SYTRD"] = flag
        t = timed(run_multi, args.reps)
        print(f"mat_eig_multi all buckets [{label:7s}]: {t:8.1f} ms",
              flush=True)


if __name__ == "__main__":
    main()
