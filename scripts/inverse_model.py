#!/usr/bin/env python3
"""Eigensolve / Cholesky-inverse cost vs factor dimension.

Reference analog: scripts/inverse_model.py (times ``torch.symeig`` /
inverse over the ResNet-50 factor dims and fits a cubic cost model used
by the load-balancing research).  Measures this framework's actual
``mat_eig`` (batched Jacobi HIP kernel or rocSOLVER eigh) and
``mat_inv`` and fits t = c3*m^3 + c0.

    python scripts/inverse_model.py
"""

import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# ResNet-50 factor dims (reference: scripts/inverse_model.py:19-20)
DIMS = [27, 64, 65, 128, 129, 147, 256, 257, 512, 513, 576, 1024, 1025,
        1152, 2048, 2049, 2304, 4608]


def timeit(fn, warmup=2, iters=5, cuda=False):
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


def fit_cubic(ms, ts):
    A = np.vstack([np.ones(len(ms)), np.array(ms, float) ** 3]).T
    (c0, c3), *_ = np.linalg.lstsq(A, np.array(ts), rcond=None)
    return c0, c3


def main():
    from kfac_pytorch_amd.ops.linalg import mat_eig, mat_inv
    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"
    eig_t, inv_t = [], []
    print(f"{'m':>6} {'eig_ms':>10} {'chol_inv_ms':>12}")
    for m in DIMS:
        x = torch.randn(m, m, device=device)
        a = x @ x.t() / m + torch.eye(m, device=device)
        te = timeit(lambda: mat_eig(a), cuda=use_cuda) * 1e3
        ti = timeit(lambda: mat_inv(a), cuda=use_cuda) * 1e3
        eig_t.append(te)
        inv_t.append(ti)
        print(f"{m:>6} {te:10.3f} {ti:12.3f}")
    for name, ts in (("eig", eig_t), ("chol_inv", inv_t)):
        c0, c3 = fit_cubic(DIMS, ts)
        print(f"{name}: t[ms] ~= {c0:.4f} + {c3:.3e} * m^3")


if __name__ == "__main__":
    main()
