"""Is rocSOLVER's batched potri the right tier, or does a
trsm+GEMM reformulation (L=chol; X=trsm(L,I); A^-1 = X^T X -- all
GEMM-class batched ops) win at K-FAC factor shapes?"""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from kfac_pytorch_amd.ops import _ext
solver = _ext.load_solver()

def spd(m, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(m, m, generator=g).to("cuda")
    return x @ x.t() / m + 0.1 * torch.eye(m, device="cuda")

def timed(fn, reps=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3

for n, b in [(4608, 3), (2304, 13), (1152, 19), (576, 22), (256, 26)]:
    A = torch.stack([spd(n, seed=n + i) for i in range(b)])
    def rocsolver_potri():
        w = A.clone()
        solver.potri_batched_(w, -1)
    def chol_trsm():
        L = torch.linalg.cholesky(A)
        eye = torch.eye(n, device="cuda").expand(b, n, n)
        X = torch.linalg.solve_triangular(L, eye, upper=False)
        inv = X.mT @ X
        return inv
    def chol_chol_inv():
        L = torch.linalg.cholesky(A)
        return torch.cholesky_inverse(L)
    t1 = timed(rocsolver_potri)
    t2 = timed(chol_trsm)
    t3 = timed(chol_chol_inv)
    # correctness spot check
    inv2 = chol_trsm()
    err = ((A[0] @ inv2[0] - torch.eye(n, device="cuda")).norm()
           / n ** 0.5).item()
    print(f"n={n:5d} b={b:3d} potri_batched={t1:7.1f} ms  "
          f"chol+trsm+gemm={t2:7.1f} ms  chol_inverse={t3:7.1f} ms  "
          f"resid={err:.2e}", flush=True)
    del A; torch.cuda.empty_cache()
