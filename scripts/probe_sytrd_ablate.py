"""Ablation: where does the custom sytrd time go?  mode 2 = barriers
only, mode 1 = everything but the matvec inner loop, mode 0 = full;
swept over workgroup targets."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from kfac_pytorch_amd.ops import _ext
solver = _ext.load_solver()

def spd(m, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(m, m, generator=g).to("cuda")
    return x @ x.t() / m + 0.1 * torch.eye(m, device="cuda")

for n, b in [(4608, 3), (2304, 13)]:
    mats = torch.stack([spd(n, seed=n + i) for i in range(b)])
    for wgs_t in (256, 512, 768):
        os.environ["KFAC_SYTRD_WGS"] = str(wgs_t)
        row = [f"n={n} b={b} wgs_t={wgs_t}:"]
        for mode in (2, 1, 0):
            os.environ["KFAC_SYTRD_MODE"] = str(mode)
            w = mats.clone(); solver.sytrd_batched_custom_(w)  # warm
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(2):
                w = mats.clone()
                solver.sytrd_batched_custom_(w)
            torch.cuda.synchronize()
            ms = (time.perf_counter() - t0) / 2 * 1e3
            row.append(f"m{mode}={ms:7.1f}")
        print(" ".join(row), flush=True)
    del mats; torch.cuda.empty_cache()
os.environ.pop("KFAC_SYTRD_MODE", None)
