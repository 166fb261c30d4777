"""Quick localization probe for the custom sytrd kernel: per-size
eigenvalue error of tridiag(D,E) vs eigvalsh(A), several n incl. tail
panels and batch > 1."""
import sys, os, torch
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from kfac_pytorch_amd.ops import _ext
solver = _ext.load_solver()
def spd(m, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(m, m, generator=g).to("cuda")
    return x @ x.t() / m + 0.1 * torch.eye(m, device="cuda")
for n, b in [(128, 2), (256, 1), (516, 3), (1152, 3)]:
    mats = torch.stack([spd(n, seed=n + i) for i in range(b)])
    work = mats.clone()
    E, tau, status = solver.sytrd_batched_custom_(work)
    torch.cuda.synchronize()
    errs = []
    for k in range(b):
        D = work[k].diagonal()
        T = (torch.diag(D) + torch.diag(E[k][:n-1], 1)
             + torch.diag(E[k][:n-1], -1))
        ev = torch.linalg.eigvalsh(T)
        ev_ref = torch.linalg.eigvalsh(mats[k])
        errs.append(float((ev - ev_ref).abs().max() / ev_ref.abs().max()))
    print(f"n={n:5d} b={b} status={status.tolist()} err={errs}", flush=True)
print("probe done")

# column-wise localization vs the numpy reference (first gross E/tau
# divergence pinpoints the buggy phase)
import numpy as np
from scripts.sytrd_ref import sytrd_blocked
n = 128
g = torch.Generator(device="cpu").manual_seed(99)
x = torch.randn(n, n, generator=g)
a_cpu = (x @ x.t() / n + 0.1 * torch.eye(n)).double().numpy()
a_gpu = torch.tensor(a_cpu, dtype=torch.float32, device="cuda")
work = a_gpu.unsqueeze(0).clone()
E, tau, status = solver.sytrd_batched_custom_(work)
torch.cuda.synchronize()
_, d_ref, e_ref, tau_ref = sytrd_blocked(a_cpu, 64)
e_gpu = E[0].cpu().double().numpy()
tau_gpu = tau[0].cpu().double().numpy()
d_gpu = work[0].diagonal().cpu().double().numpy()
first_bad = None
for jj in range(n - 1):
    de = abs(e_gpu[jj] - e_ref[jj]) / max(1e-6, abs(e_ref[jj]))
    dt = abs(tau_gpu[jj] - tau_ref[jj]) / max(1e-6, abs(tau_ref[jj]))
    dd = abs(d_gpu[jj] - d_ref[jj]) / max(1e-6, abs(d_ref[jj]))
    if max(de, dt, dd) > 2e-2 and first_bad is None:
        first_bad = jj
        print(f"first divergence at column {jj}: "
              f"E {e_gpu[jj]:.6f} vs {e_ref[jj]:.6f}, "
              f"tau {tau_gpu[jj]:.6f} vs {tau_ref[jj]:.6f}, "
              f"D {d_gpu[jj]:.6f} vs {d_ref[jj]:.6f}", flush=True)
if first_bad is None:
    print("column-wise check vs numpy ref: clean at n=128", flush=True)
