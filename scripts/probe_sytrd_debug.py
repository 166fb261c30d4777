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
