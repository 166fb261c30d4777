#!/usr/bin/env python3
"""K-FAC vs SGD convergence check on a fixed synthetic task.

Reference analog: the convergence-mode training runs (SPEED=False,
examples/pytorch_cifar10_resnet.py:350-375) -- no CIFAR download exists
in this environment, so this uses a fixed synthetic classification set
(same data every epoch, real train-loss minimization) and reports the
loss trajectory of SGD vs SGD+K-FAC at matched learning rates.

    python scripts/convergence_check.py [--steps 150] [--kfac eigen_dp]
"""

import argparse
import os
import sys

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_task(seed=0, n=512, d=32, classes=10):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, d, generator=g)
    teacher = nn.Sequential(nn.Linear(d, 64), nn.Tanh(),
                            nn.Linear(64, classes))
    with torch.no_grad():
        y = teacher(x).argmax(-1)
    return x, y


def make_model(seed, d=32, classes=10):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(d, 64), nn.ReLU(),
                         nn.Linear(64, 64), nn.ReLU(),
                         nn.Linear(64, classes))


def train(x, y, use_kfac, kfac_name, lr, steps, device):
    import kfac_pytorch_amd as kfac
    model = make_model(1).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    pre = None
    if use_kfac:
        pre = kfac.get_kfac_module(kfac_name)(
            model, lr=lr, damping=0.01, kfac_update_freq=5)
    losses = []
    for _ in range(steps):
        opt.zero_grad(set_to_none=False)
        loss = F.cross_entropy(model(x), y)
        loss.backward()
        if pre is not None:
            pre.step()
        opt.step()
        losses.append(loss.item())
    return losses


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=150)
    ap.add_argument("--lr", type=float, default=0.05)
    ap.add_argument("--kfac", default="eigen_dp")
    args = ap.parse_args()

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29761")
        dist.init_process_group("gloo", world_size=1, rank=0,
                                init_method="env://")
    import kfac_pytorch_amd.backend as backend
    backend.init("Torch")

    device = "cuda" if torch.cuda.is_available() else "cpu"
    x, y = make_task()
    x, y = x.to(device), y.to(device)

    sgd = train(x, y, False, args.kfac, args.lr, args.steps, device)
    kf = train(x, y, True, args.kfac, args.lr, args.steps, device)

    print(f"{'step':>6} {'sgd':>10} {'sgd+kfac':>10}")
    for s in range(0, args.steps, max(1, args.steps // 10)):
        print(f"{s:>6} {sgd[s]:>10.4f} {kf[s]:>10.4f}")
    print(f"{'final':>6} {sgd[-1]:>10.4f} {kf[-1]:>10.4f}")

    # steps for SGD to reach K-FAC's final loss (simple speedup proxy)
    target = kf[-1]
    reach = next((i for i, v in enumerate(sgd) if v <= target), None)
    if reach is None:
        print(f"SGD never reaches K-FAC's final loss {target:.4f} "
              f"within {args.steps} steps")
    else:
        print(f"K-FAC reaches loss {target:.4f} in {args.steps} steps; "
              f"SGD needs {reach}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
