"""Convergence evidence: K-FAC (eigen_dp) vs plain SGD on a locally
generated CIFAR-shaped classification task, accuracy vs epochs AND
wall-clock (the reference's SPEED=False mode on real CIFAR-10,
reference: batch.sh:9-14, examples/pytorch_cifar10_resnet.py:340-375;
no dataset download exists in this environment, so the data is a
deterministic generated task that requires real generalization:
10 classes x 5 latent templates, additive noise, random shifts and
flips, disjoint train/test draws).

    python scripts/convergence_cifar.py [--epochs 12] [--opt both]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch
import torch.distributed as dist
import torch.nn.functional as F


def make_dataset(n, seed, templates):
    g = torch.Generator().manual_seed(seed)
    y = torch.randint(0, 10, (n,), generator=g)
    tpl_idx = torch.randint(0, 5, (n,), generator=g)
    x = templates[y, tpl_idx].clone()
    x += 0.8 * torch.randn(x.shape, generator=g)
    # random cyclic shifts up to +-4 px and horizontal flips
    for i in range(n):
        dx = int(torch.randint(-4, 5, (1,), generator=g))
        dy = int(torch.randint(-4, 5, (1,), generator=g))
        x[i] = torch.roll(x[i], shifts=(dy, dx), dims=(1, 2))
        if int(torch.randint(0, 2, (1,), generator=g)):
            x[i] = torch.flip(x[i], dims=(2,))
    return x, y


def run(opt_name, args, xtr, ytr, xte, yte, device):
    from kfac_pytorch_amd.models import get_cifar_model
    import kfac_pytorch_amd as kfac

    torch.manual_seed(7)
    model = get_cifar_model(args.model).to(device)
    base_lr = 0.05
    optimizer = torch.optim.SGD(model.parameters(), lr=base_lr,
                                momentum=0.9, weight_decay=5e-4)
    precond = None
    if opt_name == "kfac":
        precond = kfac.KFAC_EIGEN_DP(model, lr=base_lr, damping=0.03,
                                     fac_update_freq=1,
                                     kfac_update_freq=10)
    # warmup + step decay at 50%/75% (the reference's CIFAR recipe
    # shape, examples/pytorch_cifar10_resnet.py lr schedule)
    def lr_fn(epoch):
        if epoch < 1:
            return 0.2 + 0.8 * epoch
        f = 1.0
        if epoch >= args.epochs // 2:
            f *= 0.1
        if epoch >= (3 * args.epochs) // 4:
            f *= 0.1
        return f
    scheds = [torch.optim.lr_scheduler.LambdaLR(optimizer, lr_fn)]
    if precond is not None:
        scheds.append(torch.optim.lr_scheduler.LambdaLR(precond, lr_fn))
    n = xtr.shape[0]
    bs = args.batch_size
    hist = []
    t0 = time.perf_counter()
    for epoch in range(args.epochs):
        model.train()
        perm = torch.randperm(n, generator=torch.Generator().manual_seed(
            epoch))
        for i in range(0, n - bs + 1, bs):
            idx = perm[i:i + bs]
            xb = xtr[idx].to(device, non_blocking=True)
            yb = ytr[idx].to(device, non_blocking=True)
            optimizer.zero_grad(set_to_none=False)
            loss = F.cross_entropy(model(xb), yb)
            loss.backward()
            if precond is not None:
                precond.step()
            optimizer.step()
        for s in scheds:
            s.step()
        if device.type == "cuda":
            torch.cuda.synchronize()
        wall = time.perf_counter() - t0
        model.eval()
        correct = 0
        with torch.no_grad():
            for i in range(0, xte.shape[0], 512):
                xb = xte[i:i + 512].to(device)
                pred = model(xb).argmax(1).cpu()
                correct += int((pred == yte[i:i + 512]).sum())
        acc = correct / xte.shape[0]
        hist.append((epoch, wall, acc))
        print(f"[{opt_name}] epoch {epoch:2d} wall {wall:7.1f}s "
              f"test-acc {acc:.4f}", flush=True)
    return hist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet32")
    ap.add_argument("--epochs", type=int, default=12)
    ap.add_argument("--batch-size", type=int, default=128)
    ap.add_argument("--train-n", type=int, default=20000)
    ap.add_argument("--test-n", type=int, default=2000)
    ap.add_argument("--opt", default="both",
                    choices=["both", "sgd", "kfac"])
    args = ap.parse_args()

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29761")
        dist.init_process_group("gloo", world_size=1, rank=0,
                                init_method="env://")
    import kfac_pytorch_amd.backend as backend
    backend.init("Torch")

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    g = torch.Generator().manual_seed(1234)
    templates = torch.randn(10, 5, 3, 32, 32, generator=g) * 1.2
    xtr, ytr = make_dataset(args.train_n, 100, templates)
    xte, yte = make_dataset(args.test_n, 200, templates)
    print(f"dataset: {args.train_n} train / {args.test_n} test, "
          f"10 classes x 5 templates, noise 0.8, shifts +-4, flips",
          flush=True)

    results = {}
    for opt_name in (["sgd", "kfac"] if args.opt == "both"
                     else [args.opt]):
        results[opt_name] = run(opt_name, args, xtr, ytr, xte, yte,
                                device)
    if len(results) == 2:
        for target in (0.6, 0.7, 0.8, 0.85, 0.9):
            row = {}
            for k, hist in results.items():
                hit = next(((e, w) for e, w, a in hist if a >= target),
                           None)
                row[k] = hit
            print(f"acc>={target:.2f}: sgd={row['sgd']} "
                  f"kfac={row['kfac']}  (epoch, wall s)", flush=True)


if __name__ == "__main__":
    main()
