#!/usr/bin/env python3
"""Model introspection for K-FAC comm/compute planning.

Reference analog: scripts/reader.py (extract per-layer conv shapes and
tensor sizes from training logs to feed the comm/inverse cost models,
reference :6-57, :59-115).  Logs are a lossy medium for that; this
version walks the MODEL itself and prints, per hooked layer, the
factor dims, factor bytes, eigenbasis broadcast bytes and the
per-phase message totals -- the inputs scripts/comm_models.py and
scripts/inverse_model.py consume.

    python scripts/reader.py resnet50 [--world 8]
"""

import argparse
import os
import sys
from collections import Counter

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("model")
    ap.add_argument("--world", type=int, default=8)
    ap.add_argument("--cifar", action="store_true",
                    help="use the CIFAR model zoo instead of ImageNet")
    args = ap.parse_args()

    import torch.nn as nn
    from kfac_pytorch_amd.models import (get_cifar_model,
                                         get_imagenet_model)
    from kfac_pytorch_amd.ops.factors import factor_dims

    model = (get_cifar_model(args.model) if args.cifar
             else get_imagenet_model(args.model))
    rows = []
    skipped_groups = 0
    for m in model.modules():
        cls = m.__class__.__name__
        if cls not in ("Linear", "Conv2d"):
            continue
        da, dg = factor_dims(m)
        from kfac_pytorch_amd.ops.factors import factor_groups
        if factor_groups(m) > 1:
            skipped_groups += 1  # counted: grouped = block factors
        rows.append((cls, da, dg))

    print(f"{args.model}: {len(rows)} hooked layers"
          + (f" ({skipped_groups} grouped convs with exact per-group"
             " BLOCK factors)" if skipped_groups else ""))
    print(f"{'layer':8s} {'dim A':>6s} {'dim G':>6s} "
          f"{'A bytes':>10s} {'G bytes':>10s} {'eig bcast':>10s}")
    totA = totG = tot_eig = 0
    dim_hist = Counter()
    for cls, da, dg in rows:
        ab, gb = 4 * da * da, 4 * dg * dg
        eig = 4 * (da * da + da + dg * dg + dg)
        totA += ab
        totG += gb
        tot_eig += eig
        dim_hist[da] += 1
        dim_hist[dg] += 1
        print(f"{cls:8s} {da:6d} {dg:6d} {ab:10d} {gb:10d} {eig:10d}")
    print(f"\nfactor allreduce per MPD update: "
          f"{(totA + totG) / 2**20:.1f} MiB (one flat bucket)")
    print(f"eigenbasis broadcast per MPD-eigen update: "
          f"{tot_eig / 2**20:.1f} MiB over {args.world} owner buckets "
          f"(~{tot_eig / args.world / 2**20:.1f} MiB per root)")
    print(f"pred broadcast per DP step: "
          f"{sum(4 * da * dg for _, da, dg in rows) / 2**20:.1f} MiB")
    print("\nfactor-dim histogram (dim: count):",
          dict(sorted(dim_hist.items(), reverse=True)[:12]))


if __name__ == "__main__":
    sys.exit(main())
