#!/usr/bin/env python3
"""Optimal contiguous block partition of weighted layers across workers.

Reference analog: scripts/dp_block_partition.py -- dynamic programming
that minimizes the maximum per-worker cost when assigning a contiguous
block of layers to each of P workers (research tool; the runtime
scheduler stays round-robin, kfac/kfac_preconditioner_inv.py:62-77).
"""

from __future__ import annotations

import argparse
from typing import List, Tuple

# ResNet-50 per-layer eigensolve cost-model dims
# (reference: scripts/dp_block_partition.py:92-93)
RESNET50_A = [147, 576, 64, 576, 1024, 64, 256, 576, 1024, 256, 576, 1024,
              512, 1152, 2048, 256, 1152, 2048, 512, 1152, 2048, 512, 1152,
              2048, 1024, 2304, 4096, 512, 2304, 4096, 1024, 2304, 4096,
              1024, 2304, 4096, 1024, 2304, 4096, 1024, 2304, 4096, 2048,
              4608, 8192, 1024, 4608, 8192, 2048, 4608, 8192, 2048, 2048]
RESNET50_G = [64, 64, 256, 64, 256, 64, 64, 64, 256, 128, 128, 512, 128,
              128, 512, 512, 128, 512, 128, 128, 512, 256, 256, 1024, 256,
              256, 1024, 1024, 256, 1024, 256, 256, 1024, 256, 256, 1024,
              256, 256, 1024, 512, 512, 2048, 512, 512, 2048, 2048, 512,
              2048, 512, 512, 2048, 1000, 1000]


def eig_cost(m: int) -> float:
    """O(m^3) eigensolve cost proxy (reference: scripts/inverse_model.py)."""
    return float(m) ** 3


def block_partition(weights: List[float], P: int
                    ) -> Tuple[float, List[Tuple[int, int]]]:
    """Minimize max block sum over P contiguous blocks (DP, O(n^2 P))."""
    n = len(weights)
    prefix = [0.0]
    for w in weights:
        prefix.append(prefix[-1] + w)

    INF = float("inf")
    # dp[p][i] = best max-cost for first i layers in p blocks
    dp = [[INF] * (n + 1) for _ in range(P + 1)]
    cut = [[0] * (n + 1) for _ in range(P + 1)]
    dp[0][0] = 0.0
    for p in range(1, P + 1):
        for i in range(1, n + 1):
            for j in range(p - 1, i):
                cost = max(dp[p - 1][j], prefix[i] - prefix[j])
                if cost < dp[p][i]:
                    dp[p][i] = cost
                    cut[p][i] = j
    blocks = []
    i = n
    for p in range(P, 0, -1):
        j = cut[p][i]
        blocks.append((j, i))
        i = j
    blocks.reverse()
    return dp[P][n], blocks


def round_robin_cost(weights: List[float], P: int) -> float:
    per = [0.0] * P
    for i, w in enumerate(weights):
        per[i % P] += w
    return max(per)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=8)
    args = ap.parse_args()
    weights = [eig_cost(a) + eig_cost(g)
               for a, g in zip(RESNET50_A, RESNET50_G)]
    best, blocks = block_partition(weights, args.workers)
    rr = round_robin_cost(weights, args.workers)
    total = sum(weights)
    print(f"layers={len(weights)} workers={args.workers}")
    print(f"ideal     max-cost {total / args.workers:.3e}")
    print(f"optimal   max-cost {best:.3e}  (blocks: {blocks})")
    print(f"round-rob max-cost {rr:.3e}  "
          f"({rr / best:.2f}x optimal)")


if __name__ == "__main__":
    main()
