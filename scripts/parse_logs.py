#!/usr/bin/env python3
"""Aggregate training/bench logs into a speed table.

Reference analog: scripts/parse_logs.py (extract average iteration
speeds from a directory of training logs, one row per run,
reference :4-30).  This framework's logs carry either the trainer's
``iter=..ms [...]`` lines (examples/common.py), the bench's single
JSON line, or epoch summaries ``epoch=E .. img/s=V``; this script
accepts any mix and prints one row per file: mean +- std iteration
time, throughput, and the run's config when a JSON line is present.

    python scripts/parse_logs.py logs/*.log
"""

import json
import re
import statistics
import sys

ITER_RE = re.compile(r"iter=(?P<ms>[\d.]+)ms")
EPOCH_RE = re.compile(r"img/s=(?P<ips>[\d.]+)")


def parse(path):
    iters, ips, cfg = [], [], None
    with open(path) as f:
        for line in f:
            m = ITER_RE.search(line)
            if m:
                iters.append(float(m.group("ms")))
            m = EPOCH_RE.search(line)
            if m:
                ips.append(float(m.group("ips")))
            if line.startswith("{") and '"metric"' in line:
                try:
                    d = json.loads(line)
                    cfg = d
                    if "ms_per_step" in d:
                        iters.append(float(d["ms_per_step"]))
                    if "value" in d:
                        ips.append(float(d["value"]))
                except (json.JSONDecodeError, TypeError):
                    pass
    return iters, ips, cfg


def main(paths):
    print(f"{'log':40s} {'iter ms (mean+-std)':>22s} {'img/s':>10s}  config")
    for p in paths:
        iters, ips, cfg = parse(p)
        if not iters and not ips:
            print(f"{p:40s} {'-':>22s} {'-':>10s}  (no speed lines)")
            continue
        it = (f"{statistics.mean(iters):8.1f}+-"
              f"{(statistics.stdev(iters) if len(iters) > 1 else 0):5.1f}"
              if iters else "-")
        v = f"{ips[-1]:10.1f}" if ips else f"{'-':>10s}"
        c = ""
        if cfg and isinstance(cfg.get("config"), dict):
            cc = cfg["config"]
            c = " ".join(f"{k}={cc[k]}" for k in
                         ("model", "kfac", "global_batch", "parallelism")
                         if k in cc)
        print(f"{p:40s} {it:>22s} {v}  {c}")


if __name__ == "__main__":
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    main(sys.argv[1:])
