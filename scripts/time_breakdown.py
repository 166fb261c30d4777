#!/usr/bin/env python3
"""Parse phase-timer logs into a per-phase time breakdown table.

Reference analog: scripts/time_breakdown.py + scripts/parse_logs.py +
scripts/reader.py (parse training logs for iteration speeds / phase
times, plot stacked bars of FF&BP / GradComm / FactorComp / FactorComm
/ InverseComp / InverseComm).  This framework's trainers print
``iter=..ms [io=.. fwbw=.. comm=.. kfac=.. update=..]`` lines
(examples/common.py train_loop) and ``bench.py`` prints
``KFAC_PHASES(ms/step): {...}``; this script aggregates either into a
text breakdown (no display server in this environment, so output is a
table rather than a matplotlib figure).

    python scripts/time_breakdown.py logfile [logfile ...]
"""

import json
import re
import sys
from collections import defaultdict

ITER_RE = re.compile(
    r"iter=(?P<iter>[\d.]+)ms \[io=(?P<io>[\d.]+)ms "
    r"fwbw=(?P<fwbw>[\d.]+)ms comm=(?P<comm>[\d.]+)ms "
    r"kfac=(?P<kfac>[\d.]+)ms update=(?P<update>[\d.]+)ms\]")
PHASES_RE = re.compile(r"KFAC_PHASES\(ms/step\): (\{.*\})")


def parse_file(path):
    iters = defaultdict(list)
    kfac_phases = {}
    with open(path) as f:
        for line in f:
            m = ITER_RE.search(line)
            if m:
                for k, v in m.groupdict().items():
                    iters[k].append(float(v))
            m = PHASES_RE.search(line)
            if m:
                kfac_phases = json.loads(m.group(1))
    return iters, kfac_phases


def main():
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    for path in sys.argv[1:]:
        iters, kfac_phases = parse_file(path)
        print(f"== {path}")
        if iters:
            n = len(iters["iter"])
            total = sum(iters["iter"]) / n
            print(f"  {n} samples, {total:.1f} ms/iter")
            for k in ("io", "fwbw", "comm", "kfac", "update"):
                mean = sum(iters[k]) / n
                bar = "#" * int(40 * mean / max(total, 1e-9))
                print(f"  {k:>8} {mean:8.2f} ms  {bar}")
        if kfac_phases:
            print("  K-FAC step phases (ms/step):")
            for k, v in sorted(kfac_phases.items(), key=lambda kv: -kv[1]):
                print(f"  {k:>24} {v:8.2f}")


if __name__ == "__main__":
    main()
