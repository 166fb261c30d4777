"""Summarize a rocprofv3 counter_collection.csv into per-kernel counter
sums (robust to column-name variants)."""
import csv
import sys
from collections import defaultdict

path = sys.argv[1]
with open(path) as fh:
    r = csv.DictReader(fh)
    cols = r.fieldnames
    namecol = next(c for c in cols if "Kernel" in c and "Name" in c)
    ctrcol = next(c for c in cols if "Counter" in c and "Name" in c)
    valcol = next(c for c in cols if "Value" in c)
    agg = defaultdict(lambda: defaultdict(float))
    disp = defaultdict(set)
    dispcol = next((c for c in cols if "Dispatch" in c and "Id" in c), None)
    for row in r:
        raw = row[namecol].strip('"')
        if raw.startswith("(anonymous namespace)::"):
            raw = raw[len("(anonymous namespace)::"):]
        name = raw.split("(")[0].strip() or raw[:55]
        agg[name][row[ctrcol]] += float(row[valcol])
        if dispcol:
            disp[name].add(row[dispcol])
print(f"{'kernel':55s} {'dispatches':>10s} {'MFMA_BUSY':>14s} "
      f"{'WAVE_CYC':>14s} {'LDS_CONF':>12s} {'GUI_ACTIVE':>14s} {'MFMA%':>6s}")
for name, ctrs in sorted(agg.items(),
                         key=lambda kv: -kv[1].get("GRBM_GUI_ACTIVE", 0)):
    mfma = ctrs.get("SQ_VALU_MFMA_BUSY_CYCLES", 0)
    gui = ctrs.get("GRBM_GUI_ACTIVE", 0)
    # RELATIVE MFMA-occupancy proxy: SQ_VALU_MFMA_BUSY summed over all
    # SIMDs vs chip-active cycles x 1024 SIMDs.  gfx950 ships no
    # derived-counter formulas (guide: rocprofv3 PMC slots), so compare
    # kernels against each other, not against an absolute peak.
    util = mfma / (gui * 1024) * 100 if gui else 0.0
    print(f"{name[:55]:55s} {len(disp[name]):>10d} {mfma:14.3e} "
          f"{ctrs.get('SQ_WAVE_CYCLES', 0):14.3e} "
          f"{ctrs.get('SQ_LDS_BANK_CONFLICT', 0):12.3e} "
          f"{gui:14.3e} {util:6.1f}")
