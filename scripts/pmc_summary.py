"""Aggregate rocprofv3 --pmc CSVs into a per-kernel counter table.

Usage: python scripts/pmc_summary.py gpurun_out/pmc/*.csv
Prints mean counter value per (kernel, counter) for the hand-written
kernels, weighted equally per dispatch.  Diagnostic tooling for the
profiles/ evidence — not part of the test suite.
"""
import csv
import re
import sys
from collections import defaultdict

OURS = ("gemm_nt", "attn_", "norm_", "bn_", "adamw", "sgd_", "ce_",
        "rope", "silu", "transpose")


def short(name: str) -> str:
    m = re.search(r"(\w+_kernel[^(]*)", name)
    s = m.group(1) if m else name
    return s[:60]


def main(paths):
    agg = defaultdict(list)   # (kernel, counter) -> [values]
    for p in paths:
        with open(p) as f:
            for row in csv.DictReader(f):
                name = row.get("Kernel_Name") or row.get("Kernel Name") \
                    or ""
                if not any(k in name for k in OURS):
                    continue
                ctr = row.get("Counter_Name") or row.get("Counter") or ""
                val = row.get("Counter_Value") or row.get("Value") or ""
                try:
                    agg[(short(name), ctr)].append(float(val))
                except ValueError:
                    pass
    table = defaultdict(dict)
    for (kn, ctr), vals in agg.items():
        table[kn][ctr] = sum(vals) / len(vals)
    ctrs = sorted({c for row in table.values() for c in row})
    print(f"{'kernel':60s} " + " ".join(f"{c:>16s}" for c in ctrs))
    for kn in sorted(table):
        cells = " ".join(
            f"{table[kn].get(c, float('nan')):16.2f}" for c in ctrs)
        print(f"{kn:60s} {cells}")


if __name__ == "__main__":
    main(sys.argv[1:])
