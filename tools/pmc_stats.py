#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc CSV per kernel: MFMA pipe utilization, LDS
bank conflicts, wave-parked (%WAIT_ANY) and issue-stall (%WAIT_INST_ANY)
shares of wave cycles.

Collect (own run — never combine --pmc with trace domains):
  rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES \
      SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      -d OUT -o pmc --output-format csv -- <cmd>
Usage: python tools/pmc_stats.py OUT/pmc_counter_collection.csv [top_n]
"""
import collections
import csv
import sys


def main(path, top=20):
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    with open(path) as f:
        for row in csv.DictReader(f):
            name = row.get("Kernel_Name", "?").split("(")[0][:58]
            agg[name][row["Counter_Name"]] += float(row["Counter_Value"])
    print(f"{'kernel':58s} {'MFMAbusy%':>9s} {'LDSconf%':>8s} {'parked%':>7s} "
          f"{'stall%':>6s} {'Mwavecyc':>9s}")
    rows = []
    for name, c in agg.items():
        wc = c.get("SQ_WAVE_CYCLES", 0)
        if wc == 0:
            continue
        # WAVE_CYCLES/WAIT_* count quad-cycles; MFMA_BUSY counts cycles
        rows.append((wc, name,
                     100 * c.get("SQ_VALU_MFMA_BUSY_CYCLES", 0) / 4 / wc,
                     100 * c.get("SQ_LDS_BANK_CONFLICT", 0) / wc,
                     100 * c.get("SQ_WAIT_ANY", 0) / wc,
                     100 * c.get("SQ_WAIT_INST_ANY", 0) / wc))
    for wc, name, m, l, w, wi in sorted(rows, reverse=True)[:top]:
        print(f"{name:58s} {m:9.1f} {l:8.1f} {w:7.1f} {wi:6.1f} {wc/1e6:9.0f}")


if __name__ == "__main__":
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 20)
