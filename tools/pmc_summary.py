#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc counter_collection.csv: per kernel, mean
of each counter over dispatches, plus the MFMA issue fraction
SQ_VALU_MFMA_BUSY_CYCLES / (4 * SQ_WAVE_CYCLES)  (WAVE_CYCLES counts
quad-cycles; MFMA busy counts cycles — MI355X_MICROARCH 'rocprofv3 PMC
slots').  Usage: pmc_summary.py counter_collection.csv [out.csv]"""
import csv
import sys
from collections import defaultdict


def main():
    path = sys.argv[1]
    rows = list(csv.DictReader(open(path)))
    # rocprofv3 emits one row per (dispatch, counter)
    per = defaultdict(lambda: defaultdict(list))
    for r in rows:
        kn = r.get("Kernel_Name") or r.get("Kernel Name") or r.get("kernel_name")
        cn = r.get("Counter_Name") or r.get("Counter Name") or r.get("counter_name")
        cv = r.get("Counter_Value") or r.get("Counter Value") or r.get("counter_value")
        if kn is None or cn is None:
            continue
        kn = kn.split("(")[0].strip().split("<")[0]
        per[kn][cn].append(float(cv))
    out = []
    counters = sorted({c for k in per.values() for c in k})
    for kn, cs in sorted(per.items()):
        row = {"kernel": kn,
               "dispatches": max(len(v) for v in cs.values())}
        for c in counters:
            v = cs.get(c)
            row[c] = round(sum(v) / len(v), 1) if v else ""
        wc, mb = cs.get("SQ_WAVE_CYCLES"), cs.get("SQ_VALU_MFMA_BUSY_CYCLES")
        if wc and mb and sum(wc) > 0:
            row["mfma_issue_frac"] = round(
                (sum(mb) / len(mb)) / (4 * sum(wc) / len(wc)), 3)
        out.append(row)
    cols = ["kernel", "dispatches"] + counters + ["mfma_issue_frac"]
    w = csv.DictWriter(
        open(sys.argv[2], "w") if len(sys.argv) > 2 else sys.stdout,
        fieldnames=cols)
    w.writeheader()
    for r in out:
        w.writerow(r)


if __name__ == "__main__":
    main()
