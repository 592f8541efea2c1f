#!/usr/bin/env python3
"""Summarise a rocprofv3 --pmc FETCH_SIZE WRITE_SIZE run into per-kernel
effective HBM traffic + bandwidth (needs the matching kernel durations
from a --stats run or the counter file's own timestamps).

Usage: python tools/pmc_summary.py gpurun_out/pmc > profiles/pmc_summary.md
"""
import csv
import glob
import sys
from collections import defaultdict


def main(d):
    files = glob.glob(f"{d}/**/*counter_collection.csv", recursive=True) \
        + glob.glob(f"{d}/*counter_collection.csv")
    if not files:
        print(f"no counter_collection.csv under {d}", file=sys.stderr)
        return 1
    # per-kernel: accumulate counter values and dispatch durations
    acc = defaultdict(lambda: defaultdict(float))
    dur = defaultdict(float)
    count = defaultdict(set)
    for f in files:
        with open(f) as fh:
            for row in csv.DictReader(fh):
                name = row.get("Kernel_Name", row.get("Kernel-Name", ""))
                short = name.split("(")[0].split("<")[0].strip()
                cname = row.get("Counter_Name", row.get("Counter-Name"))
                val = float(row.get("Counter_Value",
                                    row.get("Counter-Value", 0)))
                acc[short][cname] += val
                did = row.get("Dispatch_Id", row.get("Dispatch-Id"))
                if did is not None and did not in count[short]:
                    count[short].add(did)
                    s = row.get("Start_Timestamp")
                    e = row.get("End_Timestamp")
                    if s and e:
                        dur[short] += (float(e) - float(s)) / 1e3  # ns->us
    print("| kernel | dispatches | fetch GB | write GB | Σus | TB/s |")
    print("|---|---|---|---|---|---|")
    for k in sorted(acc, key=lambda k: -dur[k]):
        # FETCH_SIZE/WRITE_SIZE are reported in KB per dispatch
        fetch = acc[k].get("FETCH_SIZE", 0) / 1024 / 1024  # KB -> GB
        write = acc[k].get("WRITE_SIZE", 0) / 1024 / 1024
        us = dur[k]
        bw = (fetch + write) / (us / 1e6) / 1024 if us else 0  # GB/s->TB/s
        print(f"| {k[:60]} | {len(count[k])} | {fetch:.2f} | {write:.2f} "
              f"| {us:.0f} | {bw:.2f} |")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/pmc"))
