#!/usr/bin/env python3
"""Aggregate rocprofv3 PMC counter_collection.csv files into the
parked/issue-stall/active attribution used across profiles/ (SQ counters;
see MI355X_MICROARCH.md: WAIT_ANY + WAIT_INST_ANY + ACTIVE ~= WAVE_CYCLES)."""
import collections
import csv
import sys


def agg(path):
    a = collections.defaultdict(lambda: collections.defaultdict(float))
    with open(path) as f:
        for row in csv.DictReader(f):
            k = row.get("Kernel_Name", "?").split("(")[0].replace("void ", "")
            a[k][row["Counter_Name"]] += float(row["Counter_Value"])
    return a


def main():
    for arg in sys.argv[1:]:
        path_sep = arg.rpartition("="); label, path = path_sep[0], path_sep[2]
        print("##", label)
        for k, c in sorted(agg(path).items()):
            wc = c.get("SQ_WAVE_CYCLES", 1)
            if wc < 1e6:
                continue
            print(
                f"{k:26s} parked={c.get('SQ_WAIT_ANY', 0) / wc:5.1%} "
                f"issue-stall={c.get('SQ_WAIT_INST_ANY', 0) / wc:5.1%} "
                f"active={c.get('SQ_ACTIVE_INST_ANY', 0) / wc:5.1%}"
            )


if __name__ == "__main__":
    main()
