#!/usr/bin/env python3
"""Summarize rocprofv3 kernel_stats CSVs into a top-N breakdown."""
import csv
import glob
import sys

out_path = sys.argv[1]
pat = sys.argv[2]
rows = []
for f in glob.glob(pat, recursive=True):
    for r in csv.DictReader(open(f)):
        if r.get("Name") in ("KERNEL_DISPATCH", "MEMORY_COPY", "TOTAL"):
            continue  # aggregate rows double the total
        rows.append(r)
key = "TotalDurationNs" if rows and "TotalDurationNs" in rows[0] else "DurationNs"
rows.sort(key=lambda r: -float(r.get(key, 0)))
tot = sum(float(r.get(key, 0)) for r in rows)
with open(out_path, "w") as out:
    out.write(sys.argv[3] + "\n" if len(sys.argv) > 3 else "")
    for r in rows[:18]:
        ns = float(r.get(key, 0))
        name = r.get("Name", "?")[:72]
        out.write("%-74s %10.1f ms %5.1f%%\n" % (name, ns / 1e6,
                                                 100 * ns / max(tot, 1)))
print(open(out_path).read())
