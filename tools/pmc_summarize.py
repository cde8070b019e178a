#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc counter_collection CSV per kernel.

Usage: python tools/pmc_summarize.py <glob-dir> <out.csv>
"""

import collections
import csv
import glob
import sys


def main():
    root, out_path = sys.argv[1], sys.argv[2]
    fs = [x for x in glob.glob(root + "/**/*.csv", recursive=True)
          if "counter" in x]
    if not fs:
        sys.exit("no counter_collection csv under " + root)
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    for r in csv.DictReader(open(fs[0])):
        agg[r["Kernel_Name"]][r["Counter_Name"]] += float(r["Counter_Value"])
    cols = ["SQ_BUSY_CYCLES", "SQ_WAVE_CYCLES", "SQ_INSTS_MFMA",
            "SQ_INSTS_LDS", "SQ_LDS_BANK_CONFLICT"]
    rows = sorted(agg.items(),
                  key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))[:10]
    with open(out_path, "w") as f:
        f.write("kernel," + ",".join(c.lower() for c in cols) + "\n")
        for kname, c in rows:
            short = kname.split("(")[0][:60]
            vals = ",".join("%.0f" % c.get(k, 0) for k in cols)
            print(short, vals)
            f.write(short + "," + vals + "\n")


if __name__ == "__main__":
    main()
