#!/usr/bin/env python3
"""Sweep the XCD-constrained grid barrier (VERDICT r1 weak #1 diagnosis):
does co-locating all persistent-scan blocks on fewer XCDs cut the
per-step sync floor? Writes gpurun_out/barrier_xcd.json."""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nats_amd.ops import _hip_ext


def main():
    ext = _hip_ext()
    assert ext is not None
    results = {}
    for nwg in (32, 63, 126):
        for nxcd in (1, 2, 4, 8):
            # nxcd XCDs hold 32 CUs each; cap participants accordingly
            # (2 blocks/CU fit at 384 threads, but keep 1/CU like the
            # production scans' LDS budget would force)
            if nwg > nxcd * 32:
                continue
            us = min(ext.barrier_bench_xcd(nwg, nxcd, 2000)
                     for _ in range(3))
            results["nwg%d_xcd%d" % (nwg, nxcd)] = round(us, 3)
            print("nwg=%d nxcd=%d: %.2f us" % (nwg, nxcd, us), flush=True)
    # unconstrained reference
    for nwg in (32, 63, 126):
        us = min(ext.barrier_bench(nwg, 1, 2000) for _ in range(3))
        results["nwg%d_free" % nwg] = round(us, 3)
        print("nwg=%d free: %.2f us" % (nwg, us), flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/barrier_xcd.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
