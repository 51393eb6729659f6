#!/usr/bin/env python3
"""Build profiles/pmc_manifest.json from rocprofv3 PMC csv passes.

The manifest ties a PMC-measured per-launch HBM traffic figure to the
content-addressed key of the kernel it was captured against, so bench.py
reports `roofline.traffic` only while the live kernel is byte-identical
to the profiled one (traffic must be runtime-derived
or invalidate itself).

Counter units and corrections follow the MI355X guide's HBM/rocprofv3
section: FETCH_SIZE / WRITE_SIZE are reported in KiB; on gfx950
FETCH_SIZE under-reports by 2x, so fetch bytes = value * 1024 * 2 and
write bytes = value * 1024.

Usage:
  python tools/make_pmc_manifest.py --fetch F.csv --write W.csv \
      --workload flagship --elems 1000000000 --world 1 \
      [--kernel k_<hash>] [--out profiles/pmc_manifest.json]

Without --kernel the dominant kernel (largest average fetch+write) is
selected.  Existing manifest entries for other kernels are preserved.
"""

import argparse
import csv
import json
import os
from collections import defaultdict


def per_kernel_avg(path):
    sums = defaultdict(float)
    counts = defaultdict(int)
    with open(path) as f:
        for row in csv.DictReader(f):
            name = row["Kernel_Name"].strip('"')
            sums[name] += float(row["Counter_Value"])
            counts[name] += 1
    return {k: sums[k] / counts[k] for k in sums}, dict(counts)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fetch", required=True)
    ap.add_argument("--write", required=True)
    ap.add_argument("--workload", required=True)
    ap.add_argument("--elems", type=int, required=True)
    ap.add_argument("--world", type=int, default=1)
    ap.add_argument("--kernel", default=None)
    ap.add_argument("--out", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "profiles", "pmc_manifest.json"))
    args = ap.parse_args()

    favg, fcnt = per_kernel_avg(args.fetch)
    wavg, wcnt = per_kernel_avg(args.write)
    names = set(favg) | set(wavg)
    if args.kernel:
        name = args.kernel
        assert name in names, f"{name} not in csvs ({sorted(names)})"
    else:
        name = max(names, key=lambda n: favg.get(n, 0.0) + wavg.get(n, 0.0))
    fetch_b = favg.get(name, 0.0) * 1024.0 * 2.0   # gfx950 x2 correction
    write_b = wavg.get(name, 0.0) * 1024.0
    if not name.startswith(("k_", "ax_")):
        raise SystemExit(f"unexpected kernel name {name}")
    key = name.split("_", 1)[1]

    man = {}
    if os.path.exists(args.out):
        man = json.load(open(args.out))
    man[key] = {
        "kernel_name": name,
        "workload": args.workload,
        "elems": args.elems,
        "world": args.world,
        "fetch_bytes_per_launch": fetch_b,
        "write_bytes_per_launch": write_b,
        "bytes_per_launch": fetch_b + write_b,
        "launches_seen": {"fetch": fcnt.get(name, 0),
                          "write": wcnt.get(name, 0)},
        "source": f"rocprofv3 PMC ({os.path.basename(args.fetch)} x2-corr "
                  f"+ {os.path.basename(args.write)})",
    }
    with open(args.out, "w") as f:
        json.dump(man, f, indent=1, sort_keys=True)
    print(f"{args.out}: {name} -> "
          f"{(fetch_b + write_b) / 1e9:.2f} GB/launch")


if __name__ == "__main__":
    main()
