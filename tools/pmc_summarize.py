"""Summarize rocprofv3 --pmc counter_collection.csv output per kernel.

Usage: python tools/pmc_summarize.py OUT.json DIR [DIR...]
Walks DIRs for *counter_collection.csv (one per --pmc pass: FETCH_SIZE and
WRITE_SIZE cannot share a TCC pass on gfx950), sums each counter per kernel
and counts dispatches. Values are raw; see MI355X_MICROARCH.md for the
gfx950 FETCH_SIZE wide-read correction (x2 for 16 B/lane coalesced reads).
"""

import csv
import glob
import json
import os
import sys


def main():
    out_path = sys.argv[1]
    kernels = {}
    for d in sys.argv[2:]:
        for path in glob.glob(os.path.join(d, "**", "*counter_collection.csv"),
                              recursive=True):
            with open(path, newline="") as fh:
                for row in csv.DictReader(fh):
                    name = row.get("Kernel_Name") or row.get("Kernel-Name")
                    if name:
                        name = name.replace("(anonymous namespace)::", "")
                        name = name.split("(")[0].strip()
                    counter = row.get("Counter_Name") or row.get("Counter-Name")
                    value = float(row.get("Counter_Value")
                                  or row.get("Counter-Value") or 0)
                    disp = row.get("Dispatch_Id") or row.get("Dispatch-Id")
                    if not name or not counter:
                        continue
                    short = name.split("(")[0].strip()
                    k = kernels.setdefault(short, {"dispatches": set()})
                    k[counter] = k.get(counter, 0.0) + value
                    k["dispatches"].add((path, disp))
    for k in kernels.values():
        k["launches"] = len(k.pop("dispatches"))
    ordered = dict(sorted(kernels.items(),
                          key=lambda kv: -kv[1].get("FETCH_SIZE", 0)))
    with open(out_path, "w") as fh:
        json.dump(ordered, fh, indent=1)
    for name, k in list(ordered.items())[:8]:
        print(name, {c: v for c, v in k.items() if c != "launches"},
              "launches:", k["launches"])


if __name__ == "__main__":
    main()
