"""Direct quality comparison at benchmark scale: our GPU pipelines vs the
compiled reference's serial full partitioner on the same graph.

Usage (on a GPU box): python tools/quality_compare.py [scale] [k] [out.json]
Runs kref_compute_partition (oracle/_ref/libkaminpar_ref_full.so, serial,
seeds 1..2) and our partition() / partition_deep() on R-MAT <scale> k=<k>,
and writes cuts + timings. The reference lib must have been built in the
dev container (it travels with the snapshot).
"""

import ctypes
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import kaminpar_amd as ka
from kaminpar_amd.partition import partition, partition_deep


def main():
    scale = int(sys.argv[1]) if len(sys.argv) > 1 else 23
    k = int(sys.argv[2]) if len(sys.argv) > 2 else 16
    out_path = sys.argv[3] if len(sys.argv) > 3 else None
    kind = sys.argv[4] if len(sys.argv) > 4 else "rmat"

    if kind == "rgg2d":
        g = ka.Graph.rgg2d(1 << scale, avg_deg=16.0, seed=42)
    else:
        g = ka.Graph.rmat(scale, 8, 42)
    print(f"{kind}{scale}: n={g.n} m={g.m} k={k}", flush=True)
    out = {"graph": f"{kind}{scale}_s42", "n": g.n, "m": g.m, "k": k,
           "eps": 0.03, "cap": int(g.max_block_weight(k, 0.03))}

    lib_path = os.path.join(REPO, "oracle", "_ref", "libkaminpar_ref_full.so")
    if os.path.exists(lib_path):
        lib = ctypes.CDLL(lib_path)
        u32p = ctypes.POINTER(ctypes.c_uint32)
        lib.kref_compute_partition.restype = ctypes.c_int64
        xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
        adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
        out["reference"] = {}
        for seed in (1, 2):
            part = np.zeros(g.n, np.uint32)
            t0 = time.time()
            cut = lib.kref_compute_partition(
                ctypes.c_uint32(g.n), ctypes.c_uint64(g.m),
                xadj.ctypes.data_as(u32p), adjncy.ctypes.data_as(u32p),
                None, None, ctypes.c_uint32(k), ctypes.c_double(0.03),
                ctypes.c_int(seed), part.ctypes.data_as(u32p))
            dt = time.time() - t0
            maxb = int(np.bincount(part, minlength=k).max())
            out["reference"][f"seed{seed}"] = {
                "cut": int(cut), "max_block": maxb, "seconds": round(dt, 1)}
            print(f"reference seed{seed}: cut={cut} ({dt:.1f}s serial)",
                  flush=True)

    for name, fn in (("ours_basic", partition), ("ours_deep", partition_deep)):
        t0 = time.time()
        cut, part, levels = fn(g, k, seed=1)
        dt = time.time() - t0
        maxb = int(np.bincount(part, minlength=k).max())
        out[name] = {"cut": int(cut), "max_block": maxb,
                     "seconds": round(dt, 1)}
        print(f"{name}: cut={cut} max_block={maxb} ({dt:.1f}s)", flush=True)

    if "reference" in out and "ours_deep" in out:
        best = min(v["cut"] for v in out["reference"].values())
        out["deep_vs_reference_best"] = round(out["ours_deep"]["cut"] / best, 4)
        print("deep / reference_best =", out["deep_vs_reference_best"])

    if out_path:
        with open(out_path, "w") as fh:
            json.dump(out, fh, indent=1)


if __name__ == "__main__":
    main()
