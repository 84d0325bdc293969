"""Quality vs the compiled reference, runnable WITHOUT a GPU: partitions
through the CPU oracle mirror (tests/oracle_pipeline.py), which is
bit-identical to the GPU pipeline stage by stage, so the cuts it reports
are exactly what the GPU pipeline produces on the same inputs.

Usage: python tools/quality_cpu_oracle.py {rmat|rgg2d} SCALE K [out.json] [late]
Requires oracle/_ref/libkaminpar_ref_full.so (built by
oracle/_ref_build/Makefile.full from /root/reference in the dev
container). rgg2d uses avg_deg 16 / seed 42, rmat edgefactor 8 / seed 42
(the graphs of profiles/round1/quality_*.json). A trailing "late"
selects the full late-split quality mode (split_c = n; minutes of
host-side bisection work at scale 22+ — the 0.44x-of-reference artifact
profiles/round1/quality_rmat23_k16_late_full.json).
"""

import ctypes
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import kaminpar_amd as ka  # noqa: E402
from oracle_pipeline import oracle_partition_deep  # noqa: E402


def main():
    kind = sys.argv[1]
    scale = int(sys.argv[2])
    k = int(sys.argv[3])
    out_path = sys.argv[4] if len(sys.argv) > 4 else None
    late = len(sys.argv) > 5 and sys.argv[5] == "late"

    oracle = ctypes.CDLL(os.path.join(REPO, "oracle", "liblp_oracle.so"))
    ref = ctypes.CDLL(os.path.join(REPO, "oracle", "_ref",
                                   "libkaminpar_ref_full.so"))
    u32p = ctypes.POINTER(ctypes.c_uint32)
    ref.kref_compute_partition.restype = ctypes.c_int64

    if kind == "rgg2d":
        g = ka.Graph.rgg2d(1 << scale, 16.0, seed=42)
    else:
        g = ka.Graph.rmat(scale, 8, seed=42)
    print(f"{kind}{scale} n={g.n} m={g.m} k={k}", flush=True)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    out = {"graph": f"{kind}{scale}_s42", "n": int(g.n), "m": int(g.m),
           "k": k, "reference": {}}
    for seed in (1, 2):
        part = np.zeros(g.n, np.uint32)
        t0 = time.time()
        cut = ref.kref_compute_partition(
            ctypes.c_uint32(g.n), ctypes.c_uint64(g.m),
            xadj.ctypes.data_as(u32p), adjncy.ctypes.data_as(u32p),
            None, None, ctypes.c_uint32(k), ctypes.c_double(0.03),
            ctypes.c_int(seed), part.ctypes.data_as(u32p))
        out["reference"][f"seed{seed}"] = {
            "cut": int(cut), "seconds": round(time.time() - t0, 1)}
        print(f"reference seed{seed}: cut={cut}", flush=True)

    t0 = time.time()
    cut, part, levels = oracle_partition_deep(
        oracle, g, k, seed=1, split_c=g.n if late else None)
    dt = time.time() - t0
    out["ours_deep_oracle"] = {
        "cut": int(cut), "seconds": round(dt, 1), "late": late,
        "note": "CPU oracle mirror, bit-identical to the GPU pipeline"}
    best = min(v["cut"] for v in out["reference"].values())
    out["deep_vs_reference_best"] = round(cut / best, 4)
    print(f"ours deep: cut={cut} ratio={out['deep_vs_reference_best']} "
          f"({dt:.0f}s)", flush=True)
    if out_path:
        with open(out_path, "w") as fh:
            json.dump(out, fh, indent=1)


if __name__ == "__main__":
    main()
