"""Time partition_deep with default (coarse) vs fine-level splits at
benchmark scale, with the per-group bisections parallelized over host
cores. Decides whether fine-level splits can be the auto default beyond
2M vertices.

Usage (GPU box): python tools/time_deep_splits.py [scale] [k] [out.json]
"""

import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import kaminpar_amd as ka  # noqa: E402
from kaminpar_amd.partition import partition_deep  # noqa: E402


def main():
    scale = int(sys.argv[1]) if len(sys.argv) > 1 else 23
    k = int(sys.argv[2]) if len(sys.argv) > 2 else 16
    out_path = sys.argv[3] if len(sys.argv) > 3 else None

    g = ka.Graph.rmat(scale, 8, seed=42)
    print(f"rmat{scale} n={g.n} m={g.m} k={k} "
          f"cores={os.cpu_count()}", flush=True)
    out = {"scale": scale, "k": k, "n": int(g.n), "m": int(g.m),
           "cores": os.cpu_count()}

    for name, sc in (("deep_auto", None), ("deep_fine", 262144)):
        t0 = time.time()
        cut, part, levels = partition_deep(g, k, seed=1, split_c=sc)
        dt = time.time() - t0
        maxb = int(np.bincount(part, minlength=k).max())
        out[name] = {"cut": int(cut), "max_block": maxb,
                     "seconds": round(dt, 2), "split_c": sc}
        print(f"{name}: cut={cut} max_block={maxb} ({dt:.2f}s)", flush=True)

    if out["deep_fine"]["cut"] and out["deep_auto"]["cut"]:
        out["fine_vs_auto_cut"] = round(
            out["deep_fine"]["cut"] / out["deep_auto"]["cut"], 4)
    if out_path:
        with open(out_path, "w") as fh:
            json.dump(out, fh, indent=1)


if __name__ == "__main__":
    main()
