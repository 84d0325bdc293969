"""Generate golden fixtures pinning the compiled reference (oracle/_ref) and
the oracle outputs on small deterministic graphs.

Run in the dev container (where /root/reference and oracle/_ref exist):
    python tests/golden/generate_golden.py
Commits: tests/golden/ref_golden.json
"""

import ctypes
import json
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import kaminpar_amd as ka  # noqa: E402
from helpers import oracle_cluster, oracle_refine, ref_cluster, ref_refine  # noqa: E402


def main():
    ref = ctypes.CDLL(os.path.join(REPO, "oracle", "_ref", "libkaminpar_ref.so"))
    ref.kref_lp_refine.restype = ctypes.c_int64
    ref.kref_lp_cluster.restype = ctypes.c_int64
    ref.kref_max_block_weight.restype = ctypes.c_int64
    oracle = ctypes.CDLL(os.path.join(REPO, "oracle", "liblp_oracle.so"))
    oracle.kmp_oracle_lp_refine.restype = ctypes.c_int64
    oracle.kmp_oracle_lp_cluster.restype = ctypes.c_int64

    out = {"refine": [], "cluster": []}

    for scale, ef, gseed, k, pseed, seed in [
        (10, 8, 7, 8, 5, 1),
        (12, 8, 7, 16, 5, 2),
        (12, 8, 11, 64, 3, 3),
    ]:
        g = ka.Graph.rmat(scale, ef, seed=gseed)
        eps = 0.03
        mbw_val = g.max_block_weight(k, eps)
        part0 = ka.random_partition(g.n, k, seed=pseed)
        cut, part = ref_refine(ref, g, k, eps, part0, seed=seed)
        ocut, opart, _ = oracle_refine(
            oracle, g, k, np.full(k, mbw_val, dtype=np.int64), part0, seed=seed
        )
        out["refine"].append({
            "scale": scale, "edgefactor": ef, "gseed": gseed, "k": k,
            "pseed": pseed, "seed": seed, "eps": eps, "mbw": int(mbw_val),
            "ref_cut": int(cut), "ref_part_sum": int(np.asarray(part, np.int64).sum()),
            "oracle_cut": int(ocut),
            "oracle_part_sum": int(np.asarray(opart, np.int64).sum()),
        })

    for scale, ef, gseed, max_w, seed in [(10, 8, 7, 16, 1), (12, 8, 7, 32, 2)]:
        g = ka.Graph.rmat(scale, ef, seed=gseed)
        clus = ref_cluster(ref, g, max_w, seed=seed)
        nc, oclus, _ = oracle_cluster(oracle, g, max_w, seed=seed)
        out["cluster"].append({
            "scale": scale, "edgefactor": ef, "gseed": gseed, "max_w": max_w,
            "seed": seed, "ref_nc": int(len(np.unique(clus))),
            "oracle_nc": int(nc),
            "oracle_clus_sum": int(np.asarray(oclus, np.int64).sum()),
        })

    path = os.path.join(REPO, "tests", "golden", "ref_golden.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {path}")
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
