"""Generate golden FULL-PIPELINE cuts from the compiled reference partitioner
(oracle/_ref/libkaminpar_ref_full.so = the reference's own
KaMinPar::compute_partition built from /root/reference with serial TBB stubs,
deterministic at a fixed seed; see oracle/_ref_build/Makefile.full).

Run in the dev container (where /root/reference is mounted and the full ref
lib can be built):
    make -C oracle/_ref_build -f Makefile.full
    python tests/golden/generate_golden_partition.py
Writes tests/golden/ref_golden_partition.json: for each (graph, k, seed) the
reference's final edge cut and max block weight. The GPU pipeline test
(test_gpu_parity.py::test_partition_pipeline_*) compares our multilevel
pipeline's cut against these within a documented band.
"""

import ctypes
import json
import os

import numpy as np

import kaminpar_amd as ka
from kaminpar_amd import _lib

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))


def load_ref():
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "_ref", "libkaminpar_ref_full.so"))
    u32p = ctypes.POINTER(ctypes.c_uint32)
    i32p = ctypes.POINTER(ctypes.c_int32)
    lib.kref_compute_partition.restype = ctypes.c_int64
    lib.kref_compute_partition.argtypes = [
        ctypes.c_uint32, ctypes.c_uint64, u32p, u32p, i32p, i32p,
        ctypes.c_uint32, ctypes.c_double, ctypes.c_int, u32p,
    ]
    return lib


def ref_partition(lib, g, k, eps, seed):
    n, m = g.n, g.m
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    part = np.zeros(n, dtype=np.uint32)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    cut = lib.kref_compute_partition(
        n, m,
        xadj.ctypes.data_as(u32p), adjncy.ctypes.data_as(u32p),
        None, None, k, eps, seed,
        part.ctypes.data_as(u32p),
    )
    counts = np.bincount(part, minlength=k)
    # cross-check the reported cut
    assert cut == g.edge_cut(part), (cut, g.edge_cut(part))
    return int(cut), int(counts.max())


def main():
    lib = load_ref()
    out = {}

    def walshaw():
        d = json.load(open(os.path.join(HERE, "walshaw_data.json")))
        return ka.Graph.from_csr(np.array(d["xadj"], np.uint32),
                                 np.array(d["adjncy"], np.uint32))

    cases = [
        ("walshaw", walshaw, [2, 16]),
        ("rgg2d", lambda: ka.Graph.read_metis(os.path.join(HERE, "rgg2d.metis")), [4]),
        ("rmat14_s42", lambda: ka.Graph.rmat(14, 8, 42), [16]),
        ("rmat16_s42", lambda: ka.Graph.rmat(16, 8, 42), [16]),
        ("rmat18_s42", lambda: ka.Graph.rmat(18, 8, 42), [16, 64]),
    ]
    for name, mk, ks in cases:
        g = mk()
        for k in ks:
            entry = {}
            for seed in (1, 2, 3):
                cut, maxb = ref_partition(lib, g, k, 0.03, seed)
                entry[f"seed{seed}"] = {"cut": cut, "max_block": maxb}
            mbw = g.max_block_weight(k, 0.03)
            out[f"{name}_k{k}"] = {
                "n": g.n, "m": g.m, "k": k, "eps": 0.03,
                "cap": int(mbw), **entry,
            }
            print(name, k, out[f"{name}_k{k}"])

    with open(os.path.join(HERE, "ref_golden_partition.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
