"""Generate exact expected cuts of the PROGRESSIVE-K (deep) pipeline via the
oracle mirror (tests/oracle_pipeline.py::oracle_partition_deep) -- every
stage bit-reproducible, so the GPU pipeline must match exactly.

Run here (CPU): python tests/golden/generate_pipeline_deep_expected.py
Writes tests/golden/pipeline_deep_expected.json.
"""

import ctypes
import json
import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import kaminpar_amd as ka
from oracle_pipeline import oracle_partition_deep


def main():
    oracle = ctypes.CDLL(os.path.join(REPO, "oracle", "liblp_oracle.so"))

    def walshaw():
        d = json.load(open(os.path.join(HERE, "walshaw_data.json")))
        return ka.Graph.from_csr(np.array(d["xadj"], np.uint32),
                                 np.array(d["adjncy"], np.uint32))

    cases = [
        ("walshaw_k2", walshaw, 2),
        ("walshaw_k16", walshaw, 16),
        ("rgg2d_k4", lambda: ka.Graph.read_metis(os.path.join(HERE, "rgg2d.metis")), 4),
        ("rmat14_s42_k16", lambda: ka.Graph.rmat(14, 8, 42), 16),
        ("rmat16_s42_k16", lambda: ka.Graph.rmat(16, 8, 42), 16),
        ("rmat18_s42_k16", lambda: ka.Graph.rmat(18, 8, 42), 16),
        ("rmat18_s42_k64", lambda: ka.Graph.rmat(18, 8, 42), 64),
    ]
    out = {}
    for name, mk, k in cases:
        g = mk()
        cut, part, levels = oracle_partition_deep(oracle, g, k, seed=1)
        out[name] = {"k": k, "cut": int(cut), "levels": levels,
                     "part_checksum": int(np.bitwise_xor.reduce(
                         np.asarray(part, np.uint64) * np.arange(1, g.n + 1, dtype=np.uint64)))}
        print(name, out[name])
    with open(os.path.join(HERE, "pipeline_deep_expected.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
