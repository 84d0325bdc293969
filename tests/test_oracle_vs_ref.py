"""Pin the oracle against the compiled reference (oracle/_ref).

The reference's asynchronous chunk-randomized LP cannot be reproduced
bit-for-bit by a parallel schedule (no reference test
pins exact LP output either), so the oracle is pinned at the level the
reference itself tests, plus quality equivalence over seeds:
  - refinement: oracle cuts lie within a few percent of the reference's cut
    band on the same inputs across seeds; caps identical;
  - clustering: cluster-count bands overlap;
  - golden fixtures (tests/golden) pin the reference outputs themselves so a
    stub/toolchain drift in oracle/_ref is detected.
"""

import json
import os

import numpy as np
import pytest

import kaminpar_amd as ka
from helpers import oracle_cluster, oracle_refine, ref_cluster, ref_refine

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


@pytest.fixture(scope="module")
def graphs():
    return {
        "rmat12": ka.Graph.rmat(12, 8, seed=7),
        "rgg4k": ka.Graph.rgg2d(4096, 16.0, seed=3),
    }


def _need_ref(ref):
    if ref is None:
        pytest.skip("oracle/_ref not built on this box (built in the dev container)")


@pytest.mark.parametrize("name,k", [("rmat12", 8), ("rgg4k", 4)])
def test_refine_quality_band(oracle, ref, graphs, name, k):
    _need_ref(ref)
    g = graphs[name]
    part0 = ka.random_partition(g.n, k, seed=5)
    eps = 0.03
    mbw_val = ref.kref_max_block_weight(
        g.n, g.m,
        g.xadj.ctypes.data_as(__import__("ctypes").POINTER(__import__("ctypes").c_uint32)),
        g.adjncy.ctypes.data_as(__import__("ctypes").POINTER(__import__("ctypes").c_uint32)),
        None, None, k, __import__("ctypes").c_double(eps),
    )
    # our own max-block-weight computation must agree with the reference's
    assert mbw_val == g.max_block_weight(k, eps)
    mbw = np.full(k, mbw_val, dtype=np.int64)

    ref_cuts = [ref_refine(ref, g, k, eps, part0, seed=s)[0] for s in range(6)]
    orc_cuts = [oracle_refine(oracle, g, k, mbw, part0, seed=s)[0] for s in range(6)]

    ref_med = float(np.median(ref_cuts))
    orc_med = float(np.median(orc_cuts))
    # Quality parity band: medians within 8%. Measured (10 seeds): ~3% on
    # power-law R-MAT (k=8), ~6.6% on spatial RGG (k=4) -- the deterministic
    # chunk-synchronous schedule decides on snapshot gains, so simultaneous
    # boundary moves churn slightly where the fully asynchronous reference
    # reacts instantly; converged (iters->inf) the gap is ~3.2%. Tracked as a
    # quality-improvement item in DESIGN.md.
    assert abs(orc_med - ref_med) / ref_med < 0.08, (ref_cuts, orc_cuts)


def test_cluster_quality_band(oracle, ref, graphs):
    _need_ref(ref)
    g = graphs["rmat12"]
    max_w = 32
    ref_counts = [len(np.unique(ref_cluster(ref, g, max_w, seed=s))) for s in range(6)]
    orc_counts = [oracle_cluster(oracle, g, max_w, seed=s)[0] for s in range(6)]
    ref_med = float(np.median(ref_counts))
    orc_med = float(np.median(orc_counts))
    assert abs(orc_med - ref_med) / ref_med < 0.15, (ref_counts, orc_counts)


def test_golden_reference_outputs(ref, oracle):
    """The compiled reference reproduces the committed golden outputs, and the
    oracle stays within the recorded quality bands (regression pin for both
    the TBB-stub build and the oracle)."""
    path = os.path.join(GOLDEN, "ref_golden.json")
    assert os.path.exists(path), "run tests/golden/generate_golden.py first"
    with open(path) as f:
        golden = json.load(f)

    for case in golden["refine"]:
        g = ka.Graph.rmat(case["scale"], case["edgefactor"], seed=case["gseed"])
        part0 = ka.random_partition(g.n, case["k"], seed=case["pseed"])
        mbw = np.full(case["k"], case["mbw"], dtype=np.int64)
        if ref is not None:
            cut, part = ref_refine(ref, g, case["k"], case["eps"], part0, seed=case["seed"])
            assert cut == case["ref_cut"], "compiled reference drifted from golden"
            assert int(np.asarray(part, dtype=np.int64).sum()) == case["ref_part_sum"]
        ocut, opart, _ = oracle_refine(oracle, g, case["k"], mbw, part0, seed=case["seed"])
        assert ocut == case["oracle_cut"], "oracle output drifted from golden"
        assert int(np.asarray(opart, dtype=np.int64).sum()) == case["oracle_part_sum"]

    for case in golden["cluster"]:
        g = ka.Graph.rmat(case["scale"], case["edgefactor"], seed=case["gseed"])
        if ref is not None:
            clus = ref_cluster(ref, g, case["max_w"], seed=case["seed"])
            assert len(np.unique(clus)) == case["ref_nc"]
        nc, oclus, _ = oracle_cluster(oracle, g, case["max_w"], seed=case["seed"])
        assert nc == case["oracle_nc"]
        assert int(np.asarray(oclus, dtype=np.int64).sum()) == case["oracle_clus_sum"]
