"""CPU-side guards for the multilevel pipeline:

- the oracle-mirrored pipeline (tests/oracle_pipeline.py, bit-identical to
  the GPU pipeline stage by stage) reproduces the committed expected cuts --
  catches drift between kaminpar_amd/partition.py's schedule and the
  goldens before any GPU time is spent;
- the initial partitioner alone is balanced and deterministic.
"""

import ctypes
import json
import os

import numpy as np
import pytest

import kaminpar_amd as ka
from kaminpar_amd.partition import initial_partition
from oracle_pipeline import oracle_partition

HERE = os.path.dirname(os.path.abspath(__file__))


def _load(name):
    return json.load(open(os.path.join(HERE, "golden", name)))


def _graph(name):
    if name.startswith("walshaw"):
        d = _load("walshaw_data.json")
        return ka.Graph.from_csr(np.array(d["xadj"], np.uint32),
                                 np.array(d["adjncy"], np.uint32))
    if name.startswith("rgg2d"):
        return ka.Graph.read_metis(os.path.join(HERE, "golden", "rgg2d.metis"))
    scale = int(name.split("_")[0][4:])
    return ka.Graph.rmat(scale, 8, 42)


@pytest.mark.parametrize("name", [
    "walshaw_k2", "walshaw_k16", "rgg2d_k4", "rmat14_s42_k16",
    "rmat16_s42_k16",
])
def test_oracle_pipeline_matches_expected(oracle, name):
    exp = _load("pipeline_expected.json")[name]
    g = _graph(name)
    cut, part, levels = oracle_partition(oracle, g, exp["k"], seed=1)
    assert cut == exp["cut"], (cut, exp["cut"])
    assert levels == exp["levels"]
    checksum = int(np.bitwise_xor.reduce(
        np.asarray(part, np.uint64) * np.arange(1, g.n + 1, dtype=np.uint64)))
    assert checksum == exp["part_checksum"]


@pytest.mark.parametrize("k", [2, 4, 16, 5])
def test_initial_partition_balanced_deterministic(k):
    g = ka.Graph.read_metis(os.path.join(HERE, "golden", "rgg2d.metis"))
    cap = g.max_block_weight(k, 0.03)
    p1 = initial_partition(g, k, cap, seed=1)
    p2 = initial_partition(g, k, cap, seed=1)
    assert np.array_equal(p1, p2)
    assert p1.max() < k
    counts = np.bincount(p1, minlength=k)
    assert counts.max() <= cap


def test_pipeline_band_vs_reference_goldens(oracle):
    """Our full-pipeline cut stays within the documented band of the
    compiled reference's deep-multilevel cut (best of 3 seeds)."""
    exp = _load("pipeline_expected.json")
    band = _load("ref_golden_partition.json")
    for name in ("walshaw_k16", "rmat14_s42_k16"):
        ref_best = min(band[name][f"seed{s}"]["cut"] for s in (1, 2, 3))
        assert exp[name]["cut"] <= 1.75 * ref_best, name


@pytest.mark.parametrize("case,k", [("walshaw", 2), ("walshaw", 16),
                                    ("rgg2d", 5), ("rmat12", 16)])
def test_native_initial_partition_matches_python(case, k):
    """The C++ initial partitioner (kmp_initial_partition, used by the
    product pipeline and the C-ABI driver) is bit-identical to the numpy
    reference implementation."""
    if case == "walshaw":
        g = _graph("walshaw_k2")
    elif case == "rgg2d":
        g = _graph("rgg2d_k4")
    else:
        g = ka.Graph.rmat(12, 8, 42)
    mbw = g.max_block_weight(k, 0.03)
    py = initial_partition(g, k, mbw, seed=1, reps=8)
    cc = g.initial_partition_native(k, mbw, reps=8)
    assert np.array_equal(py, cc)


@pytest.mark.parametrize("name", [
    "walshaw_k2", "walshaw_k16", "rgg2d_k4", "rmat14_s42_k16",
])
def test_oracle_deep_pipeline_matches_expected(oracle, name):
    """The progressive-k (deep) pipeline mirror reproduces the committed
    expected cuts."""
    from oracle_pipeline import oracle_partition_deep

    exp = _load("pipeline_deep_expected.json")[name]
    g = _graph(name)
    cut, part, levels = oracle_partition_deep(oracle, g, exp["k"], seed=1)
    assert cut == exp["cut"], (cut, exp["cut"])
    assert levels == exp["levels"]
    checksum = int(np.bitwise_xor.reduce(
        np.asarray(part, np.uint64) * np.arange(1, g.n + 1, dtype=np.uint64)))
    assert checksum == exp["part_checksum"]


def test_deep_pipeline_band_vs_reference_goldens():
    """The deep pipeline's cuts stay within a band of the compiled
    reference's deep-multilevel cuts (<=1.30x the best of 3 seeds on these
    cases; measured 0.74-1.25 with the split-schedule dispatch: heavy-tail
    graphs defer splits to the finest level and BEAT the reference on all
    four R-MAT cases)."""
    exp = _load("pipeline_deep_expected.json")
    band = _load("ref_golden_partition.json")
    for name in ("walshaw_k2", "walshaw_k16", "rgg2d_k4", "rmat14_s42_k16",
                 "rmat16_s42_k16", "rmat18_s42_k16", "rmat18_s42_k64"):
        ref_best = min(band[name][f"seed{s}"]["cut"] for s in (1, 2, 3))
        assert exp[name]["cut"] <= 1.30 * ref_best, (name, exp[name]["cut"], ref_best)
    # heavy-tail cases must stay at or below the reference's best seed
    for name in ("rmat14_s42_k16", "rmat16_s42_k16", "rmat18_s42_k16",
                 "rmat18_s42_k64"):
        ref_best = min(band[name][f"seed{s}"]["cut"] for s in (1, 2, 3))
        assert exp[name]["cut"] <= ref_best, (name, exp[name]["cut"], ref_best)


def test_native_extend_partition_matches_python():
    """kmp_extend_partition (used by the C deep driver) is bit-identical to
    the Python _extend_partition."""
    import ctypes
    from kaminpar_amd import _lib, _u32p
    from kaminpar_amd.partition import _extend_partition

    _lib.kmp_extend_partition.restype = ctypes.c_int
    _lib.kmp_extend_partition.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32), ctypes.c_uint32,
        ctypes.c_int64, ctypes.c_uint32, ctypes.c_int, ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32)]

    # cases cross the bisector-dispatch thresholds (256-vertex O(n^2)
    # cutoff, heavy/light-tail CV^2 split incl. the HEM path on mesh
    # subgraphs) and the parallel per-group path (many groups at once,
    # with w<2 pass-through groups at k=23)
    def weighted_rmat(scale):
        base = ka.Graph.rmat(scale, 8, 42)
        rng = np.random.default_rng(17)
        vwgt = rng.integers(1, 12, base.n).astype(np.int32)
        # symmetric arc weights keyed on the endpoint pair
        xadj = np.asarray(base.xadj).astype(np.int64)
        adjncy = np.asarray(base.adjncy).astype(np.int64)
        src = np.repeat(np.arange(base.n, dtype=np.int64), np.diff(xadj))
        lo = np.minimum(src, adjncy)
        hi = np.maximum(src, adjncy)
        adjwgt = (1 + (lo * 31 + hi * 7) % 9).astype(np.int32)
        return ka.Graph.from_csr(np.asarray(base.xadj).copy(),
                                 np.asarray(base.adjncy).copy(),
                                 vwgt=vwgt, adjwgt=adjwgt)

    for graph, k, split_c, force in (("rmat12", 16, 48, 1),
                                     ("rmat12", 8, 200, 0),
                                     ("rmat14", 23, 1, 1),
                                     ("rmat13", 64, 16, 0),
                                     ("rgg16k", 16, 1, 1),
                                     ("rgg16k", 7, 300, 0),
                                     ("wrmat12", 16, 1, 1),
                                     ("wrgg8k", 8, 1, 1),  # weighted HEM path
                                     ("rmat18", 2, 1, 1)):  # ns > 131072
        if graph == "rgg16k":
            g = ka.Graph.rgg2d(16384, 8, seed=5)
        elif graph == "wrmat12":
            g = weighted_rmat(12)
        elif graph == "wrgg8k":
            base = ka.Graph.rgg2d(8192, 8, seed=5)
            rng = np.random.default_rng(23)
            vwgt = rng.integers(1, 9, base.n).astype(np.int32)
            xadj = np.asarray(base.xadj).astype(np.int64)
            adjncy = np.asarray(base.adjncy).astype(np.int64)
            src = np.repeat(np.arange(base.n, dtype=np.int64),
                            np.diff(xadj))
            lo = np.minimum(src, adjncy)
            hi = np.maximum(src, adjncy)
            adjwgt = (1 + (lo * 13 + hi * 5) % 7).astype(np.int32)
            g = ka.Graph.from_csr(np.asarray(base.xadj).copy(),
                                  np.asarray(base.adjncy).copy(),
                                  vwgt=vwgt, adjwgt=adjwgt)
        else:
            g = ka.Graph.rmat(int(graph[4:]), 8, 42)
        mbw = g.max_block_weight(k, 0.03)
        part_py = np.zeros(g.n, np.uint32)
        part_py, groups_py = _extend_partition(
            g, part_py, [(0, k)], mbw, k, split_c=split_c, reps=8,
            force=bool(force))
        part_c = np.zeros(g.n, np.uint32)
        lo = np.zeros(k, np.uint32)
        w = np.zeros(k, np.uint32)
        w[0] = k
        num = ctypes.c_uint32(1)
        _lib.kmp_extend_partition(g._h, _u32p(part_c), k, mbw, split_c, 8,
                                  force, _u32p(lo), _u32p(w),
                                  ctypes.byref(num))
        assert np.array_equal(part_py, part_c)
        assert groups_py == [(int(lo[i]), int(w[i])) for i in range(num.value)]


@pytest.mark.parametrize("k", [2, 8, 16])
def test_kway_fm_improves_respects_caps_deterministic(k):
    """kmp_kway_fm: lowers the cut from a random partition, never violates
    the per-block caps (0-cap blocks never receive), and is deterministic."""
    rng = np.random.default_rng(11)
    g = ka.Graph.rgg2d(4096, 8, seed=9)
    vw = np.ones(g.n, np.int64)
    part0 = rng.integers(0, k, g.n).astype(np.uint32)
    cut0 = g.edge_cut(part0)
    caps = np.full(k, g.max_block_weight(k, 0.03), np.int64)
    if k > 2:
        caps[k - 1] = 0  # closed block: must only lose vertices
    p1 = g.kway_fm(k, caps, part0.copy())
    p2 = g.kway_fm(k, caps, part0.copy())
    assert np.array_equal(p1, p2)
    cut1 = g.edge_cut(p1)
    assert cut1 < cut0
    bw = np.zeros(k, np.int64)
    np.add.at(bw, p1, vw)
    bw0 = np.zeros(k, np.int64)
    np.add.at(bw0, part0, vw)
    for b in range(k):
        assert bw[b] <= max(caps[b], bw0[b])
    if k > 2:
        assert bw[k - 1] <= bw0[k - 1]


def test_extend_partition_fuzz_python_vs_native():
    """Seeded fuzz: Python and C extension drivers stay bit-identical on
    random graphs across random k / split_c / force combinations."""
    import ctypes
    from kaminpar_amd import _lib, _u32p
    from kaminpar_amd.partition import _extend_partition

    _lib.kmp_extend_partition.restype = ctypes.c_int
    _lib.kmp_extend_partition.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32), ctypes.c_uint32,
        ctypes.c_int64, ctypes.c_uint32, ctypes.c_int, ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32)]

    rng = np.random.default_rng(99)
    for trial in range(6):
        if trial % 2:
            g = ka.Graph.rmat(int(rng.integers(10, 13)), 8,
                              seed=int(rng.integers(1, 1000)))
        else:
            g = ka.Graph.rgg2d(int(rng.integers(512, 6000)), 8,
                               seed=int(rng.integers(1, 1000)))
        k = int(rng.integers(2, 33))
        split_c = int(rng.integers(1, 500))
        force = int(rng.integers(0, 2))
        mbw = g.max_block_weight(k, 0.03)
        part_py = np.zeros(g.n, np.uint32)
        part_py, groups_py = _extend_partition(
            g, part_py, [(0, k)], mbw, k, split_c=split_c, reps=8,
            force=bool(force))
        part_c = np.zeros(g.n, np.uint32)
        lo = np.zeros(k, np.uint32)
        w = np.zeros(k, np.uint32)
        w[0] = k
        num = ctypes.c_uint32(1)
        _lib.kmp_extend_partition(g._h, _u32p(part_c), k, mbw, split_c, 8,
                                  force, _u32p(lo), _u32p(w),
                                  ctypes.byref(num))
        assert np.array_equal(part_py, part_c), (trial, k, split_c, force)
        assert groups_py == [(int(lo[i]), int(w[i]))
                             for i in range(num.value)], trial


def test_kway_fm_never_worsens_fuzz():
    """Seeded fuzz: the pass-level best-prefix rollback guarantees the cut
    never increases, from arbitrary (also infeasible) starts."""
    rng = np.random.default_rng(123)
    for trial in range(8):
        n = int(rng.integers(256, 4096))
        g = (ka.Graph.rgg2d(n, 8, seed=int(rng.integers(1, 1000)))
             if trial % 2 else
             ka.Graph.rmat(int(rng.integers(9, 12)), 8,
                           seed=int(rng.integers(1, 1000))))
        k = int(rng.integers(2, 17))
        part = rng.integers(0, k, g.n).astype(np.uint32)
        caps = np.full(k, g.max_block_weight(k, 0.03), np.int64)
        cut0 = g.edge_cut(part)
        out = g.kway_fm(k, caps, part.copy())
        assert g.edge_cut(out) <= cut0, (trial, k)
