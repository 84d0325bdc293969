"""Cluster contraction: the oracle restatement is bit-identical to the
reference implementation (contract_clustering, canonically sorted), and
invariants hold (node weight conserved, inter-cluster edge weight
conserved, coarse graph simple + symmetric)."""

import ctypes

import numpy as np
import pytest

import kaminpar_amd as ka
from helpers import i32p, oracle_cluster, u32p, u64p


def _contract(fn, g, clus):
    n, m = g.n, g.m
    xadj = np.ascontiguousarray(g.xadj)
    adjncy = np.ascontiguousarray(g.adjncy)
    mapping = np.zeros(n, np.uint32)
    c_xadj = np.zeros(n + 1, np.uint32)
    c_adj = np.zeros(m, np.uint32)
    c_vw = np.zeros(n, np.int32)
    c_wg = np.zeros(m, np.int32)
    c_m = np.zeros(1, np.uint64)
    c_n = fn(ctypes.c_uint32(n), ctypes.c_uint64(m), u32p(xadj), u32p(adjncy), None, None,
             u32p(clus), u32p(mapping), u32p(c_xadj), u32p(c_adj), i32p(c_vw), i32p(c_wg),
             u64p(c_m))
    cm = int(c_m[0])
    return (c_n, cm, mapping, c_xadj[: c_n + 1].copy(), c_adj[:cm].copy(),
            c_vw[:c_n].copy(), c_wg[:cm].copy())


@pytest.mark.parametrize("scale,max_w,seed", [(10, 16, 1), (12, 32, 1), (13, 128, 3)])
def test_oracle_contract_vs_reference(oracle, ref, scale, max_w, seed):
    oracle.kmp_oracle_contract.restype = ctypes.c_int64
    g = ka.Graph.rmat(scale, 8, seed=7)
    nc, clus, _ = oracle_cluster(oracle, g, max_w, seed=seed)
    a = _contract(oracle.kmp_oracle_contract, g, clus)
    assert a[0] == nc

    # invariants
    c_n, cm, mapping, c_xadj, c_adj, c_vw, c_wg = a
    assert int(c_vw.sum()) == g.n  # unit fine weights conserved
    # inter-cluster fine edge weight equals coarse total edge weight
    fine_u = np.repeat(np.arange(g.n), np.diff(np.asarray(g.xadj)))
    inter = int((mapping[fine_u] != mapping[np.asarray(g.adjncy)]).sum())
    assert int(cm and c_wg.sum()) == inter
    # simple symmetric coarse graph
    cu = np.repeat(np.arange(c_n), np.diff(c_xadj))
    pairs = set(zip(cu.tolist(), c_adj.tolist()))
    assert len(pairs) == cm and all((v, u) in pairs for (u, v) in pairs)
    assert not any(u == v for (u, v) in pairs)

    if ref is not None:
        ref.kref_contract.restype = ctypes.c_int64
        b = _contract(ref.kref_contract, g, clus)
        assert a[0] == b[0] and a[1] == b[1]
        for x, y in zip(a[2:], b[2:]):
            assert np.array_equal(x, y)
