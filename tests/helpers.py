"""Shared test helpers: ctypes call wrappers for the oracle / reference."""

import ctypes

import numpy as np


def u32p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))


def i32p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))


def i64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


def u64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))


def oracle_refine(oracle, g, k, maxw, part, seed=1, iters=5, vwgt=None, adjwgt=None):
    """Run the CPU oracle's deterministic LP refinement. Returns (cut, part)."""
    part = np.ascontiguousarray(part, dtype=np.uint32).copy()
    maxw = np.ascontiguousarray(maxw, dtype=np.int64)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    cut = oracle.kmp_oracle_lp_refine(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        ctypes.c_uint32(k), i64p(maxw), u32p(part),
        ctypes.c_uint64(seed), ctypes.c_int(iters), u64p(stats),
    )
    return cut, part, stats


def oracle_cluster(oracle, g, max_w, desired=0, seed=1, iters=5, vwgt=None, adjwgt=None):
    clus = np.zeros(g.n, dtype=np.uint32)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    nc = oracle.kmp_oracle_lp_cluster(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        ctypes.c_int64(max_w), ctypes.c_uint32(desired), u32p(clus),
        ctypes.c_uint64(seed), ctypes.c_int(iters), u64p(stats),
    )
    return nc, clus, stats


def ref_refine(ref, g, k, eps, part, seed=1, iters=5):
    part = np.ascontiguousarray(part, dtype=np.uint32).copy()
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    cut = ref.kref_lp_refine(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy), None, None,
        ctypes.c_uint32(k), ctypes.c_double(eps), ctypes.c_int(seed), ctypes.c_int(iters),
        u32p(part),
    )
    return cut, part


def ref_cluster(ref, g, max_w, seed=1, iters=5, desired=0):
    clus = np.zeros(g.n, dtype=np.uint32)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    ref.kref_lp_cluster(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy), None, None,
        ctypes.c_int(seed), ctypes.c_int(iters), ctypes.c_int64(max_w),
        ctypes.c_uint32(desired), u32p(clus),
    )
    return clus


def oracle_underload(oracle, g, k, maxw, minw, part, seed=1, iters=5,
                     vwgt=None, adjwgt=None):
    """Run the CPU oracle's underload-balancer LP. Returns (cut, part, stats)."""
    import ctypes
    import numpy as np
    part = np.ascontiguousarray(part, dtype=np.uint32).copy()
    maxw = np.ascontiguousarray(maxw, dtype=np.int64)
    minw = np.ascontiguousarray(minw, dtype=np.int64)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    oracle.kmp_oracle_lp_underload.restype = ctypes.c_int64
    cut = oracle.kmp_oracle_lp_underload(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        ctypes.c_uint32(k), i64p(maxw), i64p(minw), u32p(part),
        ctypes.c_uint64(seed), ctypes.c_int(iters), u64p(stats),
    )
    return cut, part, stats


def oracle_cluster_comm(oracle, g, max_w, communities, seed=1, iters=5,
                        vwgt=None, adjwgt=None, desired=0):
    """Oracle clustering with Clusterer::set_communities semantics."""
    import ctypes
    import numpy as np
    clus = np.zeros(g.n, dtype=np.uint32)
    comm = np.ascontiguousarray(communities, dtype=np.uint32)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    oracle.kmp_oracle_lp_cluster_comm.restype = ctypes.c_int64
    nc = oracle.kmp_oracle_lp_cluster_comm(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        ctypes.c_int64(int(max_w)), ctypes.c_uint32(desired), u32p(comm),
        u32p(clus), ctypes.c_uint64(seed), ctypes.c_int(iters), u64p(stats),
    )
    return nc, clus, stats


def oracle_balance(oracle, g, k, maxw, part, seed=1, iters=5, vwgt=None, adjwgt=None):
    """Run the CPU oracle's balancer-mode LP. Returns (cut, part, stats)."""
    part = np.ascontiguousarray(part, dtype=np.uint32).copy()
    maxw = np.ascontiguousarray(maxw, dtype=np.int64)
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    oracle.kmp_oracle_lp_balance.restype = ctypes.c_int64
    cut = oracle.kmp_oracle_lp_balance(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        ctypes.c_uint32(k), i64p(maxw), u32p(part),
        ctypes.c_uint64(seed), ctypes.c_int(iters), u64p(stats),
    )
    return cut, part, stats
