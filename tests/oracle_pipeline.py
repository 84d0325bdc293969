"""CPU mirror of kaminpar_amd.partition.partition built on the oracle's
cluster/contract/refine (each bit-identical to the GPU engine by the parity
tests), with the SAME level schedule, seeds and cap formulas. Because every
stage is bit-reproducible, the cut this pipeline produces is the exact cut
the GPU pipeline must produce -- used to (a) calibrate the golden band
against the compiled reference and (b) pin the GPU pipeline bit-exactly."""

import ctypes

import numpy as np

import kaminpar_amd as ka
from kaminpar_amd.partition import initial_partition, level_cluster_weight
from helpers import i32p, i64p, oracle_cluster, oracle_refine, u32p, u64p


def oracle_contract(oracle, g, clus, vwgt=None, adjwgt=None):
    oracle.kmp_oracle_contract.restype = ctypes.c_int64
    n, m = g.n, g.m
    xadj = np.ascontiguousarray(g.xadj)
    adjncy = np.ascontiguousarray(g.adjncy)
    mapping = np.zeros(n, np.uint32)
    c_xadj = np.zeros(n + 1, np.uint32)
    c_adj = np.zeros(m, np.uint32)
    c_vw = np.zeros(n, np.int32)
    c_wg = np.zeros(m, np.int32)
    c_m = np.zeros(1, np.uint64)
    c_n = oracle.kmp_oracle_contract(
        ctypes.c_uint32(n), ctypes.c_uint64(m), u32p(xadj), u32p(adjncy),
        i32p(vwgt) if vwgt is not None else None,
        i32p(adjwgt) if adjwgt is not None else None,
        u32p(clus), u32p(mapping), u32p(c_xadj), u32p(c_adj), i32p(c_vw),
        i32p(c_wg), u64p(c_m))
    cm = int(c_m[0])
    coarse = ka.Graph.from_csr(c_xadj[: c_n + 1].copy(), c_adj[:cm].copy(),
                               vwgt=c_vw[:c_n].copy(), adjwgt=c_wg[:cm].copy())
    return coarse, mapping


def oracle_partition(oracle, g, k, eps=0.03, seed=1, iters=5,
                     contraction_limit=2000, stop_n=512):
    """Same schedule as kaminpar_amd.partition.partition (keep in sync)."""
    total_w = g.total_node_weight
    mbw_val = g.max_block_weight(k, eps)
    mbw = np.full(k, mbw_val, dtype=np.int64)

    def weights(gr):
        from kaminpar_amd import _lib
        vw = _lib.kmp_graph_vwgt(gr._h)
        aw = _lib.kmp_graph_adjwgt(gr._h)
        v = np.ctypeslib.as_array(vw, shape=(gr.n,)) if vw else None
        a = np.ctypeslib.as_array(aw, shape=(gr.m,)) if aw else None
        return v, a

    graphs = [g]
    mappings = []
    while graphs[-1].n > max(stop_n, 2 * k):
        cur = graphs[-1]
        mcw = level_cluster_weight(total_w, cur.n, k, eps, contraction_limit)
        vw, aw = weights(cur)
        nc, clus, _ = oracle_cluster(oracle, cur, mcw, seed=seed + len(mappings),
                                     iters=iters, vwgt=vw, adjwgt=aw)
        coarse, mapping = oracle_contract(oracle, cur, clus, vwgt=vw, adjwgt=aw)
        if coarse.n > 0.95 * cur.n:
            break
        graphs.append(coarse)
        mappings.append(mapping)

    part = initial_partition(graphs[-1], k, mbw_val, seed=seed)

    cut = None
    fm_on = g.n <= (1 << 21)
    for level in range(len(graphs) - 1, -1, -1):
        gr = graphs[level]
        vw, aw = weights(gr)
        cut, part, _ = oracle_refine(oracle, gr, k, mbw, part, seed=seed,
                                     iters=iters, vwgt=vw, adjwgt=aw)
        # per-level k-way boundary FM (keep in sync with partition())
        if fm_on:
            part = gr.kway_fm(k, mbw, part)
            if level == 0:
                cut = g.edge_cut(part)
        if level > 0:
            part = part[mappings[level - 1]]
    return cut, part, [gr.n for gr in graphs]


def oracle_partition_deep(oracle, g, k, eps=0.03, seed=1, iters=5,
                          contraction_limit=2000, stop_n=512, split_c=None,
                          reps=8):
    """CPU mirror of kaminpar_amd.partition.partition_deep (keep in sync):
    progressive-k extension by FM-polished block bisections during
    uncoarsening."""
    from kaminpar_amd.partition import (_extend_partition, _group_caps,
                                        level_cluster_weight)

    if split_c is None:
        split_c = 262144 if g.n <= (1 << 21) else 2000
    # split-schedule dispatch by degree variance (keep in sync with
    # partition_deep): heavy-tailed fine graphs defer all splits to the
    # finest level (no eager coarsest split)
    xadj = np.asarray(g.xadj, dtype=np.int64)
    d = xadj[1:] - xadj[:-1]
    # exact integer sum-of-squares without the object-dtype blowup (an
    # object sum over 67M degrees measured ~1.3 s per partition): split the
    # int64 products into high/low 32-bit halves and recombine as Python
    # ints -- bit-exact, vectorized
    sq = d * d  # per-element fits int64 (deg < 2^31)
    ssum = (int((sq & 0xFFFFFFFF).sum(dtype=np.int64))
            + (int((sq >> 32).sum(dtype=np.int64)) << 32))
    heavy = g.n * ssum >= 2 * int(d.sum()) ** 2
    late_splits = heavy and (g.n <= (1 << 21) or split_c >= g.n)

    total_w = g.total_node_weight
    mbw_val = g.max_block_weight(k, eps)

    def weights(gr):
        from kaminpar_amd import _lib
        vw = _lib.kmp_graph_vwgt(gr._h)
        aw = _lib.kmp_graph_adjwgt(gr._h)
        v = np.ctypeslib.as_array(vw, shape=(gr.n,)) if vw else None
        a = np.ctypeslib.as_array(aw, shape=(gr.m,)) if aw else None
        return v, a

    graphs = [g]
    mappings = []
    while graphs[-1].n > max(stop_n, 2 * k):
        cur = graphs[-1]
        mcw = level_cluster_weight(total_w, cur.n, k, eps, contraction_limit)
        vw, aw = weights(cur)
        nc, clus, _ = oracle_cluster(oracle, cur, mcw, seed=seed + len(mappings),
                                     iters=iters, vwgt=vw, adjwgt=aw)
        coarse, mapping = oracle_contract(oracle, cur, clus, vwgt=vw, adjwgt=aw)
        if coarse.n > 0.95 * cur.n:
            break
        graphs.append(coarse)
        mappings.append(mapping)

    part = np.zeros(graphs[-1].n, dtype=np.uint32)
    groups = [(0, k)]
    cut = None
    coarsest = len(graphs) - 1
    for level in range(coarsest, -1, -1):
        hg = graphs[level]
        sc = (min(split_c, 48) if level == coarsest and not late_splits
              else split_c)
        if len(groups) < k and (hg.n >= 2 * sc * len(groups)
                                or level == 0):
            part, groups = _extend_partition(hg, part, groups, mbw_val, k,
                                             sc, reps,
                                             force=(level == 0))
            if len(groups) == k:
                hg.balance_partition(k, mbw_val, part)
        vw, aw = weights(hg)
        caps = _group_caps(groups, k, mbw_val)
        cut, part, _ = oracle_refine(oracle, hg, k, caps, part,
                                     seed=seed, iters=iters, vwgt=vw,
                                     adjwgt=aw)
        # per-level k-way boundary FM on small graphs (keep in sync with
        # partition_deep)
        if g.n <= (1 << 21):
            part = hg.kway_fm(k, caps, part)
            if level == 0:
                cut = g.edge_cut(part)
        if level > 0:
            part = part[mappings[level - 1]]
    return cut, part, [gr.n for gr in graphs]
