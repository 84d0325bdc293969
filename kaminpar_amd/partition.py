"""Multilevel k-way partitioning pipeline (BASELINE config 3).

Mirrors the reference's basic multilevel scheme
(kaminpar-shm/partitioning/deep/deep_multilevel.cc:55-66:
uncoarsen(initial_partition(coarsen())) with the default preset's knobs:
LP clustering with the EPSILON_BLOCK_WEIGHT cap
(coarsening/max_cluster_weights.h:18-46, contraction limit 2000,
presets.cc:185), cluster contraction, and LP refinement at every level):

  - coarsening: GPU LP clustering + GPU contraction per level;
  - initial partitioning: recursive greedy graph-growing bisection on the
    coarsest graph (CPU; a simplified stand-in for the reference's
    sequential initial-partitioner pool -- quality is validated against the
    compiled reference's full pipeline, not claimed bit-parity);
  - uncoarsening: project through the contraction mapping and run the GPU
    LP refiner at each level (the deterministic schedule; balancers are not
    implemented yet, so the initial bisection respects the caps strictly).
"""

import numpy as np

from . import LpEngine


def _greedy_bisect(xadj, adjncy, adjwgt, vwgt, n_total, nodes, target1, cap1):
    """Split `nodes` (array of vertex ids) into (part1, part2): grow part1
    from a max-degree seed by repeatedly absorbing the frontier vertex with
    the highest connection into the region, until its weight reaches
    target1. Deterministic (ties: smaller vertex id). Restates the idea of
    the reference's GreedyGraphGrowingBipartitioner
    (kaminpar-shm/initial_partitioning/initial_ggg_bipartitioner.cc), not
    its exact queue schedule."""
    nodes = np.asarray(nodes)
    loc = np.full(n_total, -1, dtype=np.int64)
    loc[nodes] = np.arange(len(nodes))

    # degree within the subgraph, for seed choice
    mask = (loc[adjncy] >= 0).astype(np.int64)
    cs = np.concatenate([[0], np.cumsum(mask)])
    deg_in = cs[xadj[nodes + 1]] - cs[xadj[nodes]]
    seed_best = deg_in == deg_in.max()
    seed_idx = int(np.where(seed_best)[0][np.argmin(nodes[seed_best])])

    in_region = np.zeros(len(nodes), dtype=bool)
    gain = np.full(len(nodes), -1, dtype=np.int64)  # -1 = not frontier
    w1 = 0

    def add(i):
        nonlocal w1
        in_region[i] = True
        w1 += int(vwgt[nodes[i]])
        gain[i] = -2  # consumed / blocked
        u = int(nodes[i])
        e0, e1 = int(xadj[u]), int(xadj[u + 1])
        j = loc[adjncy[e0:e1]]
        sel = j >= 0
        j = j[sel]
        keep = ~in_region[j] & (gain[j] != -2)
        j = j[keep]
        if len(j) == 0:
            return
        gain[j[gain[j] < 0]] = 0
        w = adjwgt[e0:e1][sel][keep] if adjwgt is not None else 1
        np.add.at(gain, j, w)

    add(seed_idx)
    while w1 < target1:
        cand = np.where(gain >= 0)[0]
        if len(cand) == 0:
            # disconnected: seed a new component (smallest id outside)
            rest = np.where(~in_region & (gain != -2))[0]
            if len(rest) == 0:
                break
            nxt = int(rest[np.argmin(nodes[rest])])
        else:
            best = cand[gain[cand] == gain[cand].max()]
            nxt = int(best[np.argmin(nodes[best])])
        if w1 + int(vwgt[nodes[nxt]]) > cap1:
            gain[nxt] = -2  # cannot take it; block and continue
            continue
        add(nxt)

    part1 = nodes[in_region]
    part2 = nodes[~in_region]
    return part1, part2


def initial_partition(g, k, max_block_weight, seed=1):
    """Recursive bisection into k blocks on the (small) coarsest graph."""
    xadj = np.asarray(g.xadj).astype(np.int64)
    adjncy = np.asarray(g.adjncy)
    vwgt = np.ones(g.n, dtype=np.int64)
    # host graph may carry weights
    from . import _lib

    vw = _lib.kmp_graph_vwgt(g._h)
    if vw:
        vwgt = np.ctypeslib.as_array(vw, shape=(g.n,)).astype(np.int64)
    aw = _lib.kmp_graph_adjwgt(g._h)
    adjwgt = np.ctypeslib.as_array(aw, shape=(g.m,)) if aw else None

    part = np.zeros(g.n, dtype=np.uint32)

    def rec(nodes, k_lo, k_hi):
        if len(nodes) == 0:
            return
        if k_hi - k_lo == 1:
            part[nodes] = k_lo
            return
        k1 = (k_hi - k_lo + 1) // 2
        k2 = (k_hi - k_lo) - k1
        total = int(vwgt[nodes].sum())
        target1 = total * k1 // (k1 + k2)
        p1, p2 = _greedy_bisect(
            xadj, adjncy, adjwgt, vwgt, g.n, nodes,
            target1, k1 * max_block_weight,
        )
        rec(p1, k_lo, k_lo + k1)
        rec(p2, k_lo + k1, k_hi)

    rec(np.arange(g.n), 0, k)
    return part


def partition(g, k, eps=0.03, seed=1, iters=5, contraction_limit=2000):
    """Full multilevel partition. Returns (cut, partition, levels_info)."""
    total_w = g.total_node_weight
    mbw_val = g.max_block_weight(k, eps)
    mbw = np.full(k, mbw_val, dtype=np.int64)

    # ---- coarsen (GPU) ----
    graphs = [g]
    mappings = []
    engines = [LpEngine(g)]
    while graphs[-1].n > max(2 * contraction_limit, 2 * k):
        cur = graphs[-1]
        shrink = min(max(cur.n // contraction_limit, 2), k)
        mcw = max(1, int(eps * total_w / shrink))
        nc, clus, _ = engines[-1].cluster(mcw, seed=seed + len(mappings), iters=iters)
        coarse, mapping = engines[-1].contract(clus)
        if coarse.n > 0.95 * cur.n:
            break
        graphs.append(coarse)
        mappings.append(mapping)
        engines.append(LpEngine(coarse))

    # ---- initial partition (CPU, coarsest) ----
    part = initial_partition(graphs[-1], k, mbw_val, seed=seed)

    # ---- uncoarsen: refine at every level (GPU) ----
    cut = None
    for level in range(len(graphs) - 1, -1, -1):
        cut, part, _ = engines[level].refine(k, mbw, part, seed=seed, iters=iters)
        if level > 0:
            part = part[mappings[level - 1]]
    levels = [gr.n for gr in graphs]
    return cut, part, levels
