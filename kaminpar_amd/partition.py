"""Multilevel k-way partitioning pipeline (BASELINE config 3).

Mirrors the reference's multilevel scheme (coarsen -> initial partition ->
uncoarsen+refine, kaminpar-shm/partitioning/): GPU LP clustering + GPU
contraction per level, CPU initial partitioning on the coarsest graph, GPU
LP refinement at every level.

Initial partitioning restates the reference's recipe in simplified form
(kaminpar-shm/initial_partitioning/): recursive bisection where each
bisection is greedy graph growing (initial_ggg_bipartitioner.cc) polished by
two-way FM (initial_two_way_fm_refiner.cc: best-prefix rollback, 100
fruitless moves, 5 passes), followed by a gain-aware overload balancer
(refinement/balancer/overload_balancer.cc in spirit). It is NOT claimed
bit-parity with the reference's initial-partitioner pool (which races many
repetitions); quality is validated end-to-end against the compiled
reference's full pipeline within a band (tests/golden/ref_golden_partition.json).

Cluster-weight rule per level: the reference's EPSILON_BLOCK_WEIGHT formula
(coarsening/max_cluster_weights.h:18-46) capped by its BLOCK_WEIGHT rule
with multiplier 1/12 (presets.cc: initial_partitioning.coarsening
.cluster_weight_limit = BLOCK_WEIGHT, multiplier = 1/12) so coarse vertices
stay small enough relative to the block capacity for a feasible initial
assignment -- without the cap, deep coarsening produces vertices weighing
~25% of a block, which makes balanced bin-packing infeasible.
"""

import os

import numpy as np

from . import LpEngine


def _subgraph_csr(xadj, adjncy, adjwgt, nodes, loc):
    """Extract local CSR of the induced subgraph (local vertex ids)."""
    rows_j = []
    rows_w = []
    sub_xadj = np.zeros(len(nodes) + 1, dtype=np.int64)
    for i, u in enumerate(nodes):
        e0, e1 = int(xadj[u]), int(xadj[u + 1])
        j = loc[adjncy[e0:e1]]
        sel = j >= 0
        rows_j.append(j[sel])
        if adjwgt is not None:
            rows_w.append(adjwgt[e0:e1][sel].astype(np.int64))
        sub_xadj[i + 1] = sub_xadj[i] + int(sel.sum())
    sub_adj = np.concatenate(rows_j) if rows_j else np.zeros(0, np.int64)
    if adjwgt is not None:
        sub_w = np.concatenate(rows_w) if rows_w else np.zeros(0, np.int64)
    else:
        sub_w = np.ones(len(sub_adj), dtype=np.int64)
    return sub_xadj, sub_adj, sub_w


def _fm_refine_bisection(sub_xadj, sub_adj, sub_w, vw, side, cap1, cap2,
                         max_passes=5, max_fruitless=100):
    """Two-way FM with best-prefix rollback on a bisection (side=True is
    part 1). Restates initial_two_way_fm_refiner.cc's schedule in simplified
    form: repeated passes, each greedily moving the best-gain movable vertex
    (ties: smallest id), locking it, tracking the best prefix."""
    n = len(side)
    w1 = int(vw[side].sum())
    w2 = int(vw.sum()) - w1
    for _ in range(max_passes):
        # gains from scratch: external - internal connection
        gains = np.zeros(n, dtype=np.int64)
        for i in range(n):
            e0, e1 = int(sub_xadj[i]), int(sub_xadj[i + 1])
            nb = sub_adj[e0:e1]
            w = sub_w[e0:e1]
            cross = side[nb] != side[i]
            gains[i] = int(w[cross].sum()) - int(w[~cross].sum())
        locked = np.zeros(n, dtype=bool)
        moves = []
        cum = best = 0
        best_len = 0
        fruitless = 0
        wa, wb = w1, w2
        while fruitless < max_fruitless:
            feas = (~locked) & (
                (side & (wb + vw <= cap2)) | (~side & (wa + vw <= cap1))
            )
            if not feas.any():
                break
            masked = np.where(feas, gains, np.iinfo(np.int64).min)
            i = int(np.argmax(masked))  # first occurrence = smallest id
            cum += int(gains[i])
            old = bool(side[i])
            e0, e1 = int(sub_xadj[i]), int(sub_xadj[i + 1])
            nb = sub_adj[e0:e1]
            w = sub_w[e0:e1]
            upd = ~locked[nb]
            same = side[nb[upd]] == old
            delta = np.where(same, 2 * w[upd], -2 * w[upd])
            np.add.at(gains, nb[upd], delta)
            gains[i] = -gains[i]
            side[i] = not old
            locked[i] = True
            if old:
                wa -= int(vw[i]); wb += int(vw[i])
            else:
                wa += int(vw[i]); wb -= int(vw[i])
            moves.append(i)
            if cum > best:
                best = cum
                best_len = len(moves)
                fruitless = 0
            else:
                fruitless += 1
        # rollback past the best prefix
        for i in moves[best_len:]:
            side[i] = not side[i]
        w1 = int(vw[side].sum())
        w2 = int(vw.sum()) - w1
        if best <= 0:
            break
    return side


def _greedy_grow(sub_xadj, sub_adj, sub_w, vw, target1, cap1, seed_rank=0):
    """Greedy graph growing: grow part 1 from a high-degree seed (the
    seed_rank-th vertex in descending-degree order -- repetitions use
    different seeds, like the reference's bipartitioner pool), absorbing
    the frontier vertex with the highest connection (ties: smallest id)."""
    n = len(vw)
    deg = np.diff(sub_xadj)
    order = np.lexsort((np.arange(n), -deg))
    seed_idx = int(order[seed_rank % n])

    side = np.zeros(n, dtype=bool)
    gain = np.full(n, -1, dtype=np.int64)  # -1 not frontier, -2 blocked/in
    w1 = 0

    def add(i):
        nonlocal w1
        side[i] = True
        w1 += int(vw[i])
        gain[i] = -2
        e0, e1 = int(sub_xadj[i]), int(sub_xadj[i + 1])
        j = sub_adj[e0:e1]
        keep = ~side[j] & (gain[j] != -2)
        j2 = j[keep]
        if len(j2) == 0:
            return
        gain[j2[gain[j2] < 0]] = 0
        np.add.at(gain, j2, sub_w[e0:e1][keep])

    add(seed_idx)
    while w1 < target1:
        cand = np.flatnonzero(gain >= 0)
        if len(cand) == 0:
            rest = np.flatnonzero(~side & (gain != -2))
            if len(rest) == 0:
                break
            nxt = int(rest[0])  # disconnected: smallest id outside
        else:
            best = cand[gain[cand] == gain[cand].max()]
            nxt = int(best[0])
        if w1 + int(vw[nxt]) > cap1:
            gain[nxt] = -2
            continue
        add(nxt)
    return side


def _balance(xadj, adjncy, adjwgt, vwgt, part, k, cap):
    """Gain-aware overload balancer: while a block exceeds cap, move the
    best vertex out of the most overloaded block -- preferring the
    max-gain feasible move, else any move that strictly lowers the
    overloaded block below the target's new weight (monotone, terminates)."""
    n = len(part)
    vw = vwgt if vwgt is not None else np.ones(n, dtype=np.int64)
    bw = np.zeros(k, dtype=np.int64)
    np.add.at(bw, part, vw)
    guard = 8 * (k + 16)
    while bw.max() > cap and guard > 0:
        guard -= 1
        b = int(np.argmax(bw))
        vs = np.flatnonzero(part == b)
        best_feas = None   # (gain, -weight, v, t)
        best_force = None  # (new_target_weight, v, t)
        for v in vs:
            e0, e1 = int(xadj[v]), int(xadj[v + 1])
            nb = part[adjncy[e0:e1]]
            w = adjwgt[e0:e1] if adjwgt is not None else np.ones(e1 - e0, np.int64)
            conn = np.zeros(k, dtype=np.int64)
            np.add.at(conn, nb, w)
            internal = int(conn[b])
            for t in range(k):
                if t == b:
                    continue
                nw = int(bw[t]) + int(vw[v])
                g = int(conn[t]) - internal
                if nw <= cap:
                    key = (g, -int(vw[v]), -v, t)
                    if best_feas is None or key > best_feas[:4]:
                        best_feas = (g, -int(vw[v]), -v, t, v)
                elif nw < int(bw[b]):
                    key = (-nw, g, -v)
                    if best_force is None or key > best_force[:3]:
                        best_force = (-nw, g, -v, t, v)
        if best_feas is not None:
            t, v = best_feas[3], best_feas[4]
        elif best_force is not None:
            t, v = best_force[3], best_force[4]
        else:
            break
        part[v] = t
        bw[b] -= int(vw[v])
        bw[t] += int(vw[v])
    return part


def initial_partition(g, k, max_block_weight, seed=1, reps=8):
    """Recursive bisection into k blocks on the (small) coarsest graph:
    per bisection, `reps` greedy-graph-growing attempts from different
    high-degree seeds, each polished by two-way FM, best cut kept
    (restates the reference's bipartitioner-pool repetition idea,
    initial_partitioning with min_num_non_adaptive_repetitions=5); then a
    k-way overload balancer."""
    xadj = np.asarray(g.xadj).astype(np.int64)
    adjncy = np.asarray(g.adjncy).astype(np.int64)
    from . import _lib

    vwgt = None
    vw_p = _lib.kmp_graph_vwgt(g._h)
    if vw_p:
        vwgt = np.ctypeslib.as_array(vw_p, shape=(g.n,)).astype(np.int64)
    adjwgt = None
    aw_p = _lib.kmp_graph_adjwgt(g._h)
    if aw_p:
        adjwgt = np.ctypeslib.as_array(aw_p, shape=(g.m,)).astype(np.int64)
    vwgt_all = vwgt if vwgt is not None else np.ones(g.n, dtype=np.int64)

    part = np.zeros(g.n, dtype=np.uint32)
    loc = np.full(g.n, -1, dtype=np.int64)

    def rec(nodes, k_lo, k_hi):
        if len(nodes) == 0:
            return
        if k_hi - k_lo == 1:
            part[nodes] = k_lo
            return
        k1 = (k_hi - k_lo + 1) // 2
        k2 = (k_hi - k_lo) - k1
        total = int(vwgt_all[nodes].sum())
        target1 = total * k1 // (k1 + k2)
        cap1 = k1 * max_block_weight
        cap2 = k2 * max_block_weight

        loc[nodes] = np.arange(len(nodes))
        sub_xadj, sub_adj, sub_w = _subgraph_csr(xadj, adjncy, adjwgt, nodes, loc)
        loc[nodes] = -1
        vw = vwgt_all[nodes]

        def bisection_cut(s):
            u = np.repeat(np.arange(len(nodes)), np.diff(sub_xadj))
            return int(sub_w[s[u] != s[sub_adj]].sum())

        best_side = None
        best_cut = None
        for rep in range(reps):
            side = _greedy_grow(sub_xadj, sub_adj, sub_w, vw, target1, cap1,
                                seed_rank=rep)
            side = _fm_refine_bisection(sub_xadj, sub_adj, sub_w, vw, side,
                                        cap1, cap2)
            c = bisection_cut(side)
            if best_cut is None or c < best_cut:
                best_cut = c
                best_side = side
        rec(nodes[best_side], k_lo, k_lo + k1)
        rec(nodes[~best_side], k_lo + k1, k_hi)

    rec(np.arange(g.n), 0, k)
    part = _balance(xadj, adjncy, adjwgt, vwgt, part, k, max_block_weight)
    return part


def level_cluster_weight(total_w, n, k, eps, contraction_limit=2000):
    """Per-level max cluster weight: the reference's EPSILON_BLOCK_WEIGHT
    formula (max_cluster_weights.h:18-46) capped by the BLOCK_WEIGHT rule
    with the reference's IP multiplier 1/12 (presets.cc)."""
    shrink = min(max(n // contraction_limit, 2), k)
    eps_rule = int(eps * total_w / shrink)
    block_rule = total_w // (12 * k)
    return max(1, min(eps_rule, block_rule) if block_rule > 0 else eps_rule)


def partition(g, k, eps=0.03, seed=1, iters=5, contraction_limit=2000,
              stop_n=512, engine=None, return_arcs=False):
    """Full multilevel partition on the GPU engine.

    `engine` reuses an existing LpEngine for the fine graph (keeps the
    fine CSR resident in HBM across repeated runs, e.g. in bench.py).
    Returns (cut, partition, level_sizes); with return_arcs also the LP
    arcs scanned and phase-A kernel nanoseconds summed over all levels."""
    total_w = g.total_node_weight
    mbw_val = g.max_block_weight(k, eps)
    mbw = np.full(k, mbw_val, dtype=np.int64)

    # ---- coarsen (GPU, device-resident chain: the coarse CSR is handed
    # engine-to-engine in HBM; only the mapping comes back to the host) ----
    sizes = [g.n]
    mappings = []
    engines = [engine if engine is not None else LpEngine(g)]
    arcs_total = 0
    ns_total = 0
    while sizes[-1] > max(stop_n, 2 * k):
        cur_n = sizes[-1]
        mcw = level_cluster_weight(total_w, cur_n, k, eps, contraction_limit)
        nc, clus, cst = engines[-1].cluster(mcw, seed=seed + len(mappings), iters=iters)
        arcs_total += cst.arcs_scanned
        ns_total += cst.phase_a_ns
        coarse_eng, mapping = engines[-1].contract_engine(clus)
        if coarse_eng.n > 0.95 * cur_n:
            del coarse_eng
            break
        engines.append(coarse_eng)
        mappings.append(mapping)
        sizes.append(coarse_eng.n)

    # ---- initial partition (CPU, coarsest downloaded from HBM; the C++
    # implementation, bit-identical to initial_partition() below) ----
    coarsest = engines[-1].download_graph() if len(engines) > 1 else g
    part = coarsest.initial_partition_native(k, mbw_val)

    # ---- uncoarsen: refine at every level (GPU), plus per-level k-way
    # boundary FM on small graphs (<= ~2M fine vertices; same recipe as
    # partition_deep) ----
    cut = None
    fm_on = g.n <= (1 << 21)
    for level in range(len(engines) - 1, -1, -1):
        cut, part, rst = engines[level].refine(k, mbw, part, seed=seed, iters=iters)
        arcs_total += rst.arcs_scanned
        ns_total += rst.phase_a_ns
        if fm_on:
            hg = g if level == 0 else engines[level].download_graph()
            part = hg.kway_fm(k, mbw, part)
            if level == 0:
                cut = g.edge_cut(part)
        if level > 0:
            part = part[mappings[level - 1]]
    levels = sizes
    if return_arcs:
        return cut, part, levels, int(arcs_total), int(ns_total)
    return cut, part, levels


def _group_caps(groups, k, mbw_val):
    caps = np.zeros(k, dtype=np.int64)
    for b, w in groups:
        caps[b] = w * mbw_val
    return caps


def _extend_partition(hg, part, groups, mbw_val, k, split_c=256, reps=8,
                      force=False):
    """Split every splittable block group in half via FM-polished bisection
    of its induced subgraph (the shape of the reference's deep-multilevel
    partition extension, kaminpar-shm/partitioning/deep/deep_multilevel.cc:
    bipartition blocks while uncoarsening instead of full-k at the coarsest).
    groups is a list of (first_block_id, width); repeats until every block
    would drop below split_c vertices (or all widths are 1)."""
    from . import _lib

    vwp = _lib.kmp_graph_vwgt(hg._h)
    vw = (np.ctypeslib.as_array(vwp, shape=(hg.n,)).astype(np.int64)
          if vwp else np.ones(hg.n, np.int64))
    from concurrent.futures import ThreadPoolExecutor

    def _bisect_group(task):
        _b, _k1, nodes, t1, reps_eff, cap1, cap2 = task
        # deterministic dispatch (keep in sync with the C twin
        # kmp_extend_partition): pinned O(n^2) bisector <= 128
        # vertices; above that by degree variance (CV^2 >= 1):
        # heavy-tailed subgraphs use flat FM (O(n^2) to 4096, then
        # lazy-PQ -- HEM collapses hubs), low-variance
        # (geometric/mesh-like) subgraphs use the HEM multilevel
        # bisector, where flat FM gets lost (measured: rgg2d k=2 at
        # 2.5x the reference with flat vs 1.0x with HEM)
        ns = len(nodes)
        if ns <= 128:
            bisect = hg.bisect_subset
        else:
            xadj = np.asarray(hg.xadj)
            d = (xadj[nodes.astype(np.int64) + 1]
                 - xadj[nodes.astype(np.int64)]).astype(object)
            s = int(np.sum(d))
            sq = int(np.sum(d * d))
            heavy_tail = ns * sq >= 2 * s * s
            if heavy_tail:
                bisect = (hg.bisect_subset if ns <= 4096
                          else hg.bisect_subset_fast)
            else:
                bisect = hg.bisect_subset_ml
        return bisect(nodes, t1, cap1, cap2, reps=reps_eff)

    while True:
        num = len(groups)
        if num >= k:
            break
        if not force and hg.n < 2 * split_c * num:
            break
        new_groups = []
        tasks = []
        for b, w in groups:
            if w < 2:
                new_groups.append((b, w))
                continue
            k1 = (w + 1) // 2
            k2 = w - k1
            nodes = np.flatnonzero(part == b).astype(np.uint32)
            if len(nodes) == 0:
                new_groups += [(b, k1), (b + k1, k2)]
                continue
            total = int(vw[nodes].sum())
            t1 = total * k1 // w
            ns = len(nodes)
            reps_eff = reps if ns <= 16384 else 4
            reps_eff = min(reps, reps_eff)
            tasks.append((b, k1, nodes, t1, reps_eff,
                          k1 * mbw_val, k2 * mbw_val))
            new_groups += [(b, k1), (b + k1, k2)]
        # groups are independent subproblems (disjoint part[] writes); the
        # ctypes bisector calls release the GIL, so a thread pool gives the
        # same per-group parallelism as the C twin's OpenMP loop, with
        # bit-identical results (sides applied serially afterwards)
        if len(tasks) > 1:
            with ThreadPoolExecutor(max_workers=os.cpu_count()) as ex:
                sides = list(ex.map(_bisect_group, tasks))
        else:
            sides = [_bisect_group(t) for t in tasks]
        for (b, k1, nodes, _t1, _r, _c1, _c2), side in zip(tasks, sides):
            part[nodes[~side]] = b + k1
        groups = new_groups
    return part, groups


def partition_deep(g, k, eps=0.03, seed=1, iters=5, contraction_limit=2000,
                   stop_n=512, split_c=None, reps=8, engine=None,
                   return_arcs=False):
    """Progressive-k multilevel partition: coarsen as in partition(), then
    instead of full-k initial partitioning at the coarsest level, grow k by
    FM-polished block bisections DURING uncoarsening whenever every block
    still holds >= split_c vertices -- the shape of the reference's deep
    multilevel mode. LP refinement runs at every level with per-group block
    caps (width x uniform cap; unopened block ids get cap 0).

    Returns (cut, partition, level_sizes)."""
    if split_c is None:
        # fine-level structure formation is affordable (and valuable) up to
        # ~2M vertices; beyond that CPU bisection cost explodes while the
        # quality difference vanishes (measured at scale 26: one clustering
        # level already destroyed the structure fine splits would need), so
        # fall back to reference-like block sizes (~2x contraction limit)
        split_c = 262144 if g.n <= (1 << 21) else 2000
    # Split-schedule dispatch by degree variance of the FINE graph (same
    # CV^2 >= 1 statistic as the bisector dispatch): on heavy-tailed graphs
    # every clustering level destroys cut structure (measured: mid-level
    # splits cost 1.4-1.6x vs <=1.0x for finest-level splits on R-MAT), so
    # defer ALL splits to the finest affordable level by skipping the eager
    # coarsest-level split; on low-variance (mesh-like) graphs multilevel
    # structure transfers well and the eager coarsest split wins.
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
    # split_c >= n is the explicit full-late quality mode: every split at
    # the finest level regardless of size (measured at scale 23: cut 0.44x
    # the reference's best seed, profiles/round1/quality_rmat23_k16_late_full.json)
    late_splits = heavy and (g.n <= (1 << 21) or split_c >= g.n)
    total_w = g.total_node_weight
    mbw_val = g.max_block_weight(k, eps)

    import os as _os
    import time as _time
    _tdbg = _os.environ.get("KMP_TIME") == "1"
    _tt = {"cluster": 0.0, "contract": 0.0, "extend": 0.0, "refine": 0.0,
           "download": 0.0, "fm": 0.0, "other": 0.0}
    _t0 = _time.perf_counter()
    sizes = [g.n]
    mappings = []
    engines = [engine if engine is not None else LpEngine(g)]
    arcs_total = 0
    ns_total = 0
    while sizes[-1] > max(stop_n, 2 * k):
        cur_n = sizes[-1]
        mcw = level_cluster_weight(total_w, cur_n, k, eps, contraction_limit)
        _tc = _time.perf_counter()
        nc, clus, cst = engines[-1].cluster(mcw, seed=seed + len(mappings),
                                           iters=iters)
        _tt["cluster"] += _time.perf_counter() - _tc
        arcs_total += cst.arcs_scanned
        ns_total += cst.phase_a_ns
        _tc = _time.perf_counter()
        coarse_eng, mapping = engines[-1].contract_engine(clus)
        _tt["contract"] += _time.perf_counter() - _tc
        if coarse_eng.n > 0.95 * cur_n:
            del coarse_eng
            break
        engines.append(coarse_eng)
        mappings.append(mapping)
        sizes.append(coarse_eng.n)

    part = np.zeros(sizes[-1], dtype=np.uint32)
    groups = [(0, k)]
    cut = None
    coarsest = len(engines) - 1
    for level in range(coarsest, -1, -1):
        # at the coarsest level split eagerly (down to ~32-vertex blocks,
        # like the reference's initial bipartition of the coarsest graph);
        # afterwards extend only when every block keeps >= split_c vertices
        sc = (min(split_c, 48) if level == coarsest and not late_splits
              else split_c)
        hg = None
        if len(groups) < k and (sizes[level] >= 2 * sc * len(groups)
                                or level == 0):
            _tc = _time.perf_counter()
            hg = g if level == 0 else engines[level].download_graph()
            _tt["download"] += _time.perf_counter() - _tc
            _tc = _time.perf_counter()
            part, groups = _extend_partition(hg, part, groups, mbw_val, k,
                                             sc, reps,
                                             force=(level == 0))
            if len(groups) == k:
                hg.balance_partition(k, mbw_val, part)
            _tt["extend"] += _time.perf_counter() - _tc
        caps = _group_caps(groups, k, mbw_val)
        _tc = _time.perf_counter()
        cut, part, rst = engines[level].refine(
            k, caps, part, seed=seed, iters=iters)
        _tt["refine"] += _time.perf_counter() - _tc
        arcs_total += rst.arcs_scanned
        ns_total += rst.phase_a_ns
        # per-level k-way boundary FM on small graphs (<= ~2M fine
        # vertices): recovers the bisection quality LP refinement alone
        # cannot on mesh-like graphs (classic multilevel FM recipe)
        if g.n <= (1 << 21):
            _tc = _time.perf_counter()
            if hg is None:
                hg = g if level == 0 else engines[level].download_graph()
            part = hg.kway_fm(k, caps, part)
            if level == 0:
                cut = g.edge_cut(part)
            _tt["fm"] += _time.perf_counter() - _tc
        if level > 0:
            part = part[mappings[level - 1]]
    if _tdbg:
        _tt["other"] = (_time.perf_counter() - _t0) - sum(
            v for q, v in _tt.items() if q != "other")
        import sys as _sys
        print("[deep-timing] " + " ".join(f"{q}={v:.2f}s"
                                          for q, v in _tt.items()),
              file=_sys.stderr)
    if return_arcs:
        return cut, part, sizes, int(arcs_total), int(ns_total)
    return cut, part, sizes
