// Host-side initial partitioning for the multilevel pipeline: recursive
// bisection (greedy graph growing from distinct high-degree seeds + two-way
// FM with best-prefix rollback, best-of-`reps` per bisection) followed by a
// gain-aware k-way overload balancer.
//
// Restates the reference's initial-partitioning recipe in simplified form
// (kaminpar-shm/initial_partitioning/: initial_ggg_bipartitioner.cc,
// initial_two_way_fm_refiner.cc with num_fruitless_moves=100 and 5 passes,
// refinement/balancer/overload_balancer.cc in spirit). Deterministic and
// EXACTLY equivalent to kaminpar_amd/partition.py's numpy implementation --
// tests/test_pipeline_cpu.py pins the equivalence, and the committed
// pipeline goldens (tests/golden/pipeline_expected.json) pin the results.
// CPU-only: no HIP; runs in the dev container.

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <omp.h>
#include <vector>

using u32 = uint32_t;
using u64 = uint64_t;
using i32 = int32_t;
using i64 = int64_t;

extern "C" {
// from graph_gen.cpp
typedef struct kmp_graph_t kmp_graph_t;
u32 kmp_graph_n(const kmp_graph_t *);
u64 kmp_graph_m(const kmp_graph_t *);
const u32 *kmp_graph_xadj(const kmp_graph_t *);
const u32 *kmp_graph_adjncy(const kmp_graph_t *);
const i32 *kmp_graph_vwgt(const kmp_graph_t *);
const i32 *kmp_graph_adjwgt(const kmp_graph_t *);
}

namespace {

struct SubCsr {
  std::vector<i64> xadj;
  std::vector<i64> adj; // local ids
  std::vector<i64> w;
};

// Induced-subgraph CSR in local ids (nodes ascending; local id = position).
SubCsr subgraph_csr(
    const u32 *xadj, const u32 *adjncy, const i32 *adjwgt,
    const std::vector<i64> &nodes, std::vector<i64> &loc
) {
  SubCsr s;
  s.xadj.assign(nodes.size() + 1, 0);
  for (size_t i = 0; i < nodes.size(); ++i) {
    const u32 u = static_cast<u32>(nodes[i]);
    for (u64 e = xadj[u]; e < xadj[u + 1]; ++e) {
      const i64 j = loc[adjncy[e]];
      if (j >= 0) {
        s.adj.push_back(j);
        s.w.push_back(adjwgt ? adjwgt[e] : 1);
      }
    }
    s.xadj[i + 1] = static_cast<i64>(s.adj.size());
  }
  return s;
}

// Greedy graph growing (partition.py _greedy_grow): grow part 1 from the
// seed_rank-th vertex in descending-degree order (ties: smaller local id).
std::vector<uint8_t> greedy_grow(
    const SubCsr &s, const std::vector<i64> &vw, i64 target1, i64 cap1,
    int seed_rank
) {
  const size_t n = vw.size();
  std::vector<i64> deg(n);
  for (size_t i = 0; i < n; ++i) {
    deg[i] = s.xadj[i + 1] - s.xadj[i];
  }
  std::vector<u32> order(n);
  for (size_t i = 0; i < n; ++i) {
    order[i] = static_cast<u32>(i);
  }
  std::stable_sort(order.begin(), order.end(),
                   [&](u32 a, u32 b) { return deg[a] > deg[b]; });
  const u32 seed_idx = order[static_cast<size_t>(seed_rank) % n];

  std::vector<uint8_t> side(n, 0);
  std::vector<i64> gain(n, -1); // -1 not frontier, -2 blocked/in
  i64 w1 = 0;

  auto add = [&](u32 i) {
    side[i] = 1;
    w1 += vw[i];
    gain[i] = -2;
    for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
      const i64 j = s.adj[e];
      if (!side[j] && gain[j] != -2) {
        if (gain[j] < 0) {
          gain[j] = 0;
        }
        gain[j] += s.w[e];
      }
    }
  };

  add(seed_idx);
  while (w1 < target1) {
    // frontier argmax (ties: smallest id), else smallest unblocked outside
    i64 best_g = -1;
    i64 nxt = -1;
    for (size_t i = 0; i < n; ++i) {
      if (gain[i] > best_g) {
        best_g = gain[i];
        nxt = static_cast<i64>(i);
      }
    }
    if (best_g < 0) {
      nxt = -1;
      for (size_t i = 0; i < n; ++i) {
        if (!side[i] && gain[i] != -2) {
          nxt = static_cast<i64>(i);
          break;
        }
      }
      if (nxt < 0) {
        break;
      }
    }
    if (w1 + vw[nxt] > cap1) {
      gain[nxt] = -2;
      continue;
    }
    add(static_cast<u32>(nxt));
  }
  return side;
}

// Two-way FM with best-prefix rollback (partition.py _fm_refine_bisection).
void fm_refine(
    const SubCsr &s, const std::vector<i64> &vw, std::vector<uint8_t> &side,
    i64 cap1, i64 cap2, int max_passes = 5, int max_fruitless = 100
) {
  const size_t n = side.size();
  i64 total = 0;
  for (i64 w : vw) {
    total += w;
  }
  i64 w1 = 0;
  for (size_t i = 0; i < n; ++i) {
    if (side[i]) {
      w1 += vw[i];
    }
  }
  std::vector<i64> gains(n);
  std::vector<uint8_t> locked(n);
  std::vector<u32> moves;
  for (int pass = 0; pass < max_passes; ++pass) {
    for (size_t i = 0; i < n; ++i) {
      i64 cross = 0, intern = 0;
      for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
        if (side[s.adj[e]] != side[i]) {
          cross += s.w[e];
        } else {
          intern += s.w[e];
        }
      }
      gains[i] = cross - intern;
    }
    std::fill(locked.begin(), locked.end(), 0);
    moves.clear();
    i64 cum = 0, best = 0;
    size_t best_len = 0;
    int fruitless = 0;
    i64 wa = w1, wb = total - w1;
    while (fruitless < max_fruitless) {
      i64 best_g = 0;
      i64 pick = -1;
      bool found = false;
      for (size_t i = 0; i < n; ++i) {
        if (locked[i]) {
          continue;
        }
        const bool feas =
            side[i] ? (wb + vw[i] <= cap2) : (wa + vw[i] <= cap1);
        if (!feas) {
          continue;
        }
        if (!found || gains[i] > best_g) {
          best_g = gains[i];
          pick = static_cast<i64>(i);
          found = true;
        }
      }
      if (!found) {
        break;
      }
      const u32 i = static_cast<u32>(pick);
      cum += gains[i];
      const uint8_t old = side[i];
      for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
        const i64 j = s.adj[e];
        if (!locked[j]) {
          gains[j] += (side[j] == old) ? 2 * s.w[e] : -2 * s.w[e];
        }
      }
      gains[i] = -gains[i];
      side[i] = !old;
      locked[i] = 1;
      if (old) {
        wa -= vw[i];
        wb += vw[i];
      } else {
        wa += vw[i];
        wb -= vw[i];
      }
      moves.push_back(i);
      if (cum > best) {
        best = cum;
        best_len = moves.size();
        fruitless = 0;
      } else {
        ++fruitless;
      }
    }
    for (size_t m = best_len; m < moves.size(); ++m) {
      side[moves[m]] = !side[moves[m]];
    }
    w1 = 0;
    for (size_t i = 0; i < n; ++i) {
      if (side[i]) {
        w1 += vw[i];
      }
    }
    if (best <= 0) {
      break;
    }
  }
}

// ---- O(m log n) bisection (lazy max-gain priority queues) -------------
// Same greedy-grow + FM recipe as above, but selection uses lazy PQs with
// (gain, smaller-id-wins) ordering instead of O(n) scans, so bisections of
// 10^5..10^6-vertex subgraphs are affordable -- which lets the progressive-k
// driver split blocks at FINE levels, where the cut structure still exists
// (one clustering level destroys R-MAT cut structure).
// Deterministic: pure sequential code, total order (gain, -id).

struct LazyPQ {
  // max-heap of (gain, vertex); ties -> smaller vertex id first
  std::vector<std::pair<i64, i64>> h; // (gain, -id)
  void push(i64 gain, u32 v) {
    h.emplace_back(gain, -static_cast<i64>(v));
    std::push_heap(h.begin(), h.end());
  }
  bool empty() const { return h.empty(); }
  std::pair<i64, u32> pop() {
    std::pop_heap(h.begin(), h.end());
    auto t = h.back();
    h.pop_back();
    return {t.first, static_cast<u32>(-t.second)};
  }
};

std::vector<uint8_t> greedy_grow_fast(
    const SubCsr &s, const std::vector<i64> &vw, i64 target1, i64 cap1,
    int seed_rank
) {
  const size_t n = vw.size();
  std::vector<i64> deg(n);
  for (size_t i = 0; i < n; ++i) {
    deg[i] = s.xadj[i + 1] - s.xadj[i];
  }
  std::vector<u32> order(n);
  for (size_t i = 0; i < n; ++i) {
    order[i] = static_cast<u32>(i);
  }
  std::stable_sort(order.begin(), order.end(),
                   [&](u32 a, u32 b) { return deg[a] > deg[b]; });
  const u32 seed_idx = order[static_cast<size_t>(seed_rank) % n];

  std::vector<uint8_t> side(n, 0), blocked(n, 0);
  std::vector<i64> gain(n, -1);
  LazyPQ pq;
  i64 w1 = 0;
  size_t scan_pos = 0; // for the disconnected fallback (monotone cursor)

  auto add = [&](u32 i) {
    side[i] = 1;
    w1 += vw[i];
    for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
      const i64 j = s.adj[e];
      if (!side[j] && !blocked[j]) {
        if (gain[j] < 0) {
          gain[j] = 0;
        }
        gain[j] += s.w[e];
        pq.push(gain[j], static_cast<u32>(j));
      }
    }
  };

  add(seed_idx);
  while (w1 < target1) {
    i64 v = -1;
    while (!pq.empty()) {
      auto [pg, pv] = pq.pop();
      if (side[pv] || blocked[pv] || gain[pv] != pg) {
        continue; // stale
      }
      v = pv;
      break;
    }
    if (v < 0) {
      // disconnected: first unassigned, unblocked vertex
      while (scan_pos < n && (side[scan_pos] || blocked[scan_pos])) {
        ++scan_pos;
      }
      if (scan_pos >= n) {
        break;
      }
      v = static_cast<i64>(scan_pos);
    }
    if (w1 + vw[v] > cap1) {
      blocked[v] = 1;
      continue;
    }
    add(static_cast<u32>(v));
  }
  return side;
}

void fm_refine_fast(
    const SubCsr &s, const std::vector<i64> &vw, std::vector<uint8_t> &side,
    i64 cap1, i64 cap2, int max_passes = 5, int max_fruitless = 100
) {
  const size_t n = side.size();
  i64 total = 0;
  for (i64 w : vw) {
    total += w;
  }
  i64 w1 = 0;
  for (size_t i = 0; i < n; ++i) {
    if (side[i]) {
      w1 += vw[i];
    }
  }
  std::vector<i64> gains(n);
  std::vector<uint8_t> locked(n);
  std::vector<u32> moves, waitA, waitB;
  for (int pass = 0; pass < max_passes; ++pass) {
    for (size_t i = 0; i < n; ++i) {
      i64 cross = 0, intern = 0;
      for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
        if (side[s.adj[e]] != side[i]) {
          cross += s.w[e];
        } else {
          intern += s.w[e];
        }
      }
      gains[i] = cross - intern;
    }
    std::fill(locked.begin(), locked.end(), 0);
    moves.clear();
    waitA.clear();
    waitB.clear();
    LazyPQ pq;
    for (size_t i = 0; i < n; ++i) {
      pq.push(gains[i], static_cast<u32>(i));
    }
    i64 cum = 0, best = 0;
    size_t best_len = 0;
    int fruitless = 0;
    i64 wa = w1, wb = total - w1;
    const size_t pop_budget = 64 * n + 4096; // stale/wait safety bound
    size_t pops = 0;
    while (fruitless < max_fruitless && pops < pop_budget) {
      if (pq.empty()) {
        break;
      }
      ++pops;
      auto [pg, v] = pq.pop();
      if (locked[v] || gains[v] != pg) {
        continue;
      }
      const uint8_t old = side[v];
      const bool feas = old ? (wb + vw[v] <= cap2) : (wa + vw[v] <= cap1);
      if (!feas) {
        (old ? waitA : waitB).push_back(v);
        continue;
      }
      cum += gains[v];
      for (i64 e = s.xadj[v]; e < s.xadj[v + 1]; ++e) {
        const i64 j = s.adj[e];
        if (!locked[j]) {
          gains[j] += (side[j] == old) ? 2 * s.w[e] : -2 * s.w[e];
          pq.push(gains[j], static_cast<u32>(j));
        }
      }
      gains[v] = -gains[v];
      side[v] = !old;
      locked[v] = 1;
      if (old) {
        wa -= vw[v];
        wb += vw[v];
        // side B lost headroom? no: A shrank -> B->A movers gain room
        for (u32 u : waitB) {
          if (!locked[u]) {
            pq.push(gains[u], u);
          }
        }
        waitB.clear();
      } else {
        wa += vw[v];
        wb -= vw[v];
        for (u32 u : waitA) {
          if (!locked[u]) {
            pq.push(gains[u], u);
          }
        }
        waitA.clear();
      }
      moves.push_back(v);
      if (cum > best) {
        best = cum;
        best_len = moves.size();
        fruitless = 0;
      } else {
        ++fruitless;
      }
    }
    for (size_t m = best_len; m < moves.size(); ++m) {
      side[moves[m]] = !side[moves[m]];
    }
    w1 = 0;
    for (size_t i = 0; i < n; ++i) {
      if (side[i]) {
        w1 += vw[i];
      }
    }
    if (best <= 0) {
      break;
    }
  }
}

u64 mix64(u64 x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// Multilevel bisection: heavy-edge matching -> contract -> recurse -> FM at
// every level on the way back up (the shape of the reference's
// initial-partitioner pool, which runs its own mini-multilevel per
// bisection with TWOWAY_SIMPLE_FM refinement at every level,
// presets.cc: initial_partitioning.coarsening contraction_limit 20 +
// pool.refinement TWOWAY_SIMPLE_FM). rep varies the matching visit order
// and the base-case grow seed.
void ml_bisect_rec(
    const SubCsr &s, const std::vector<i64> &vw, i64 target1, i64 cap1,
    i64 cap2, int rep, std::vector<uint8_t> &side
) {
  const size_t n = vw.size();
  i64 total = 0;
  for (i64 w : vw) {
    total += w;
  }
  if (n <= 128) {
    side = greedy_grow(s, vw, target1, cap1, rep);
    fm_refine(s, vw, side, cap1, cap2);
    return;
  }

  // heavy-edge matching (visit order: hash of (id, rep); ties smaller id)
  std::vector<u32> order(n);
  for (size_t i = 0; i < n; ++i) {
    order[i] = static_cast<u32>(i);
  }
  std::stable_sort(order.begin(), order.end(), [&](u32 a, u32 b) {
    return mix64(a * 2654435761u + rep) < mix64(b * 2654435761u + rep);
  });
  const i64 wlimit = std::max<i64>(1, total / 32);
  std::vector<i64> match(n, -1);
  for (u32 u : order) {
    if (match[u] >= 0) {
      continue;
    }
    i64 best = -1, bw = -1;
    for (i64 e = s.xadj[u]; e < s.xadj[u + 1]; ++e) {
      const i64 v = s.adj[e];
      if (v == (i64)u || match[v] >= 0 || vw[u] + vw[v] > wlimit) {
        continue;
      }
      if (s.w[e] > bw || (s.w[e] == bw && v < best)) {
        bw = s.w[e];
        best = v;
      }
    }
    match[u] = best >= 0 ? best : (i64)u;
    if (best >= 0) {
      match[best] = u;
    }
  }

  // coarse ids in fine-index order
  std::vector<i64> cid(n, -1);
  u32 cn = 0;
  for (size_t u = 0; u < n; ++u) {
    if (cid[u] < 0) {
      cid[u] = cn;
      if (match[u] != (i64)u) {
        cid[match[u]] = cn;
      }
      ++cn;
    }
  }
  if (cn > 0.95 * n) { // matching stalled: fall back to flat
    side = greedy_grow(s, vw, target1, cap1, rep);
    fm_refine(s, vw, side, cap1, cap2);
    return;
  }

  // contract
  std::vector<i64> cvw(cn, 0);
  for (size_t u = 0; u < n; ++u) {
    cvw[cid[u]] += vw[u];
  }
  std::vector<std::pair<u64, i64>> arcs;
  arcs.reserve(s.adj.size());
  for (size_t u = 0; u < n; ++u) {
    for (i64 e = s.xadj[u]; e < s.xadj[u + 1]; ++e) {
      const i64 cu = cid[u], cv = cid[s.adj[e]];
      if (cu != cv) {
        arcs.emplace_back((static_cast<u64>(cu) << 32) | static_cast<u32>(cv),
                          s.w[e]);
      }
    }
  }
  std::sort(arcs.begin(), arcs.end());
  SubCsr cs;
  cs.xadj.assign(cn + 1, 0);
  for (size_t i = 0; i < arcs.size();) {
    size_t j = i;
    i64 wsum = 0;
    while (j < arcs.size() && arcs[j].first == arcs[i].first) {
      wsum += arcs[j].second;
      ++j;
    }
    cs.adj.push_back(static_cast<i64>(arcs[i].first & 0xFFFFFFFFu));
    cs.w.push_back(wsum);
    ++cs.xadj[(arcs[i].first >> 32) + 1];
    i = j;
  }
  for (u32 c = 0; c < cn; ++c) {
    cs.xadj[c + 1] += cs.xadj[c];
  }

  std::vector<uint8_t> cside;
  ml_bisect_rec(cs, cvw, target1, cap1, cap2, rep, cside);
  side.resize(n);
  for (size_t u = 0; u < n; ++u) {
    side[u] = cside[cid[u]];
  }
  fm_refine(s, vw, side, cap1, cap2);
}

// 2x the bisection cut (both arc directions), for best-of-reps selection.
i64 bisection_cut2(const SubCsr &s, const std::vector<uint8_t> &side) {
  i64 c = 0;
  for (size_t i = 0; i < side.size(); ++i) {
    for (i64 e = s.xadj[i]; e < s.xadj[i + 1]; ++e) {
      if (side[s.adj[e]] != side[i]) {
        c += s.w[e];
      }
    }
  }
  return c;
}

struct IpCtx {
  const u32 *xadj;
  const u32 *adjncy;
  const i32 *adjwgt;
  std::vector<i64> vwgt;
  std::vector<i64> loc;
  i64 mbw;
  int reps;
  u32 *part;
};

void rec(IpCtx &c, std::vector<i64> &nodes, u32 k_lo, u32 k_hi) {
  if (nodes.empty()) {
    return;
  }
  if (k_hi - k_lo == 1) {
    for (i64 u : nodes) {
      c.part[u] = k_lo;
    }
    return;
  }
  const u32 k1 = (k_hi - k_lo + 1) / 2;
  const u32 k2 = (k_hi - k_lo) - k1;
  i64 total = 0;
  for (i64 u : nodes) {
    total += c.vwgt[u];
  }
  const i64 target1 = total * k1 / (k1 + k2);
  const i64 cap1 = static_cast<i64>(k1) * c.mbw;
  const i64 cap2 = static_cast<i64>(k2) * c.mbw;

  for (size_t i = 0; i < nodes.size(); ++i) {
    c.loc[nodes[i]] = static_cast<i64>(i);
  }
  SubCsr s = subgraph_csr(c.xadj, c.adjncy, c.adjwgt, nodes, c.loc);
  for (i64 u : nodes) {
    c.loc[u] = -1;
  }
  std::vector<i64> vw(nodes.size());
  for (size_t i = 0; i < nodes.size(); ++i) {
    vw[i] = c.vwgt[nodes[i]];
  }

  std::vector<std::vector<uint8_t>> sides(c.reps);
  std::vector<i64> cuts(c.reps);
#pragma omp parallel for schedule(dynamic, 1) if (!omp_in_parallel())
  for (int rep = 0; rep < c.reps; ++rep) {
    sides[rep] = greedy_grow(s, vw, target1, cap1, rep);
    fm_refine(s, vw, sides[rep], cap1, cap2);
    cuts[rep] = bisection_cut2(s, sides[rep]);
  }
  int bi = 0;
  for (int rep = 1; rep < c.reps; ++rep) {
    if (cuts[rep] < cuts[bi]) {
      bi = rep;
    }
  }
  const std::vector<uint8_t> &best_side = sides[bi];

  std::vector<i64> p1, p2;
  for (size_t i = 0; i < nodes.size(); ++i) {
    (best_side[i] ? p1 : p2).push_back(nodes[i]);
  }
  rec(c, p1, k_lo, k_lo + k1);
  rec(c, p2, k_lo + k1, k_hi);
}

// Gain-aware overload balancer (partition.py _balance).
void balance(
    const u32 *xadj, const u32 *adjncy, const i32 *adjwgt,
    const std::vector<i64> &vw, u32 *part, u32 n, u32 k, i64 cap
) {
  std::vector<i64> bw(k, 0);
  for (u32 u = 0; u < n; ++u) {
    bw[part[u]] += vw[u];
  }
  i64 guard = 8 * (static_cast<i64>(k) + 16);
  std::vector<i64> conn(k);
  while (guard-- > 0) {
    u32 b = 0;
    for (u32 t = 1; t < k; ++t) {
      if (bw[t] > bw[b]) {
        b = t;
      }
    }
    if (bw[b] <= cap) {
      break;
    }
    // best feasible move: max (gain, -weight, -v, t); else the move that
    // most lowers the overloaded block: max (-new_target_weight, gain, -v)
    bool have_f = false, have_o = false;
    i64 f0 = 0, f1 = 0, f2 = 0;
    i64 o0 = 0, o1 = 0, o2 = 0;
    u32 f3 = 0, fv = 0, ot = 0, ov = 0;
    for (u32 v = 0; v < n; ++v) {
      if (part[v] != b) {
        continue;
      }
      std::fill(conn.begin(), conn.end(), 0);
      for (u64 e = xadj[v]; e < xadj[v + 1]; ++e) {
        conn[part[adjncy[e]]] += adjwgt ? adjwgt[e] : 1;
      }
      const i64 internal = conn[b];
      for (u32 t = 0; t < k; ++t) {
        if (t == b) {
          continue;
        }
        const i64 nw = bw[t] + vw[v];
        const i64 g = conn[t] - internal;
        if (nw <= cap) {
          const i64 k0 = g, k1v = -vw[v], k2v = -static_cast<i64>(v);
          if (!have_f || k0 > f0 ||
              (k0 == f0 && (k1v > f1 ||
               (k1v == f1 && (k2v > f2 ||
                (k2v == f2 && t > f3)))))) {
            have_f = true;
            f0 = k0; f1 = k1v; f2 = k2v; f3 = t; fv = v;
          }
        } else if (nw < bw[b]) {
          const i64 k0 = -nw, k1v = g, k2v = -static_cast<i64>(v);
          if (!have_o || k0 > o0 ||
              (k0 == o0 && (k1v > o1 || (k1v == o1 && k2v > o2)))) {
            have_o = true;
            o0 = k0; o1 = k1v; o2 = k2v; ot = t; ov = v;
          }
        }
      }
    }
    u32 t, v;
    if (have_f) {
      t = f3; v = fv;
    } else if (have_o) {
      t = ot; v = ov;
    } else {
      break;
    }
    part[v] = t;
    bw[b] -= vw[v];
    bw[t] += vw[v];
  }
}

} // namespace

extern "C" {

int kmp_bisect_subset_fast(
    const kmp_graph_t *g, const u32 *nodes_in, u32 n_sub, i64 target1,
    i64 cap1, i64 cap2, int reps, uint8_t *side_out);
int kmp_bisect_subset_ml(
    const kmp_graph_t *g, const u32 *nodes_in, u32 n_sub, i64 target1,
    i64 cap1, i64 cap2, int reps, uint8_t *side_out);

// Bisect an arbitrary vertex subset of a host graph: `reps` greedy-grow
// attempts from distinct high-degree seeds, each FM-polished, best
// (2x directed) bisection cut kept. side_out[i] = 1 puts nodes[i] in part 1
// (weight target target1, cap cap1). Used by the progressive-k partition
// extension (the shape of the reference's deep-multilevel bisection
// extension, kaminpar-shm/partitioning/deep/deep_multilevel.cc) and
// equivalent to one recursion step of kmp_initial_partition.
int kmp_bisect_subset(
    const kmp_graph_t *g, const u32 *nodes_in, u32 n_sub, i64 target1,
    i64 cap1, i64 cap2, int reps, uint8_t *side_out
) {
  const u32 n = kmp_graph_n(g);
  const u32 *xadj = kmp_graph_xadj(g);
  const u32 *adjncy = kmp_graph_adjncy(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  const i32 *adjwgt = kmp_graph_adjwgt(g);

  std::vector<i64> nodes(nodes_in, nodes_in + n_sub);
  std::vector<i64> loc(n, -1);
  for (u32 i = 0; i < n_sub; ++i) {
    loc[nodes[i]] = i;
  }
  SubCsr s = subgraph_csr(xadj, adjncy, adjwgt, nodes, loc);
  std::vector<i64> vw(n_sub);
  for (u32 i = 0; i < n_sub; ++i) {
    vw[i] = vwgt ? vwgt[nodes[i]] : 1;
  }

  // reps run in parallel when not already inside the per-group parallel
  // loop; first-minimal-rep selection matches the sequential
  // keep-if-strictly-better semantics bit-exactly.
  std::vector<std::vector<uint8_t>> sides(reps);
  std::vector<i64> cuts(reps);
#pragma omp parallel for schedule(dynamic, 1) if (!omp_in_parallel())
  for (int rep = 0; rep < reps; ++rep) {
    sides[rep] = greedy_grow(s, vw, target1, cap1, rep);
    fm_refine(s, vw, sides[rep], cap1, cap2);
    cuts[rep] = bisection_cut2(s, sides[rep]);
  }
  int best = 0;
  for (int rep = 1; rep < reps; ++rep) {
    if (cuts[rep] < cuts[best]) {
      best = rep;
    }
  }
  for (u32 i = 0; i < n_sub; ++i) {
    side_out[i] = sides[best][i];
  }
  return 0;
}

// Progressive-k partition extension on a host graph (C twin of
// kaminpar_amd.partition._extend_partition -- keep in sync): split every
// block group with >= 2 target blocks in half via FM-polished bisection,
// repeating while every block keeps >= split_c vertices (or force).
// groups arrays (size k): group_lo[i], group_w[i] for i < *num_groups,
// updated in place. Returns 0.
int kmp_extend_partition(
    const kmp_graph_t *g, u32 *part, u32 k, i64 mbw_val, u32 split_c,
    int reps, int force, u32 *group_lo, u32 *group_w, u32 *num_groups
) {
  const u32 n = kmp_graph_n(g);
  const i32 *vwgt = kmp_graph_vwgt(g);

  while (true) {
    const u32 num = *num_groups;
    if (num >= k) {
      break;
    }
    if (!force && n < 2ull * split_c * num) {
      break;
    }
    // Bin vertices by group in ONE pass (part values are exactly the
    // group_lo ids during extension), replacing the O(n * num) per-group
    // scans; ascending-u order within each bin matches the scan order.
    std::vector<u32> gid_of_block(k, 0);
    for (u32 i = 0; i < num; ++i) {
      gid_of_block[group_lo[i]] = i;
    }
    std::vector<u32> off(num + 1, 0);
    for (u32 u = 0; u < n; ++u) {
      off[gid_of_block[part[u]] + 1]++;
    }
    for (u32 i = 0; i < num; ++i) {
      off[i + 1] += off[i];
    }
    std::vector<u32> binned(n);
    {
      std::vector<u32> cur(off.begin(), off.end() - 1);
      for (u32 u = 0; u < n; ++u) {
        binned[cur[gid_of_block[part[u]]]++] = u;
      }
    }
    std::vector<u32> nlo, nw;
    // Groups are independent subproblems (disjoint part[] writes, the
    // extracted subgraph keeps intra-subset arcs only), so the bisections
    // run in parallel over host cores; results are bit-identical to the
    // serial loop.
#pragma omp parallel for schedule(dynamic, 1)
    for (u32 i = 0; i < num; ++i) {
      const u32 b = group_lo[i], w = group_w[i];
      if (w < 2) {
        continue;
      }
      const u32 k1 = (w + 1) / 2, k2 = w - k1;
      std::vector<u32> nodes(binned.begin() + off[i],
                             binned.begin() + off[i + 1]);
      if (!nodes.empty()) {
        i64 total = 0;
        for (u32 u : nodes) {
          total += vwgt ? vwgt[u] : 1;
        }
        const i64 t1 = total * k1 / w;
        std::vector<uint8_t> side(nodes.size());
        // deterministic dispatch (keep in sync with partition.py
        // _extend_partition): <=128 vertices use the pinned O(n^2)
        // bisector; above that by degree variance (squared coefficient
        // of variation >= 1, i.e. n * sum(d^2) >= 2 * sum(d)^2):
        // heavy-tailed subgraphs use flat FM (O(n^2) to 4096, then the
        // lazy-PQ bisector -- HEM collapses hubs), low-variance
        // (geometric/mesh-like) subgraphs use the HEM multilevel
        // bisector, where flat FM gets lost (measured: rgg2d k=2 at
        // 2.5x the reference with flat vs 1.0x with HEM).
        const size_t ns = nodes.size();
        int reps_eff = reps;
        if (ns > 16384) {
          reps_eff = std::min(reps, 4);
        }
        auto *bisect = kmp_bisect_subset;
        if (ns > 128) {
          const u32 *xadj = kmp_graph_xadj(g);
          unsigned __int128 sum = 0, sq = 0;
          for (u32 u : nodes) {
            const u64 d = xadj[u + 1] - xadj[u];
            sum += d;
            sq += d * d;
          }
          const bool heavy_tail =
              static_cast<unsigned __int128>(ns) * sq >= 2 * sum * sum;
          if (heavy_tail) {
            bisect = ns <= 4096 ? kmp_bisect_subset : kmp_bisect_subset_fast;
          } else {
            bisect = kmp_bisect_subset_ml;
          }
        }
        bisect(g, nodes.data(), ns, t1,
               static_cast<i64>(k1) * mbw_val,
               static_cast<i64>(k2) * mbw_val, reps_eff, side.data());
        for (size_t i2 = 0; i2 < nodes.size(); ++i2) {
          if (!side[i2]) {
            part[nodes[i2]] = b + k1;
          }
        }
      }
    }
    for (u32 i = 0; i < num; ++i) {
      const u32 b = group_lo[i], w = group_w[i];
      if (w < 2) {
        nlo.push_back(b);
        nw.push_back(w);
      } else {
        const u32 k1 = (w + 1) / 2;
        nlo.push_back(b);
        nw.push_back(k1);
        nlo.push_back(b + k1);
        nw.push_back(w - k1);
      }
    }
    *num_groups = static_cast<u32>(nlo.size());
    for (u32 i = 0; i < *num_groups; ++i) {
      group_lo[i] = nlo[i];
      group_w[i] = nw[i];
    }
  }
  return 0;
}

// Deterministic k-way boundary FM with pass-level best-prefix rollback --
// the serial-deterministic restatement of the shape of the reference's
// k-way FM refiner (kaminpar-shm/refinement/fm/fm_refiner.cc: gain-PQ over
// boundary nodes, moves to the best feasible adjacent block, bounded
// negative-gain hill climbing, rollback to the best seen prefix). Used by
// the multilevel drivers on fine graphs (<= ~2M vertices), where LP
// refinement alone cannot recover bisection quality on mesh-like graphs.
// caps[k]: per-block hard weight caps (0 closes a block). Deterministic:
// serial, lazy max-heap ordered by (gain, smaller vertex id), target ties
// broken by smaller block id.
int kmp_kway_fm(
    const kmp_graph_t *g, u32 k, const i64 *caps, u32 *part,
    int max_passes, int max_fruitless
) {
  const u32 n = kmp_graph_n(g);
  const u32 *xadj = kmp_graph_xadj(g);
  const u32 *adjncy = kmp_graph_adjncy(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  const i32 *adjwgt = kmp_graph_adjwgt(g);
  if (max_passes <= 0) {
    max_passes = 3;
  }
  if (max_fruitless <= 0) {
    max_fruitless = 300;
  }

  std::vector<i64> bw(k, 0);
  for (u32 u = 0; u < n; ++u) {
    bw[part[u]] += vwgt ? vwgt[u] : 1;
  }

  // scratch connectivity (k-sized, cleared via touched list)
  std::vector<i64> conn(k, 0);
  std::vector<u32> touched;
  touched.reserve(64);

  // best feasible move for v under the CURRENT state: (gain, target) with
  // target among adjacent blocks, ties -> smaller block id. Returns false
  // if v has no adjacent block other than its own or no feasible target.
  auto best_move = [&](u32 v, i64 *gain_out, u32 *t_out) -> bool {
    const u32 b = part[v];
    const i64 wv = vwgt ? vwgt[v] : 1;
    touched.clear();
    for (u64 e = xadj[v]; e < xadj[v + 1]; ++e) {
      const u32 t = part[adjncy[e]];
      if (conn[t] == 0) {
        touched.push_back(t);
      }
      conn[t] += adjwgt ? adjwgt[e] : 1;
    }
    const i64 internal = conn[b];
    bool found = false;
    i64 bg = 0;
    u32 bt = 0;
    for (u32 t : touched) {
      if (t == b || bw[t] + wv > caps[t]) {
        continue;
      }
      const i64 gn = conn[t] - internal;
      if (!found || gn > bg || (gn == bg && t < bt)) {
        found = true;
        bg = gn;
        bt = t;
      }
    }
    for (u32 t : touched) {
      conn[t] = 0;
    }
    *gain_out = bg;
    *t_out = bt;
    return found;
  };

  constexpr i64 kNone = INT64_MIN;
  std::vector<i64> key(n);      // gain currently in the heap (kNone = out)
  std::vector<uint8_t> locked(n);
  std::vector<std::pair<i64, i64>> heap; // (gain, -v) max-heap
  struct Move {
    u32 v, from, to;
    i64 w;
  };
  std::vector<Move> moves;

  for (int pass = 0; pass < max_passes; ++pass) {
    std::fill(key.begin(), key.end(), kNone);
    std::fill(locked.begin(), locked.end(), 0);
    heap.clear();
    moves.clear();
    for (u32 v = 0; v < n; ++v) {
      bool boundary = false;
      for (u64 e = xadj[v]; e < xadj[v + 1] && !boundary; ++e) {
        boundary = part[adjncy[e]] != part[v];
      }
      if (!boundary) {
        continue;
      }
      i64 gn;
      u32 t;
      if (best_move(v, &gn, &t)) {
        key[v] = gn;
        heap.emplace_back(gn, -static_cast<i64>(v));
      }
    }
    std::make_heap(heap.begin(), heap.end());

    i64 cum = 0, best = 0;
    size_t best_len = 0;
    int fruitless = 0;
    while (!heap.empty() && fruitless < max_fruitless) {
      std::pop_heap(heap.begin(), heap.end());
      const i64 gn = heap.back().first;
      const u32 v = static_cast<u32>(-heap.back().second);
      heap.pop_back();
      if (locked[v] || key[v] != gn) {
        continue; // stale entry
      }
      i64 cur_g;
      u32 t;
      if (!best_move(v, &cur_g, &t)) {
        key[v] = kNone;
        continue;
      }
      if (cur_g != gn) {
        key[v] = cur_g;
        heap.emplace_back(cur_g, -static_cast<i64>(v));
        std::push_heap(heap.begin(), heap.end());
        continue;
      }
      // perform the move
      const u32 b = part[v];
      const i64 wv = vwgt ? vwgt[v] : 1;
      part[v] = t;
      bw[b] -= wv;
      bw[t] += wv;
      locked[v] = 1;
      key[v] = kNone;
      moves.push_back({v, b, t, wv});
      cum += cur_g;
      if (cum > best) {
        best = cum;
        best_len = moves.size();
        fruitless = 0;
      } else {
        ++fruitless;
      }
      // refresh unlocked neighbours (lazy: push the new exact key)
      for (u64 e = xadj[v]; e < xadj[v + 1]; ++e) {
        const u32 u = adjncy[e];
        if (locked[u]) {
          continue;
        }
        i64 ug;
        u32 ut;
        if (best_move(u, &ug, &ut)) {
          if (key[u] != ug) {
            key[u] = ug;
            heap.emplace_back(ug, -static_cast<i64>(u));
            std::push_heap(heap.begin(), heap.end());
          }
        } else {
          key[u] = kNone;
        }
      }
    }
    // roll back past the best prefix
    for (size_t i = moves.size(); i-- > best_len;) {
      const Move &m = moves[i];
      part[m.v] = m.from;
      bw[m.to] -= m.w;
      bw[m.from] += m.w;
    }
    if (best <= 0) {
      break;
    }
  }
  return 0;
}

// Gain-aware overload balancer on a host graph (uniform cap), exposed for
// the progressive-k driver.
int kmp_balance_partition(
    const kmp_graph_t *g, u32 k, i64 cap, u32 *part
) {
  const u32 n = kmp_graph_n(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  std::vector<i64> vw(n);
  for (u32 u = 0; u < n; ++u) {
    vw[u] = vwgt ? vwgt[u] : 1;
  }
  balance(kmp_graph_xadj(g), kmp_graph_adjncy(g), kmp_graph_adjwgt(g), vw,
          part, n, k, cap);
  return 0;
}

// O(m log n) bisection of a vertex subset (lazy-PQ greedy grow + FM);
// used for subgraphs beyond a few thousand vertices where the O(n^2)
// selection of kmp_bisect_subset would dominate. Deterministic.
int kmp_bisect_subset_fast(
    const kmp_graph_t *g, const u32 *nodes_in, u32 n_sub, i64 target1,
    i64 cap1, i64 cap2, int reps, uint8_t *side_out
) {
  const u32 n = kmp_graph_n(g);
  const u32 *xadj = kmp_graph_xadj(g);
  const u32 *adjncy = kmp_graph_adjncy(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  const i32 *adjwgt = kmp_graph_adjwgt(g);

  std::vector<i64> nodes(nodes_in, nodes_in + n_sub);
  std::vector<i64> loc(n, -1);
  for (u32 i = 0; i < n_sub; ++i) {
    loc[nodes[i]] = i;
  }
  SubCsr s = subgraph_csr(xadj, adjncy, adjwgt, nodes, loc);
  std::vector<i64> vw(n_sub);
  for (u32 i = 0; i < n_sub; ++i) {
    vw[i] = vwgt ? vwgt[nodes[i]] : 1;
  }

  std::vector<std::vector<uint8_t>> sides(reps);
  std::vector<i64> cuts(reps);
#pragma omp parallel for schedule(dynamic, 1) if (!omp_in_parallel())
  for (int rep = 0; rep < reps; ++rep) {
    sides[rep] = greedy_grow_fast(s, vw, target1, cap1, rep);
    fm_refine_fast(s, vw, sides[rep], cap1, cap2);
    cuts[rep] = bisection_cut2(s, sides[rep]);
  }
  int best = 0;
  for (int rep = 1; rep < reps; ++rep) {
    if (cuts[rep] < cuts[best]) {
      best = rep;
    }
  }
  for (u32 i = 0; i < n_sub; ++i) {
    side_out[i] = sides[best][i];
  }
  return 0;
}

// Multilevel bisection of a vertex subset (heavy-edge matching coarsening
// with FM at every level; best of `reps` matching orders). Higher quality
// than kmp_bisect_subset on subgraphs beyond a few hundred vertices; used
// by the progressive-k partition extension.
int kmp_bisect_subset_ml(
    const kmp_graph_t *g, const u32 *nodes_in, u32 n_sub, i64 target1,
    i64 cap1, i64 cap2, int reps, uint8_t *side_out
) {
  const u32 n = kmp_graph_n(g);
  const u32 *xadj = kmp_graph_xadj(g);
  const u32 *adjncy = kmp_graph_adjncy(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  const i32 *adjwgt = kmp_graph_adjwgt(g);

  std::vector<i64> nodes(nodes_in, nodes_in + n_sub);
  std::vector<i64> loc(n, -1);
  for (u32 i = 0; i < n_sub; ++i) {
    loc[nodes[i]] = i;
  }
  SubCsr s = subgraph_csr(xadj, adjncy, adjwgt, nodes, loc);
  std::vector<i64> vw(n_sub);
  for (u32 i = 0; i < n_sub; ++i) {
    vw[i] = vwgt ? vwgt[nodes[i]] : 1;
  }

  std::vector<std::vector<uint8_t>> sides(reps);
  std::vector<i64> cuts(reps);
#pragma omp parallel for schedule(dynamic, 1) if (!omp_in_parallel())
  for (int rep = 0; rep < reps; ++rep) {
    ml_bisect_rec(s, vw, target1, cap1, cap2, rep, sides[rep]);
    cuts[rep] = bisection_cut2(s, sides[rep]);
  }
  int best = 0;
  for (int rep = 1; rep < reps; ++rep) {
    if (cuts[rep] < cuts[best]) {
      best = rep;
    }
  }
  for (u32 i = 0; i < n_sub; ++i) {
    side_out[i] = sides[best][i];
  }
  return 0;
}

// Recursive-bisection initial partitioning on a (small) host graph.
// Equivalent to kaminpar_amd.partition.initial_partition (numpy); the
// equivalence is pinned by tests/test_pipeline_cpu.py.
int kmp_initial_partition(
    const kmp_graph_t *g, u32 k, i64 max_block_weight, int reps, u32 *part_out
) {
  const u32 n = kmp_graph_n(g);
  const u32 *xadj = kmp_graph_xadj(g);
  const u32 *adjncy = kmp_graph_adjncy(g);
  const i32 *vwgt = kmp_graph_vwgt(g);
  const i32 *adjwgt = kmp_graph_adjwgt(g);

  IpCtx c;
  c.xadj = xadj;
  c.adjncy = adjncy;
  c.adjwgt = adjwgt;
  c.vwgt.resize(n);
  for (u32 u = 0; u < n; ++u) {
    c.vwgt[u] = vwgt ? vwgt[u] : 1;
  }
  c.loc.assign(n, -1);
  c.mbw = max_block_weight;
  c.reps = reps;
  c.part = part_out;

  std::vector<i64> all(n);
  for (u32 u = 0; u < n; ++u) {
    all[u] = u;
  }
  rec(c, all, 0, k);
  balance(xadj, adjncy, adjwgt, c.vwgt, part_out, n, k, max_block_weight);
  return 0;
}

} // extern "C"
