// CPU oracle for the MI355X-native KaMinPar label-propagation hot path.
//
// This is a from-scratch C++ restatement of the reference LP semantics
// (gain accumulation, cluster selection, weight-constrained moves, active
// set, isolated-node and two-hop handling) of
//   /root/reference/kaminpar-shm/label_propagation.h        (engine)
//   /root/reference/kaminpar-shm/coarsening/clustering/lp_clusterer.cc:181-280
//       (clusterer select_best_cluster)
//   /root/reference/kaminpar-shm/refinement/lp/lp_refiner.cc:151-285
//       (refiner select_best_cluster)
//   /root/reference/kaminpar-shm/label_propagation.h:2139-2152
//       (move_cluster_weight), partitioned_graph.h:231-259 (move_block_weight)
// under the DETERMINISTIC CHUNK-SYNCHRONOUS SCHEDULE defined below, which is
// the parity contract between this oracle and the HIP implementation
// (kaminpar_amd). The reference's own asynchronous chunk-randomized schedule
// is inherently sequential (labels are read as they change mid-sweep); the
// deterministic schedule fixes a processing order that a GPU can reproduce
// bit-exactly. Quality parity with the reference algorithm proper is
// validated statistically against oracle/_ref (the reference itself compiled
// with serial TBB stubs) -- see tests/test_oracle_vs_ref.py.
//
// ======================= DETERMINISTIC SCHEDULE =======================
// Inputs: CSR graph (n, m, xadj u32[n+1], adjncy u32[m], optional vwgt i32[n],
// adjwgt i32[m]), initial labels, per-cluster max weights, seed, #iterations.
//
// perm:   vertices are grouped into UNITS of 64 consecutive ids (the
//         reference's own randomization granularity, kPermutationSize = 64,
//         label_propagation.h:52); the units are permuted by a 4-round
//         Feistel network keyed by mix(seed, iter) (cycle-walking; see
//         FeistelPerm below). Position p maps to vertex
//         u = feistel(p/64)*64 + p%64; u >= n is skipped. Consecutive ids in
//         a unit keep CSR reads coalesced on the GPU.
// chunks: 64 chunks per sweep over the position space [0, ceil(n/64)*64);
//         C = ceil(ceil(n/64)/64)*64 positions per chunk. 64 synchronous
//         commit points per sweep let label chains propagate (approximating
//         the reference's continuous asynchronous updates) while keeping GPU
//         launch overhead bounded.
// sweep (one iteration): for each chunk in order:
//   phase A (snapshot = state after the previous chunk's commit):
//     for every position p in the chunk, u = pi_iter(p):
//       skip if degree(u) > max_degree or !active[u];
//       arcs_scanned += degree(u); mark u processed;
//       accumulate ratings: map[label[v]] += w(u,v) over neighbours v
//         (label_propagation.h:487-505 semantics);
//       select best cluster (order-free restatement, below); if best !=
//       current, emit proposal (rank = p - chunk_start, u, from, to, w_u).
//   phase B (commit, greatest-fixpoint rollback): tentatively admit every
//     proposal, then iteratively trim each target cluster's arrivals (kept as
//     an ascending-rank prefix) until w_start(c) + admitted_arrivals(c)
//     - admitted_departures(c) <= max_weight(c) for every cluster. Monotone
//     (admissions only shrink), hence deterministic and order-free; restates
//     the optimistic-move + feasibility-check + rollback protocol of the
//     reference's distributed LP refiner
//     (kaminpar-dist/refinement/lp/lp_refiner.cc:296-333). The cap is never
//     overshot.
//     Apply admitted moves: labels, cluster weights (dest += w, src -= w);
//     clear active for all processed vertices, then set active[v] = 1 for
//     every neighbour v of every admitted mover (activate_neighbors,
//     label_propagation.h:848-870);
//     clusterer: decrement the live-cluster count for clusters whose weight
//     dropped to zero; stop the sweep early once count <= desired
//     (should_stop, label_propagation.h:260-265).
// Iterations stop early when a sweep commits zero moves
// (lp_clusterer.cc:94-105, lp_refiner.cc:80-86).
//
// Order-free cluster selection (tie-breaking): the reference's UNIFORM
// tie-breaking collects all max-gain candidates and picks uniformly at random
// (lp_clusterer.cc:238-248). The deterministic schedule replaces the random
// pick with an order-independent pseudo-random hash: pick the candidate
// maximizing the lexicographic key (rating, h(u, c)) among candidates passing
// the weight-acceptance predicate (refiner: lp_refiner.cc:185-230 accept
// rules; clusterer: lp_clusterer.cc:199-250). The refiner's secondary
// min-overload preference among gain ties is NOT reproduced: under snapshot
// semantics it is identical for every vertex in a chunk and funnels all tied
// vertices into one block (measured: cuts 20% worse); the hash tie-break
// matches the reference's cut quality (tests/test_oracle_vs_ref.py).
// The favored cluster (two-hop) is the (rating, h) argmax over ALL
// candidates. h(u, c) = splitmix64(mix(seed,iter) ^ u*0x9E3779B97F4A7C15 ^ c).
//
// Isolated nodes / two-hop (clusterer, defaults of presets.cc:140-153):
// described at their implementations below.
// ======================================================================
//
// This file is TEST INFRASTRUCTURE (the parity referee): only tests/,
// __graft_entry__.smoke() and bench.py's cpu_baseline leg may call it. It is
// never part of the product path.

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace {
// Cap the oracle's thread count: beyond ~32 threads the serial commit and
// NUMA effects dominate and 256-thread runs collapse (measured 15 M arcs/s
// at 256 threads vs 70 M at 8). KMP_ORACLE_THREADS overrides.
int oracle_threads() {
#ifdef _OPENMP
  static int n = [] {
    if (const char *env = getenv("KMP_ORACLE_THREADS")) {
      return atoi(env);
    }
    int hw = omp_get_max_threads();
    return hw > 32 ? 32 : hw;
  }();
  return n;
#else
  return 1;
#endif
}
} // namespace

namespace {

using u32 = uint32_t;
using u64 = uint64_t;
using i32 = int32_t;
using i64 = int64_t;

// ---------------------------------------------------------------- PRNG bits
inline u64 splitmix64(u64 x) {
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

inline u64 mix_seed(u64 seed, u64 salt) { return splitmix64(seed ^ (salt * 0xD1B54A32D192ED03ULL)); }

// Tie-breaking hash (must match kaminpar_amd/csrc exactly).
inline u64 tie_hash(u64 iter_seed, u32 u, u32 c) {
  return splitmix64(iter_seed ^ (static_cast<u64>(u) * 0x9E3779B97F4A7C15ULL) ^ c);
}

// 4-round Feistel permutation of [0, n) with cycle-walking.
// Bit width nb = smallest even number of bits covering n-1; half = nb/2.
struct FeistelPerm {
  u32 n;
  u32 half_bits;
  u32 half_mask;
  u64 keys[4];

  FeistelPerm(u32 n_, u64 seed) : n(n_) {
    u32 nb = 2;
    while ((1ULL << nb) < n) {
      nb += 2;
    }
    half_bits = nb / 2;
    half_mask = (1u << half_bits) - 1;
    for (int r = 0; r < 4; ++r) {
      keys[r] = splitmix64(seed ^ (0xA5A5A5A5ULL + r));
    }
  }

  inline u32 apply_once(u32 x) const {
    u32 l = x & half_mask;
    u32 r = (x >> half_bits) & half_mask;
    for (int i = 0; i < 4; ++i) {
      u32 nl = r;
      u32 nr = l ^ static_cast<u32>(splitmix64(keys[i] ^ r) & half_mask);
      l = nl;
      r = nr;
    }
    return (r << half_bits) | l;
  }

  inline u32 operator()(u32 p) const {
    u32 x = apply_once(p);
    while (x >= n) {
      x = apply_once(x);
    }
    return x;
  }
};

// Fixed chunk count (64 commit points per sweep).
constexpr u32 kNumChunks = 64;

// Block permutation over 64-vertex units (independent restatement of the
// schedule's BlockPerm; see kaminpar_amd/csrc/lp_common.h).
constexpr u32 kUnit = 64;

inline u32 num_units(u32 n) { return (n + kUnit - 1) / kUnit; }

inline u32 chunk_size_for_pos(u32 n) {
  const u32 nu = num_units(n);
  return ((nu + kNumChunks - 1) / kNumChunks) * kUnit;
}

inline u32 pos_count(u32 n) { return num_units(n) * kUnit; }

struct BlockPerm {
  FeistelPerm fp;
  u32 n;

  BlockPerm(u32 n_, u64 seed) : fp(num_units(n_), seed), n(n_) {}

  // maps position -> vertex id; result >= n means "no vertex" (skip)
  inline u32 operator()(u32 p) const { return fp(p / kUnit) * kUnit + (p % kUnit); }
};

// ---------------------------------------------------------------- Graph view
struct Csr {
  u32 n;
  u64 m;
  const u32 *xadj;
  const u32 *adjncy;
  const i32 *vwgt;   // may be null -> unit weights
  const i32 *adjwgt; // may be null -> unit weights

  inline u32 degree(u32 u) const { return xadj[u + 1] - xadj[u]; }
  inline i32 node_weight(u32 u) const { return vwgt ? vwgt[u] : 1; }
  inline i32 edge_weight(u64 e) const { return adjwgt ? adjwgt[e] : 1; }
};

// ------------------------------------------------------------- rating map
// Simple open-addressing map (cluster -> rating) used by the oracle for gain
// accumulation; restates the role of RatingMap/FixedSizeSparseMap
// (kaminpar-common/datastructures/rating_map.h:94, fixed_size_sparse_map.h:53)
// -- exact values matter, the container does not.
struct RatingMapOracle {
  std::vector<u32> keys;
  std::vector<i64> vals;
  std::vector<u32> used;
  u32 mask = 0;

  void reserve(u32 max_entries) {
    u32 cap = 16;
    while (cap < max_entries * 2) {
      cap <<= 1;
    }
    if (keys.size() < cap) {
      keys.assign(cap, 0xFFFFFFFFu);
      vals.assign(cap, 0);
    }
    mask = cap - 1;
  }

  inline void add(u32 c, i64 w) {
    u32 slot = static_cast<u32>(splitmix64(c)) & mask;
    while (true) {
      if (keys[slot] == c) {
        vals[slot] += w;
        return;
      }
      if (keys[slot] == 0xFFFFFFFFu) {
        keys[slot] = c;
        vals[slot] = w;
        used.push_back(slot);
        return;
      }
      slot = (slot + 1) & mask;
    }
  }

  inline void clear() {
    for (u32 slot : used) {
      keys[slot] = 0xFFFFFFFFu;
      vals[slot] = 0;
    }
    used.clear();
  }
};

// ------------------------------------------------------------ proposals
struct Proposal {
  u32 rank; // position within chunk (admission order)
  u32 u;
  u32 from;
  u32 to; // 0xFFFFFFFF marks an empty slot (parallel phase A)
  i32 w;
};

// ------------------------------------------------------- LP engine (oracle)
struct LpParams {
  u32 n;
  i64 uniform_max_weight = 0;       // clusterer: uniform cap
  const i64 *max_weights = nullptr; // refiner: per-block caps (len k)
  u64 seed = 1;
  int iters = 5;
  u32 max_degree = 0xFFFFFFFFu;
  u32 desired_clusters = 0; // clusterer stop threshold (0 = never)
  bool clusterer = false;   // select semantics variant
  bool balance = false;     // overloaded vertices lose "stay" (balancer mode)
  bool underload = false;   // underload-balancer mode (min weights set)
  const i64 *min_weights = nullptr; // per-block minimums (underload mode)
  const u32 *communities = nullptr; // clusterer: merges stay within community
                                    // (clusterer.h:35, lp_clusterer.cc:193)
  u32 k = 0;                // number of clusters (refiner: k; clusterer: n)
};

struct LpStats {
  u64 arcs_scanned = 0;
  u64 moves = 0;
  u32 num_nonempty_clusters = 0;
};


// One full deterministic LP run (shared by clusterer and refiner).
// labels: in/out, len n. weights: in/out cluster weights, len k.
// favored: optional out (clusterer two-hop), len n.
void lp_run(
    const Csr &g,
    const LpParams &par,
    u32 *labels,
    i64 *weights,
    u32 *favored,
    std::vector<uint8_t> &active,
    LpStats &stats,
    u32 *live_clusters_io // clusterer: in/out live cluster count (null for refiner)
) {
  const u32 n = g.n;
  const u32 C = chunk_size_for_pos(n);
  const u32 P = pos_count(n);
  const u32 num_chunks = (P + C - 1) / C;

  std::vector<Proposal> proposals;
  std::vector<Proposal> slots; // per-position results (parallel phase A)
  std::vector<u32> processed;

  u32 live_clusters = live_clusters_io ? *live_clusters_io : 0;

  for (int iter = 0; iter < par.iters; ++iter) {
    const u64 iter_seed = mix_seed(par.seed, 0x17E5ULL + static_cast<u64>(iter));
    BlockPerm perm(n, iter_seed);
    u64 sweep_moves = 0;
    bool stopped = false;

    for (u32 chunk = 0; chunk < num_chunks && !stopped; ++chunk) {
      const u32 pos_begin = chunk * C;
      const u32 pos_end = std::min<u64>(static_cast<u64>(pos_begin) + C, P);

      proposals.clear();
      processed.clear();
      slots.assign(pos_end - pos_begin, Proposal{0, 0, 0, 0xFFFFFFFFu, 0});

      // balance mode: per-chunk fallback target = lightest block with room
      // (keep in sync with kmp_lp_phase_a on the device)
      u32 balance_fallback = 0xFFFFFFFFu;
      if (par.balance) {
        i64 bw = -1;
        for (u32 c = 0; c < par.k; ++c) {
          if (weights[c] < par.max_weights[c] && (bw < 0 || weights[c] < bw)) {
            bw = weights[c];
            balance_fallback = c;
          }
        }
      }

      // ---- phase A: gains + selection against the chunk-start snapshot ----
      // (parallel over positions; results land in per-position slots so the
      // compacted order is deterministic, mirroring the GPU design)
#ifdef _OPENMP
// parallelism only pays for large chunks; fork/join + spin-wait overhead
// dominates on small ones (measured 300x slowdown on 4K-vertex graphs)
#pragma omp parallel for schedule(dynamic, 256) num_threads(oracle_threads()) \
    if (pos_end - pos_begin >= 65536)
#endif
      for (long long pp = pos_begin; pp < static_cast<long long>(pos_end); ++pp) {
        const u32 p = static_cast<u32>(pp);
        static thread_local RatingMapOracle map;
        const u32 u = perm(p);
        if (u >= n) {
          continue; // tail of the last unit
        }
        const u32 deg = g.degree(u);
        if (deg > par.max_degree) {
          continue;
        }
        if (!active[u]) {
          continue;
        }

        const u32 u_cluster = labels[u];
        const i32 u_weight = g.node_weight(u);
        const i64 init_weight = weights[u_cluster];

        // Underload mode (underload_balancer.cc is_movable_from): only
        // vertices whose block is NOT underloaded and stays above its
        // minimum may move.
        if (par.underload) {
          const i64 mnw = par.min_weights[u_cluster];
          if (init_weight < mnw || init_weight - u_weight < mnw) {
            continue;
          }
        }

        map.reserve(std::min<u32>(deg, par.k) + 2);
        const u64 row_begin = g.xadj[u];
        const u64 row_end = g.xadj[u + 1];
        for (u64 e = row_begin; e < row_end; ++e) {
          map.add(labels[g.adjncy[e]], g.edge_weight(e));
        }

        // ---- order-free select_best_cluster ----
        u32 best = u_cluster;
        i64 best_gain = 0;
        i64 best_over = 0;
        u64 best_h = 0;
        bool have = false;

        u32 fav = u_cluster;
        i64 fav_gain = 0;
        u64 fav_h = 0;

        const bool store_favored =
            par.clusterer && favored != nullptr && u_weight == init_weight &&
            init_weight <= par.uniform_max_weight / 2;

        for (u32 slot : map.used) {
          const u32 c = map.keys[slot];
          const i64 r = map.vals[slot];
          const u64 h = tie_hash(iter_seed, u, c);

          // community filter (lp_clusterer.cc:193-194): cluster ids are
          // vertex ids, so communities[] indexes both
          if (par.communities != nullptr &&
              par.communities[c] != par.communities[u_cluster]) {
            continue;
          }

          if (store_favored) {
            if (r > fav_gain || (r == fav_gain && (h > fav_h || (h == fav_h && c < fav)))) {
              fav_gain = r;
              fav = c;
              fav_h = h;
            }
          }

          const i64 cw = weights[c];
          const i64 maxw = par.clusterer ? par.uniform_max_weight : par.max_weights[c];
          bool accept;
          i64 over = 0;
          if (par.clusterer) {
            accept = (cw + u_weight <= maxw) || (c == u_cluster);
          } else if (par.underload) {
            // underload_balancer.cc is_movable_to: only underloaded targets
            // with room; the current block is never a candidate
            accept = c != u_cluster && cw < par.min_weights[c] &&
                     cw + u_weight <= maxw;
          } else if (c == u_cluster) {
            // balance mode: a vertex in an over-cap block loses "stay"
            accept = !(par.balance &&
                       init_weight > par.max_weights[u_cluster]);
          } else if (par.balance) {
            // room-only in balance mode (matches accept_refine on device)
            accept = cw + u_weight <= maxw;
          } else {
            over = cw - maxw;
            const i64 init_over = init_weight - par.max_weights[u_cluster];
            accept = (cw + u_weight <= maxw) || (over < init_over);
          }
          if (!accept) {
            continue;
          }

          // Tie key: (gain, hash). The reference's refiner additionally
          // prefers lower overload among gain ties (lp_refiner.cc:201-229);
          // under snapshot semantics that preference is identical for every
          // vertex in a chunk and funnels all tied vertices into one block,
          // so the deterministic schedule replaces it with the pseudo-random
          // hash (the UNIFORM tie-breaking spirit: uniform among max-gain).
          bool better;
          if (!have) {
            better = true;
          } else if (r != best_gain) {
            better = r > best_gain;
          } else if (h != best_h) {
            better = h > best_h;
          } else {
            better = c < best; // total order (matches device key_better)
          }
          if (better) {
            best = c;
            best_gain = r;
            best_over = over;
            best_h = h;
            have = true;
          }
        }
        map.clear();

        if (store_favored) {
          favored[u] = fav;
        }

        if (!have && par.balance &&
            init_weight > par.max_weights[u_cluster] &&
            balance_fallback != 0xFFFFFFFFu && balance_fallback != u_cluster) {
          // no admissible adjacent target: shed to the chunk's fallback
          best = balance_fallback;
          have = true;
        }
        if (have && best != u_cluster) {
          slots[p - pos_begin] = Proposal{p - pos_begin, u, u_cluster, best, u_weight};
        }
      }
      // compact valid slots (position order) + arcs/processed bookkeeping
      for (u32 p = pos_begin; p < pos_end; ++p) {
        if (slots[p - pos_begin].to != 0xFFFFFFFFu) {
          proposals.push_back(slots[p - pos_begin]);
        }
      }
      for (u32 p = pos_begin; p < pos_end; ++p) {
        const u32 u = perm(p);
        if (u >= n) {
          continue;
        }
        const u32 deg = g.degree(u);
        if (deg > par.max_degree || !active[u]) {
          continue;
        }
        processed.push_back(u);
        stats.arcs_scanned += deg;
      }

      // ---- phase B (underload mode): serial rank-order admission ----
      // The underload balancer must respect BOTH per-block minima (source
      // side) and maxima (target side); instead of a 2-dimensional
      // fixpoint, admission is sequential in deterministic rank order with
      // both caps re-checked against live weights (the batch analogue of
      // underload_balancer.cc's locked per-move checks; proposal counts in
      // this mode are small). Targets are re-checked to still be
      // underloaded at admission time (is_movable_to semantics).
      if (par.underload) {
        std::vector<const Proposal *> admitted;
        admitted.reserve(proposals.size());
        for (const Proposal &pr : proposals) {
          const i32 w = pr.w;
          const i64 mn_from = par.min_weights[pr.from];
          const i64 mn_to = par.min_weights[pr.to];
          if (weights[pr.from] >= mn_from && weights[pr.from] - w >= mn_from &&
              weights[pr.to] < mn_to &&
              weights[pr.to] + w <= par.max_weights[pr.to]) {
            weights[pr.from] -= w;
            weights[pr.to] += w;
            labels[pr.u] = pr.to;
            admitted.push_back(&pr);
          }
        }
        for (u32 u : processed) {
          active[u] = 0;
        }
        for (const Proposal *pr : admitted) {
          const u64 row_begin = g.xadj[pr->u];
          const u64 row_end = g.xadj[pr->u + 1];
          for (u64 e = row_begin; e < row_end; ++e) {
            active[g.adjncy[e]] = 1;
          }
        }
        sweep_moves += admitted.size();
        stats.moves += admitted.size();
        continue;
      }

      // ---- phase B: deterministic commit (greatest-fixpoint rollback) ----
      // Tentatively admit every proposal, then iteratively trim each target
      // cluster's arrivals (kept as an ascending-rank prefix) until every
      // cluster respects its cap given the departures that still happen.
      // This restates the optimistic-move + feasibility-check + rollback
      // protocol of the reference's distributed LP refiner
      // (kaminpar-dist/refinement/lp/lp_refiner.cc:296-333) as a monotone
      // fixpoint: admissions only shrink per round, so the result is
      // deterministic and order-free, and the cap is never overshot.
      std::sort(proposals.begin(), proposals.end(), [](const Proposal &a, const Proposal &b) {
        return a.to != b.to ? a.to < b.to : a.rank < b.rank;
      });

      const size_t np = proposals.size();
      std::vector<uint8_t> admitted_flag(np, 1);
      // Arrival segments: [seg_begin[i], seg_end[i]) in `proposals` per target.
      std::vector<std::pair<size_t, size_t>> segs;
      for (size_t i = 0; i < np;) {
        size_t j = i;
        while (j < np && proposals[j].to == proposals[i].to) {
          ++j;
        }
        segs.emplace_back(i, j);
        i = j;
      }
      // Per-segment admitted prefix length (in proposals); starts at full.
      std::vector<size_t> prefix_len(segs.size());
      for (size_t s = 0; s < segs.size(); ++s) {
        prefix_len[s] = segs[s].second - segs[s].first;
      }

      bool changed = true;
      while (changed) {
        changed = false;
        // Departures of currently-admitted proposals, per source cluster.
        // (Sparse accumulation keyed by source cluster.)
        RatingMapOracle dep;
        dep.reserve(static_cast<u32>(std::min<size_t>(np, 1u << 20)) + 2);
        for (size_t i = 0; i < np; ++i) {
          if (admitted_flag[i]) {
            dep.add(proposals[i].from, proposals[i].w);
          }
        }
        auto dep_of = [&](u32 c) -> i64 {
          u32 slot = static_cast<u32>(splitmix64(c)) & dep.mask;
          while (true) {
            if (dep.keys[slot] == c) {
              return dep.vals[slot];
            }
            if (dep.keys[slot] == 0xFFFFFFFFu) {
              return 0;
            }
            slot = (slot + 1) & dep.mask;
          }
        };

        for (size_t s = 0; s < segs.size(); ++s) {
          const auto [b, e] = segs[s];
          const u32 c = proposals[b].to;
          const i64 maxw = par.clusterer ? par.uniform_max_weight : par.max_weights[c];
          const i64 capacity = maxw - weights[c] + dep_of(c);
          i64 acc = 0;
          size_t t = 0;
          while (t < prefix_len[s] && acc + proposals[b + t].w <= capacity) {
            acc += proposals[b + t].w;
            ++t;
          }
          if (t < prefix_len[s]) {
            for (size_t i = b + t; i < b + prefix_len[s]; ++i) {
              admitted_flag[i] = 0;
            }
            prefix_len[s] = t;
            changed = true;
          }
        }
      }

      std::vector<const Proposal *> admitted;
      admitted.reserve(np);
      for (size_t i = 0; i < np; ++i) {
        if (admitted_flag[i]) {
          admitted.push_back(&proposals[i]);
        }
      }
      if (getenv("KMP_ORACLE_DEBUG")) {
        fprintf(
            stderr,
            "  chunk %u: processed %zu proposed %zu admitted %zu\n",
            chunk,
            processed.size(),
            np,
            admitted.size()
        );
      }

      // Clear active for processed vertices first, then activate neighbours
      // of admitted movers.
      for (u32 u : processed) {
        active[u] = 0;
      }

      // Record which clusters might empty (clusterer bookkeeping).
      for (const Proposal *pr : admitted) {
        labels[pr->u] = pr->to;
        weights[pr->to] += pr->w;
        weights[pr->from] -= pr->w;
      }
      for (const Proposal *pr : admitted) {
        const u64 row_begin = g.xadj[pr->u];
        const u64 row_end = g.xadj[pr->u + 1];
        for (u64 e = row_begin; e < row_end; ++e) {
          active[g.adjncy[e]] = 1;
        }
      }
      if (live_clusters_io) {
        // Count each emptied source cluster exactly once (dedupe the `from`
        // list; several movers can leave the same cluster in one chunk).
        std::vector<u32> froms;
        froms.reserve(admitted.size());
        for (const Proposal *pr : admitted) {
          froms.push_back(pr->from);
        }
        std::sort(froms.begin(), froms.end());
        froms.erase(std::unique(froms.begin(), froms.end()), froms.end());
        for (u32 c : froms) {
          if (weights[c] == 0) {
            --live_clusters;
          }
        }
      }
      sweep_moves += admitted.size();
      stats.moves += admitted.size();

      if (live_clusters_io && par.desired_clusters > 0 && live_clusters <= par.desired_clusters) {
        stopped = true;
      }
    }

    if (sweep_moves == 0) {
      break;
    }
  }

  if (live_clusters_io) {
    *live_clusters_io = live_clusters;
  }
}

} // namespace

// ======================================================================
// C API (ctypes-friendly). Only tests / bench cpu_baseline may call this.
// ======================================================================
extern "C" {

// Expose the Feistel permutation for cross-checking against the HIP side.
// out must have pos_count(n) = ceil(n/64)*64 entries; entries >= n mark
// skipped tail positions.
void kmp_oracle_perm(u32 n, u64 seed, int iter, u32 *out) {
  const u64 iter_seed = mix_seed(seed, 0x17E5ULL + static_cast<u64>(iter));
  BlockPerm perm(n, iter_seed);
  const u32 P = pos_count(n);
  for (u32 p = 0; p < P; ++p) {
    out[p] = perm(p);
  }
}

i64 kmp_oracle_edge_cut(
    u32 n, [[maybe_unused]] u64 m, const u32 *xadj, const u32 *adjncy, const i32 *adjwgt, const u32 *labels
) {
  i64 cut = 0;
  for (u32 u = 0; u < n; ++u) {
    for (u64 e = xadj[u]; e < xadj[u + 1]; ++e) {
      if (labels[u] != labels[adjncy[e]]) {
        cut += adjwgt ? adjwgt[e] : 1;
      }
    }
  }
  return cut / 2;
}

// Deterministic LP refinement. partition: in/out. max_block_weights: len k.
// Returns final edge cut. stats_out (len 3): arcs_scanned, moves, 0.
i64 kmp_oracle_lp_refine(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    u32 k,
    const i64 *max_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    u64 *stats_out
) {
  Csr g{n, m, xadj, adjncy, vwgt, adjwgt};

  LpParams par;
  par.n = n;
  par.max_weights = max_block_weights;
  par.seed = seed;
  par.iters = iters;
  par.clusterer = false;
  par.k = k;

  std::vector<i64> weights(k, 0);
  for (u32 u = 0; u < n; ++u) {
    weights[partition[u]] += g.node_weight(u);
  }

  std::vector<uint8_t> active(n, 1);
  LpStats stats;
  lp_run(g, par, partition, weights.data(), nullptr, active, stats, nullptr);

  if (stats_out) {
    stats_out[0] = stats.arcs_scanned;
    stats_out[1] = stats.moves;
    stats_out[2] = 0;
  }
  return kmp_oracle_edge_cut(n, m, xadj, adjncy, adjwgt, partition);
}

// Balancer-mode twin of kmp_oracle_lp_refine (see kmp_lp_balance in
// include/kaminpar_lp.h): overloaded vertices lose "stay" in selection.
i64 kmp_oracle_lp_balance(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    u32 k,
    const i64 *max_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    u64 *stats_out
) {
  Csr g{n, m, xadj, adjncy, vwgt, adjwgt};

  LpParams par;
  par.n = n;
  par.max_weights = max_block_weights;
  par.seed = seed;
  par.iters = iters;
  par.clusterer = false;
  par.balance = true;
  par.k = k;

  std::vector<i64> weights(k, 0);
  for (u32 u = 0; u < n; ++u) {
    weights[partition[u]] += g.node_weight(u);
  }

  // isolated pre-pass (keep in sync with kmp_lp_balance): deg-0 vertices in
  // over-cap blocks go to the lightest block with room, ascending id order
  for (u32 u = 0; u < n; ++u) {
    if (g.degree(u) != 0) {
      continue;
    }
    const u32 b = partition[u];
    if (weights[b] <= max_block_weights[b]) {
      continue;
    }
    const i32 uw = g.node_weight(u);
    i64 best = -1;
    u32 t = b;
    for (u32 c = 0; c < k; ++c) {
      if (c != b && weights[c] + uw <= max_block_weights[c] &&
          (best < 0 || weights[c] < best)) {
        best = weights[c];
        t = c;
      }
    }
    if (t != b) {
      weights[b] -= uw;
      weights[t] += uw;
      partition[u] = t;
    }
  }

  std::vector<uint8_t> active(n, 1);
  LpStats stats;
  lp_run(g, par, partition, weights.data(), nullptr, active, stats, nullptr);

  if (stats_out) {
    stats_out[0] = stats.arcs_scanned;
    stats_out[1] = stats.moves;
    stats_out[2] = 0;
  }
  return kmp_oracle_edge_cut(n, m, xadj, adjncy, adjwgt, partition);
}

// Underload-balancer mode (the role of the reference's UNDERLOAD_BALANCER
// closing the default refiner chain, presets.cc:332-338; semantics restated
// from refinement/balancer/underload_balancer.cc): fill blocks below their
// minimum weight with best-gain admissible vertices, never dropping any
// source below its own minimum and never overshooting any maximum.
i64 kmp_oracle_lp_underload(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    u32 k,
    const i64 *max_block_weights,
    const i64 *min_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    u64 *stats_out
) {
  Csr g{n, m, xadj, adjncy, vwgt, adjwgt};

  LpParams par;
  par.n = n;
  par.max_weights = max_block_weights;
  par.min_weights = min_block_weights;
  par.seed = seed;
  par.iters = iters;
  par.clusterer = false;
  par.underload = true;
  par.k = k;

  std::vector<i64> weights(k, 0);
  for (u32 u = 0; u < n; ++u) {
    weights[partition[u]] += g.node_weight(u);
  }

  std::vector<uint8_t> active(n, 1);
  LpStats stats;
  lp_run(g, par, partition, weights.data(), nullptr, active, stats, nullptr);

  if (stats_out) {
    stats_out[0] = stats.arcs_scanned;
    stats_out[1] = stats.moves;
    stats_out[2] = 0;
  }
  return kmp_oracle_edge_cut(n, m, xadj, adjncy, adjwgt, partition);
}

// Deterministic LP clustering (coarsening instantiation;
// lp_clusterer.cc:89-109 driver semantics + isolated/two-hop passes).
// clustering: out, len n. Returns number of non-empty clusters.
i64 kmp_oracle_lp_cluster_comm(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    i64 max_cluster_weight,
    u32 desired_clusters,
    const u32 *communities, // null = unrestricted (clusterer.h:35)
    u32 *clustering,
    u64 seed,
    int iters,
    u64 *stats_out
) {
  Csr g{n, m, xadj, adjncy, vwgt, adjwgt};

  LpParams par;
  par.n = n;
  par.uniform_max_weight = max_cluster_weight;
  par.seed = seed;
  par.iters = iters;
  par.clusterer = true;
  par.k = n;
  par.desired_clusters = desired_clusters;
  par.communities = communities;

  std::vector<i64> weights(n);
  std::vector<u32> favored(n);
  for (u32 u = 0; u < n; ++u) {
    clustering[u] = u;                  // initial_cluster (lp_clusterer.cc:169)
    weights[u] = g.node_weight(u);      // initial_cluster_weight (:173)
    favored[u] = u;
  }

  std::vector<uint8_t> active(n, 1);
  LpStats stats;
  u32 live_clusters = n;
  lp_run(g, par, clustering, weights.data(), favored.data(), active, stats, &live_clusters);

  // ---- isolated nodes + two-hop handling, default preset strategies ----
  // (presets.cc:147-152: isolated = MATCH_DURING_TWO_HOP, two-hop =
  // MATCH_THREADWISE at threshold 0.5.)
  const bool handle_two_hop =
      (1.0 - 1.0 * live_clusters / n) <= 0.5; // lp_clusterer.cc:164-166

  if (handle_two_hop) {
    // Isolated nodes, MATCH semantics (label_propagation.h:884-917):
    // deterministic sequential chain over isolated nodes in ascending order;
    // merge the current node's cluster into the pending node's cluster when
    // the cap permits, then reset (pairs).
    u32 pending = 0xFFFFFFFFu;
    for (u32 u = 0; u < n; ++u) {
      if (g.degree(u) != 0) {
        continue;
      }
      const u32 cu = clustering[u];
      // communities are a hard constraint: isolated chains break at
      // community boundaries (strengthens the reference engine, whose
      // isolated handler predates set_communities)
      if (pending != 0xFFFFFFFFu && communities != nullptr &&
          communities[pending] != communities[cu]) {
        pending = cu;
        continue;
      }
      if (pending != 0xFFFFFFFFu && weights[pending] + weights[cu] <= max_cluster_weight) {
        weights[pending] += weights[cu];
        weights[cu] = 0;
        clustering[u] = pending;
        --live_clusters;
        pending = 0xFFFFFFFFu; // match mode: pair formed, reset
      } else {
        pending = cu;
      }
    }

    // Two-hop matching, MATCH_THREADWISE semantics at one thread
    // (label_propagation.h:931-1016): nodes still in their own singleton
    // cluster are grouped by favored cluster; consecutive candidates in
    // ascending node order are paired (second joins the first's cluster).
    auto considered = [&](u32 u) {
      if (g.degree(u) == 0 || clustering[u] != u) {
        return false;
      }
      const i64 w = weights[u];
      return w <= max_cluster_weight / 2 && w == g.node_weight(u);
    };

    // rep[f] = pending representative node for favored cluster f + 1 (0 = none)
    std::vector<u32> rep(n, 0);
    for (u32 u = 0; u < n; ++u) {
      if (!considered(u)) {
        continue;
      }
      const u32 f = favored[u];
      if (rep[f] == 0) {
        rep[f] = u + 1;
      } else {
        const u32 r = rep[f] - 1;
        const u32 cr = clustering[r];
        if (weights[cr] + weights[u] <= max_cluster_weight) {
          weights[cr] += weights[u];
          weights[u] = 0;
          clustering[u] = cr;
          --live_clusters;
          rep[f] = 0; // match mode: pair formed, reset
        } else {
          rep[f] = u + 1;
        }
      }
    }
  }

  // Exact final count of non-empty clusters (the tracked count is only used
  // for the should_stop heuristic during the sweeps).
  u64 nonempty = 0;
  for (u32 c = 0; c < n; ++c) {
    nonempty += (weights[c] != 0);
  }

  if (stats_out) {
    stats_out[0] = stats.arcs_scanned;
    stats_out[1] = stats.moves;
    stats_out[2] = nonempty;
  }
  return static_cast<i64>(nonempty);
}

} // extern "C"

// ======================================================================
// Cluster contraction (restates kaminpar-shm/coarsening/contraction/
// cluster_contraction_preprocessing.cc:17-52 mapping semantics -- coarse id
// = prefix rank of occupied cluster ids -- and the coarse-graph
// construction of cluster_contraction.cc: intra-cluster edges dropped,
// parallel edges merged with summed weights, node weights summed).
// Coarse adjacency is emitted sorted by (coarse_u, coarse_v): a canonical
// order shared with the GPU implementation (the reference's own adjacency
// order is scheduling-dependent; sorted comparison is used for pinning).
// ======================================================================
extern "C" {

// Returns coarse node count; fills mapping[n], c_xadj[c_n+1], c_adjncy,
// c_vwgt, c_adjwgt (buffers sized for the fine graph are always enough).
// c_m_out receives the coarse arc count.
i64 kmp_oracle_lp_cluster(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    i64 max_cluster_weight,
    u32 desired_clusters,
    u32 *clustering,
    u64 seed,
    int iters,
    u64 *stats_out
) {
  return kmp_oracle_lp_cluster_comm(
      n, m, xadj, adjncy, vwgt, adjwgt, max_cluster_weight, desired_clusters, nullptr,
      clustering, seed, iters, stats_out
  );
}

i64 kmp_oracle_contract(
    u32 n,
    u64 m,
    const u32 *xadj,
    const u32 *adjncy,
    const i32 *vwgt,
    const i32 *adjwgt,
    const u32 *clustering,
    u32 *mapping,
    u32 *c_xadj,
    u32 *c_adjncy,
    i32 *c_vwgt,
    i32 *c_adjwgt,
    u64 *c_m_out
) {
  // coarse ids: prefix rank of occupied cluster ids
  std::vector<u32> rank(n, 0);
  for (u32 u = 0; u < n; ++u) {
    rank[clustering[u]] = 1;
  }
  u32 acc = 0;
  for (u32 c = 0; c < n; ++c) {
    const u32 occ = rank[c];
    rank[c] = acc;
    acc += occ;
  }
  const u32 c_n = acc;
  for (u32 u = 0; u < n; ++u) {
    mapping[u] = rank[clustering[u]];
  }

  // coarse node weights
  for (u32 c = 0; c < c_n; ++c) {
    c_vwgt[c] = 0;
  }
  for (u32 u = 0; u < n; ++u) {
    c_vwgt[mapping[u]] += vwgt ? vwgt[u] : 1;
  }

  // inter-cluster arcs, merged and sorted by (cu, cv)
  std::vector<std::pair<u64, i64>> arcs;
  arcs.reserve(m);
  for (u32 u = 0; u < n; ++u) {
    const u32 cu = mapping[u];
    for (u64 e = xadj[u]; e < xadj[u + 1]; ++e) {
      const u32 cv = mapping[adjncy[e]];
      if (cu != cv) {
        arcs.emplace_back((static_cast<u64>(cu) << 32) | cv, adjwgt ? adjwgt[e] : 1);
      }
    }
  }
  std::sort(arcs.begin(), arcs.end(), [](const auto &a, const auto &b) {
    return a.first < b.first;
  });

  u64 c_m = 0;
  for (u32 c = 0; c <= c_n; ++c) {
    c_xadj[c] = 0;
  }
  for (size_t i = 0; i < arcs.size();) {
    const u64 key = arcs[i].first;
    i64 w = 0;
    while (i < arcs.size() && arcs[i].first == key) {
      w += arcs[i].second;
      ++i;
    }
    c_adjncy[c_m] = static_cast<u32>(key & 0xFFFFFFFFu);
    c_adjwgt[c_m] = static_cast<i32>(w);
    ++c_xadj[(key >> 32) + 1];
    ++c_m;
  }
  for (u32 c = 0; c < c_n; ++c) {
    c_xadj[c + 1] += c_xadj[c];
  }
  *c_m_out = c_m;
  return static_cast<i64>(c_n);
}

} // extern "C"
