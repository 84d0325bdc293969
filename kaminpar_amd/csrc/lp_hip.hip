// MI355X (gfx950) HIP implementation of the KaMinPar label-propagation hot
// path under the deterministic chunk-synchronous schedule (parity contract:
// oracle/lp_oracle.cpp header). Semantics restated from
//   kaminpar-shm/label_propagation.h (engine: gains :487-505, active set
//   :848-870,1908-1912, try_node_move :817-841)
//   kaminpar-shm/refinement/lp/lp_refiner.cc:151-285 (refiner select)
//   kaminpar-shm/coarsening/clustering/lp_clusterer.cc:181-280 (clusterer
//   select), with the commit fixpoint mirroring
//   kaminpar-dist/refinement/lp/lp_refiner.cc:296-333 (rollback protocol).
//
// Kernel design (CDNA4): irregular integer gather workload; the roofline
// bound is HBM bandwidth (8 B per directed arc: 4 B adjncy + 4 B labels
// gather), not MFMA. The schedule's 64-vertex units keep per-vertex state
// reads coalesced. Phase A uses NO append atomics: every position owns a
// 16-byte proposal slot, written by exactly one kernel (S: deg<=16 with
// 16-lane subgroup shuffle-waterfall gains; M: one wavefront per vertex with
// dense per-cluster LDS gains for k<=2048; L: one workgroup per high-degree
// vertex), then rocprim::select compacts valid slots in position order
// (stable), so the commit's stable 32-bit radix sort by target cluster
// yields the deterministic (to, rank) admission order.

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string.h>
#include <vector>

#include <hip/hip_runtime.h>

#include <rocprim/rocprim.hpp>

#include <rccl/rccl.h>

#include "../../include/kaminpar_lp.h"
#include "lp_common.h"

using kmp::i32;
using kmp::i64;
using kmp::u32;
using kmp::u64;
using kmp::BlockPerm;
using kmp::iter_seed_of;
using kmp::tie_hash;

// Check the sticky error right after a kernel launch (configuration errors
// surface there, not at the launch statement).
#define LAUNCH_CHECK()                                                                             \
  do {                                                                                             \
    hipError_t lerr_ = hipGetLastError();                                                          \
    if (lerr_ != hipSuccess) {                                                                     \
      fprintf(stderr, "kaminpar_amd: launch error %s at %s:%d\n", hipGetErrorString(lerr_),       \
              __FILE__, __LINE__);                                                                 \
      abort();                                                                                     \
    }                                                                                              \
  } while (0)

#define HIP_CHECK(cmd)                                                                             \
  do {                                                                                             \
    hipError_t err_ = (cmd);                                                                       \
    if (err_ != hipSuccess) {                                                                      \
      fprintf(stderr, "kaminpar_amd: HIP error %s at %s:%d\n", hipGetErrorString(err_), __FILE__,  \
              __LINE__);                                                                           \
      abort();                                                                                     \
    }                                                                                              \
  } while (0)

namespace {

constexpr u32 kSmallDeg = 16;    // S path: <= 16 neighbours, 16 lanes/vertex
constexpr u32 kMidDeg = 2048;    // M path: one wavefront per vertex
constexpr u32 kMaxDenseK = 2048; // dense per-cluster LDS gains limit (refine)
constexpr u32 kWave = 64;
constexpr u32 kInvalid = 0xFFFFFFFFu;

struct Prop { // 16-byte proposal record (ABI: uint32x4)
  u32 u;
  u32 to;   // kInvalid marks an empty slot (pre-compaction)
  u32 rank; // position - chunk_base (admission order within the chunk)
  u32 w;    // node weight bits (i32 >= 0)
};

struct PropValid {
  __host__ __device__ bool operator()(const Prop &p) const { return p.to != kInvalid; }
};

// ------------------------------------------------- select helpers (device)
struct BestState {
  i32 gain;
  u64 h;
  u32 c;
  bool have;
};

__device__ inline bool key_better(i32 g, u64 h, u32 c, const BestState &b) {
  if (!b.have) {
    return true;
  }
  if (g != b.gain) {
    return g > b.gain;
  }
  if (h != b.h) {
    return h > b.h;
  }
  return c < b.c;
}

// Gain accumulation uses REPLICATED per-cluster LDS counters: lane l adds
// into replica l % R, so same-address LDS atomic serialization (the dominant
// cost with a single counter per cluster) is cut by ~R. The replicas are
// merged once per vertex. R is chosen so k * R stays within the LDS budget.
__host__ __device__ inline u32 gain_replicas(u32 k) {
  if (k <= 64) {
    return 8;
  }
  if (k <= 256) {
    return 4;
  }
  return 1;
}

// Weight-acceptance predicate (refiner variant, lp_refiner.cc:185-230).
// excl_cur implements BALANCE mode (the overload balancer's forcing rule,
// refinement/balancer/overload_balancer.cc in spirit): a vertex whose block
// exceeds its cap loses "stay" as a candidate, so it proposes its best
// admissible target even at negative gain; everything else (commit
// admission, caps, determinism) is the normal refiner machinery.
__device__ inline bool accept_refine(
    u32 c, u32 cur, i32 u_w, i64 cw, i64 maxw, i64 cur_w, i64 cur_maxw,
    bool balance = false
) {
  if (c == cur) {
    return !(balance && cur_w > cur_maxw);
  }
  if (balance) {
    // room-only in balance mode: the relative-overload clause would let the
    // deterministic select livelock on an already-full favourite target
    return cw + u_w <= maxw;
  }
  return (cw + u_w <= maxw) || ((cw - maxw) < (cur_w - cur_maxw));
}

// Underload-balancer acceptance (underload_balancer.cc is_movable_to): only
// underloaded targets with room; the current block is never a candidate.
__device__ inline bool accept_underload(u32 c, u32 cur, i32 u_w, i64 cw, i64 maxw_c, i64 minw_c) {
  return c != cur && cw < minw_c && cw + u_w <= maxw_c;
}

// Weight-acceptance predicate (clusterer variant, lp_clusterer.cc:199-204).
__device__ inline bool accept_cluster(u32 c, u32 cur, i32 u_w, i64 cw, i64 maxw_uniform) {
  return (cw + u_w <= maxw_uniform) || (c == cur);
}

__device__ inline u32 hash_u32(u32 c) {
  return static_cast<u32>(kmp::splitmix64(c));
}

// ------------------------------------------------------------ S path
// Covers EVERY position of the slice: 4 positions per wave, 16 lanes each.
// Owns the slot for: tail positions (u >= n), inactive or degree-filtered
// vertices (invalid slot), and active deg <= 16 vertices (computed result).
// Leaves deg in (16, inf) active slots for the M/L kernels.
template <typename LT>
__global__ void k_phase_s(
    u32 pos_lo,
    u32 pos_hi,
    u32 chunk_base,
    u32 n,
    u64 iter_seed,
    u32 mode, // 0 = refine, 1 = overload balance, 2 = underload balance
    u32 fallback,
    u32 max_degree,
    u32 s_clear, // v2 path: S owns the chunk's active-flag clearing
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const i64 *__restrict__ minw, // per-block minimums (mode 2 only)
    const LT *__restrict__ labels_s, // u8 shadow for k <= 256, else u16
    uint8_t *__restrict__ active,
    const uint8_t *__restrict__ unit_active,
    Prop *__restrict__ slots
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 sub = lane >> 4;  // subgroup 0..3
  const u32 slot = lane & 15; // lane within subgroup
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 p = pos_lo + wave_id * 4 + sub;
  if (p >= pos_hi) {
    return;
  }

  const BlockPerm perm(n, iter_seed);
  // all 4 positions of this wave share one 64-vertex unit; one byte decides
  // whether anything in it is active (inactive units are skipped downstream)
  const u32 vb = perm.fp(p / kmp::kUnit);
  if (!unit_active[vb]) {
    return;
  }
  const u32 u = vb * kmp::kUnit + (p % kmp::kUnit);
  const u32 sidx = p - pos_lo;

  bool skip = false;
  u64 row = 0;
  u32 deg = 0;
  if (u >= n) {
    skip = true;
  } else {
    row = xadj[u];
    deg = static_cast<u32>(xadj[u + 1] - row);
    const bool act = active[u] != 0;
    if (!act || deg > max_degree) {
      skip = true;
    }
    // v2: clear the processed flag inline (replaces k_clear_active's pass;
    // exactly the legacy condition: deg <= max_degree && active)
    if (s_clear && slot == 0 && act && deg <= max_degree) {
      active[u] = 0;
    }
  }

  // M/L work lists are built by k_build_lists (one reservation atomic per
  // wave-row; appending from here serialized millions of waves on a single
  // counter -- measured 2x on k_phase_s at scale 26).
  if (skip) {
    // always-write contract: every position of an ACTIVE unit gets a fresh
    // slot each chunk (proposal or invalid), so the memset-free v2 commit
    // can read raw slots gated only by unit_active
    if (slot == 0) {
      slots[sidx] = Prop{0u, kInvalid, 0u, 0u};
    }
    return;
  }
  if (deg > kSmallDeg) {
    return; // M/L kernels own (and always write) these slots
  }

  // candidate load: lane handles one edge
  u32 c = kInvalid;
  i32 w = 0;
  if (slot < deg) {
    // non-temporal: the adjncy stream is read once per sweep and must not
    // evict the label shadow from L2/MALL (the gather working set)
    const u32 v = __builtin_nontemporal_load(&adjncy[row + slot]);
    c = labels_s[v]; // narrow shadow: u8 for k <= 256 (64 MB at scale 26,
                     // MALL-resident), u16 up to k <= 2048
    w = adjwgt ? adjwgt[row + slot] : 1;
  }

  // dedupe within subgroup: sum weights of equal clusters; lowest slot owns
  i32 gain = w;
  bool owner = (slot < deg);
  const u32 base = sub * 16;
  for (u32 j = 0; j < 16; ++j) {
    const u32 cj = __shfl(c, base + j, kWave);
    const i32 wj = __shfl(w, base + j, kWave);
    if (j != slot && c != kInvalid && cj == c) {
      gain += wj;
      if (j < slot) {
        owner = false;
      }
    }
  }

  const u32 cur = labels[u];
  const i32 u_w = vwgt ? vwgt[u] : 1;
  const i64 cur_w = weights[cur];
  const i64 cur_maxw = maxw[cur];
  const bool excl_cur = mode == 1;
  bool vgate = true;
  if (mode == 2) {
    const i64 mnw = minw[cur];
    vgate = cur_w >= mnw && cur_w - u_w >= mnw; // is_movable_from
  }

  BestState best{0, 0, 0, false};
  if (owner && c != kInvalid && vgate) {
    const i64 cw = weights[c];
    const i64 mw = maxw[c];
    const bool ok = (mode == 2)
                        ? accept_underload(c, cur, u_w, cw, mw, minw[c])
                        : accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw, excl_cur);
    if (ok) {
      best = BestState{gain, tie_hash(iter_seed, u, c), c, true};
    }
  }

  // subgroup argmax over 16 lanes
  for (int off = 8; off > 0; off >>= 1) {
    const i32 og = __shfl_down(best.gain, off, kWave);
    const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
    const u32 oc = __shfl_down(best.c, off, kWave);
    const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
    if ((slot + off) < 16 && ohave && key_better(og, oh, oc, best)) {
      best = BestState{og, oh, oc, true};
    }
  }

  if (slot == 0) {
    BestState fin = best;
    if (!fin.have && mode == 1 && cur_w > cur_maxw && fallback != kInvalid &&
        fallback != cur) {
      // balance fallback: no admissible adjacent target (e.g. everything in
      // one block) -- propose the lightest block with room at gain 0
      fin = BestState{0, tie_hash(iter_seed, u, fallback), fallback, true};
    }
    if (fin.have && fin.c != cur) {
      slots[sidx] = Prop{u, fin.c, p - chunk_base, static_cast<u32>(u_w)};
    } else {
      slots[sidx] = Prop{0u, kInvalid, 0u, 0u};
    }
  }
}

// ------------------------------------------------------------ M path
// One wavefront per position (dead waves for non-M positions retire in a
// few cycles; list building would need a serializing append counter).
// Handles active vertices with kSmallDeg < deg <= kMidDeg: dense per-wave
// LDS gains (k <= kMaxDenseK), ballot-waterfall accumulation. Active
// deg > kMidDeg vertices were appended to the L list by k_phase_s.
// blockDim.x = 256 (4 waves); dynamic LDS = 4 * k * sizeof(i32).
template <bool kUnitWeights, typename LT>
__global__ void k_phase_m(
    u32 pos_lo,
    u32 chunk_base,
    u32 k,
    u64 iter_seed,
    u32 mode,
    u32 fallback,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const LT *__restrict__ labels_s,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const i64 *__restrict__ minw,
    const u64 *__restrict__ m_list,
    const u32 *__restrict__ m_count,
    Prop *__restrict__ slots
) {
  extern __shared__ i32 lds[];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wave_in_wg = threadIdx.x >> 6;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 num_waves = (gridDim.x * blockDim.x) >> 6;
  const u32 R = gain_replicas(k);
  i32 *gains = lds + wave_in_wg * k * R; // R replicas of k counters

  const u32 count = *m_count;
  for (u32 vid = wave_id; vid < count; vid += num_waves) {
  const u64 rec = m_list[vid];
  const u32 p = static_cast<u32>(rec >> 32);
  const u32 u = static_cast<u32>(rec);
  const u64 row = xadj[u];
  const u32 deg = static_cast<u32>(xadj[u + 1] - row);

  for (u32 c = lane; c < k * R; c += kWave) {
    gains[c] = 0;
  }
  __threadfence_block(); // LDS ordering; gains slice is private to this wave

  const u32 rep_off = (lane % R) * k;
  for (u32 e = lane; e < deg; e += kWave) {
    const u32 v = __builtin_nontemporal_load(&adjncy[row + e]);
    const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
    atomicAdd(&gains[rep_off + labels_s[v]], w);
  }
  __threadfence_block();

  const u32 cur = labels[u];
  const i32 u_w = vwgt ? vwgt[u] : 1;
  const i64 cur_w = weights[cur];
  const i64 cur_maxw = maxw[cur];
  const bool excl_cur = mode == 1;
  bool vgate = true;
  if (mode == 2) {
    const i64 mnw = minw[cur];
    vgate = cur_w >= mnw && cur_w - u_w >= mnw;
  }

  BestState best{0, 0, 0, false};
  for (u32 c = lane; vgate && c < k; c += kWave) {
    i32 g = gains[c];
    for (u32 r = 1; r < R; ++r) {
      g += gains[r * k + c];
    }
    if (g <= 0) {
      continue;
    }
    const i64 cw = weights[c];
    const i64 mw = maxw[c];
    const bool ok = (mode == 2)
                        ? accept_underload(c, cur, u_w, cw, mw, minw[c])
                        : accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw, excl_cur);
    if (!ok) {
      continue;
    }
    const u64 h = tie_hash(iter_seed, u, c);
    if (key_better(g, h, c, best)) {
      best = BestState{g, h, c, true};
    }
  }

  for (int off = 32; off > 0; off >>= 1) {
    const i32 og = __shfl_down(best.gain, off, kWave);
    const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
    const u32 oc = __shfl_down(best.c, off, kWave);
    const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
    if (ohave && key_better(og, oh, oc, best)) {
      best = BestState{og, oh, oc, true};
    }
  }

  if (lane == 0) {
    BestState fin = best;
    if (!fin.have && mode == 1 && cur_w > cur_maxw && fallback != kInvalid &&
        fallback != cur) {
      fin = BestState{0, tie_hash(iter_seed, u, fallback), fallback, true};
    }
    if (fin.have && fin.c != cur) {
      slots[p - pos_lo] = Prop{u, fin.c, p - chunk_base, static_cast<u32>(u_w)};
    } else {
      slots[p - pos_lo] = Prop{0u, kInvalid, 0u, 0u};
    }
  }
  __threadfence_block(); // gains reuse across grid-stride iterations
  }
}

// ------------------------------------------------------------ L path
// High-degree vertices are processed slice-parallel: rows are cut into
// kLSlice-edge slices, each handled by one workgroup accumulating into a
// per-vertex global gains row (via a per-WG replicated LDS histogram), then
// a selection kernel reduces each row. This keeps mega-hubs (100K+ edges)
// from serializing on a single workgroup.
constexpr u32 kLSlice = 8192;

// L prep (refine): write per-vertex slice counts in parallel into a
// zeroed l_cap+1 array; the exclusive prefix runs as a rocprim scan on the
// host side (entries beyond the live count are zero, so l_off[count] holds
// the total).
__global__ void k_l_sizes(
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    const u64 *__restrict__ xadj,
    u32 l_cap,
    u32 *__restrict__ sizes
) {
  const u32 count = *l_count < l_cap ? *l_count : l_cap;
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    const u32 u = static_cast<u32>(l_list[i]);
    const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
    sizes[i] = (deg + kLSlice - 1) / kLSlice;
  }
}

// Accumulate one slice per workgroup into the vertex's global gains row.
template <bool kUnitWeights, typename LT>
__global__ void k_phase_l_acc(
    u32 k,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const LT *__restrict__ labels_s,
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    u32 l_cap,
    const u32 *__restrict__ l_off,
    i32 *__restrict__ l_gains // l_cap x k
) {
  extern __shared__ i32 lds[];
  const u32 R = gain_replicas(k);
  i32 *hist = lds;

  const u32 count = *l_count < l_cap ? *l_count : l_cap;
  const u32 total = l_off[count];
  for (u32 s = blockIdx.x; s < total; s += gridDim.x) {
    // binary search the vertex owning slice s
    u32 lo = 0, hi = count - 1;
    while (lo < hi) {
      const u32 mid = (lo + hi + 1) >> 1;
      if (l_off[mid] <= s) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    const u32 vid = lo;
    const u32 u = static_cast<u32>(l_list[vid]);
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    const u32 e_lo = (s - l_off[vid]) * kLSlice;
    const u32 e_hi = e_lo + kLSlice < deg ? e_lo + kLSlice : deg;

    for (u32 c = threadIdx.x; c < k * R; c += blockDim.x) {
      hist[c] = 0;
    }
    __syncthreads();
    const u32 rep_off = (threadIdx.x % R) * k;
    for (u32 e = e_lo + threadIdx.x; e < e_hi; e += blockDim.x) {
      const u32 v = __builtin_nontemporal_load(&adjncy[row + e]);
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      atomicAdd(&hist[rep_off + labels_s[v]], w);
    }
    __syncthreads();
    i32 *grow = l_gains + static_cast<size_t>(vid) * k;
    for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
      i32 g = hist[c];
      for (u32 r = 1; r < R; ++r) {
        g += hist[r * k + c];
      }
      if (g) {
        atomicAdd(&grow[c], g);
      }
    }
    __syncthreads();
  }
}

// Select per L vertex (one workgroup each, grid-stride), write the slot,
// and reset the gains row for the next chunk.
__global__ void k_phase_l_sel(
    u32 mode,
    u32 fallback,
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    u32 k,
    const u64 *__restrict__ xadj,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const i64 *__restrict__ minw,
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    u32 l_cap,
    i32 *__restrict__ l_gains,
    Prop *__restrict__ slots
) {
  __shared__ i64 red[16];
  const u32 count = *l_count < l_cap ? *l_count : l_cap;
  for (u32 vid = blockIdx.x; vid < count; vid += gridDim.x) {
    const u64 rec = l_list[vid];
    const u32 p = static_cast<u32>(rec >> 32);
    const u32 u = static_cast<u32>(rec);
    const u32 cur = labels[u];
    const i32 u_w = vwgt ? vwgt[u] : 1;
    const i64 cur_w = weights[cur];
    const i64 cur_maxw = maxw[cur];
    const bool excl_cur = mode == 1;
    bool vgate = true;
    if (mode == 2) {
      const i64 mnw = minw[cur];
      vgate = cur_w >= mnw && cur_w - u_w >= mnw;
    }
    i32 *grow = l_gains + static_cast<size_t>(vid) * k;

    BestState best{0, 0, 0, false};
    for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
      const i32 g = grow[c];
      grow[c] = 0; // reset for the next chunk
      if (g <= 0 || !vgate) {
        continue;
      }
      const i64 cw = weights[c];
      const i64 mw = maxw[c];
      const bool ok = (mode == 2)
                          ? accept_underload(c, cur, u_w, cw, mw, minw[c])
                          : accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw, excl_cur);
      if (!ok) {
        continue;
      }
      const u64 h = tie_hash(iter_seed, u, c);
      if (key_better(g, h, c, best)) {
        best = BestState{g, h, c, true};
      }
    }
    const u32 lane = threadIdx.x & (kWave - 1);
    for (int off = 32; off > 0; off >>= 1) {
      const i32 og = __shfl_down(best.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
      const u32 oc = __shfl_down(best.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
      if (ohave && key_better(og, oh, oc, best)) {
        best = BestState{og, oh, oc, true};
      }
    }
    const u32 wave_in_wg = threadIdx.x >> 6;
    if (lane == 0) {
      red[wave_in_wg * 2] = (static_cast<i64>(best.gain) << 1) | (best.have ? 1 : 0);
      red[wave_in_wg * 2 + 1] = static_cast<i64>(best.h);
      reinterpret_cast<u32 *>(red + 8)[wave_in_wg] = best.c;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      BestState total{0, 0, 0, false};
      const u32 waves = blockDim.x >> 6;
      for (u32 wv = 0; wv < waves; ++wv) {
        const i64 packed = red[wv * 2];
        if (packed & 1) {
          const i32 g = static_cast<i32>(packed >> 1);
          const u64 h = static_cast<u64>(red[wv * 2 + 1]);
          const u32 c = reinterpret_cast<u32 *>(red + 8)[wv];
          if (key_better(g, h, c, total)) {
            total = BestState{g, h, c, true};
          }
        }
      }
      if (!total.have && mode == 1 && cur_w > cur_maxw &&
          fallback != kInvalid && fallback != cur) {
        total = BestState{0, tie_hash(iter_seed, u, fallback), fallback, true};
      }
      if (total.have && total.c != cur) {
        slots[p - pos_lo] = Prop{u, total.c, p - chunk_base, static_cast<u32>(u_w)};
      } else {
        slots[p - pos_lo] = Prop{0u, kInvalid, 0u, 0u};
      }
    }
    __syncthreads();
  }
}

// Fallback for L entries beyond l_cap (pathological): one workgroup per
// vertex, whole row, replicated LDS histogram.
template <bool kUnitWeights, typename LT>
__global__ void k_phase_l_direct(
    u32 mode,
    u32 fallback,
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    u32 k,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const LT *__restrict__ labels_s,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const i64 *__restrict__ minw,
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    u32 l_cap,
    Prop *__restrict__ slots
) {
  extern __shared__ i32 lds[];
  const u32 R = gain_replicas(k);
  i32 *gains = lds;
  i64 *red = reinterpret_cast<i64 *>(lds + ((k * R + 1) & ~1u));

  const u32 count = *l_count;
  for (u32 vid = l_cap + blockIdx.x; vid < count; vid += gridDim.x) {
    for (u32 c = threadIdx.x; c < k * R; c += blockDim.x) {
      gains[c] = 0;
    }
    __syncthreads();

    const u64 rec = l_list[vid];
    const u32 p = static_cast<u32>(rec >> 32);
    const u32 u = static_cast<u32>(rec);
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);

    const u32 rep_off = (threadIdx.x % R) * k;
    for (u32 e = threadIdx.x; e < deg; e += blockDim.x) {
      const u32 v = adjncy[row + e];
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      atomicAdd(&gains[rep_off + labels_s[v]], w);
    }
    __syncthreads();

    const u32 cur = labels[u];
    const i32 u_w = vwgt ? vwgt[u] : 1;
    const i64 cur_w = weights[cur];
    const i64 cur_maxw = maxw[cur];
    const bool excl_cur = mode == 1;
    bool vgate = true;
    if (mode == 2) {
      const i64 mnw = minw[cur];
      vgate = cur_w >= mnw && cur_w - u_w >= mnw;
    }

    BestState best{0, 0, 0, false};
    for (u32 c = threadIdx.x; vgate && c < k; c += blockDim.x) {
      i32 g = gains[c];
      for (u32 r = 1; r < R; ++r) {
        g += gains[r * k + c];
      }
      if (g <= 0) {
        continue;
      }
      const i64 cw = weights[c];
      const i64 mw = maxw[c];
      const bool ok = (mode == 2)
                          ? accept_underload(c, cur, u_w, cw, mw, minw[c])
                          : accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw, excl_cur);
      if (!ok) {
        continue;
      }
      const u64 h = tie_hash(iter_seed, u, c);
      if (key_better(g, h, c, best)) {
        best = BestState{g, h, c, true};
      }
    }
    const u32 lane = threadIdx.x & (kWave - 1);
    for (int off = 32; off > 0; off >>= 1) {
      const i32 og = __shfl_down(best.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
      const u32 oc = __shfl_down(best.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
      if (ohave && key_better(og, oh, oc, best)) {
        best = BestState{og, oh, oc, true};
      }
    }
    const u32 wave_in_wg = threadIdx.x >> 6;
    if (lane == 0) {
      red[wave_in_wg * 2] = (static_cast<i64>(best.gain) << 1) | (best.have ? 1 : 0);
      red[wave_in_wg * 2 + 1] = static_cast<i64>(best.h);
      reinterpret_cast<u32 *>(red + 8)[wave_in_wg] = best.c;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      BestState total{0, 0, 0, false};
      const u32 waves = blockDim.x >> 6;
      for (u32 wv = 0; wv < waves; ++wv) {
        const i64 packed = red[wv * 2];
        if (packed & 1) {
          const i32 g = static_cast<i32>(packed >> 1);
          const u64 h = static_cast<u64>(red[wv * 2 + 1]);
          const u32 c = reinterpret_cast<u32 *>(red + 8)[wv];
          if (key_better(g, h, c, total)) {
            total = BestState{g, h, c, true};
          }
        }
      }
      if (!total.have && mode == 1 && cur_w > cur_maxw &&
          fallback != kInvalid && fallback != cur) {
        total = BestState{0, tie_hash(iter_seed, u, fallback), fallback, true};
      }
      if (total.have && total.c != cur) {
        slots[p - pos_lo] = Prop{u, total.c, p - chunk_base, static_cast<u32>(u_w)};
      } else {
        slots[p - pos_lo] = Prop{0u, kInvalid, 0u, 0u};
      }
    }
    __syncthreads();
  }
}

// ==================== clustering (clusters = vertices) ====================
// LP clusterer instantiation (lp_clusterer.cc:23-28): ClusterID space = n,
// clusters start as singletons, uniform weight cap, favored-cluster tracking
// for two-hop merging. Gain maps are hash-based (cluster ids are unbounded):
// deg <= 16 register waterfall, deg <= 512 per-wave LDS hash, larger rows
// slice-parallel into a pooled global hash.

constexpr u32 kClusterMidDeg = 512; // per-wave LDS-hash path bound
constexpr u32 kHashSlots = 1024;    // per-wave LDS hash (8 KB per wave)
constexpr u32 kClusterM2Deg = 1536; // per-WG LDS-hash path bound (load <= 0.375)
constexpr u32 kM2HashSlots = 4096;  // per-WG LDS hash (32 KB per WG)

// S path for clustering: identical structure to k_phase_s, clusterer accept
// + favored-cluster argmax over all candidates.
__global__ void k_phase_s_c(
    u32 pos_lo,
    u32 pos_hi,
    u32 chunk_base,
    u32 n,
    u64 iter_seed,
    u32 max_degree,
    i64 maxw_uniform,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const u32 *__restrict__ comm, // null = unrestricted (clusterer.h:35)
    const uint8_t *__restrict__ active,
    const uint8_t *__restrict__ unit_active,
    u32 *__restrict__ favored,
    Prop *__restrict__ slots
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 sub = lane >> 4;
  const u32 slot = lane & 15;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 p = pos_lo + wave_id * 4 + sub;
  if (p >= pos_hi) {
    return;
  }

  const BlockPerm perm(n, iter_seed);
  // all 4 positions of this wave share one 64-vertex unit (slots pre-invalid)
  const u32 vb = perm.fp(p / kmp::kUnit);
  if (!unit_active[vb]) {
    return;
  }
  const u32 u = vb * kmp::kUnit + (p % kmp::kUnit);
  const u32 sidx = p - pos_lo;

  bool skip = false;
  u64 row = 0;
  u32 deg = 0;
  if (u >= n) {
    skip = true;
  } else {
    row = xadj[u];
    deg = static_cast<u32>(xadj[u + 1] - row);
    if (!active[u] || deg > max_degree) {
      skip = true;
    }
  }

  // M/L work lists are built by k_build_lists
  if (skip || deg > kSmallDeg) {
    return; // M/L own larger degrees
  }

  u32 c = kInvalid;
  i32 w = 0;
  if (slot < deg) {
    const u32 v = adjncy[row + slot];
    c = labels[v];
    w = adjwgt ? adjwgt[row + slot] : 1;
  }

  i32 gain = w;
  bool owner = (slot < deg);
  const u32 base = sub * 16;
  for (u32 j = 0; j < 16; ++j) {
    const u32 cj = __shfl(c, base + j, kWave);
    const i32 wj = __shfl(w, base + j, kWave);
    if (j != slot && c != kInvalid && cj == c) {
      gain += wj;
      if (j < slot) {
        owner = false;
      }
    }
  }

  const u32 cur = labels[u];
  const i32 u_w = vwgt ? vwgt[u] : 1;
  const i64 cur_w = weights[cur];

  BestState best{0, 0, 0, false};
  BestState fav{0, 0, 0, false};
  if (owner && c != kInvalid &&
      (comm == nullptr || comm[c] == comm[cur])) { // lp_clusterer.cc:193-194
    const u64 h = tie_hash(iter_seed, u, c);
    fav = BestState{gain, h, c, true};
    if (accept_cluster(c, cur, u_w, weights[c], maxw_uniform)) {
      best = BestState{gain, h, c, true};
    }
  }

  for (int off = 8; off > 0; off >>= 1) {
    {
      const i32 og = __shfl_down(best.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
      const u32 oc = __shfl_down(best.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
      if ((slot + off) < 16 && ohave && key_better(og, oh, oc, best)) {
        best = BestState{og, oh, oc, true};
      }
    }
    {
      const i32 og = __shfl_down(fav.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(fav.h), off, kWave);
      const u32 oc = __shfl_down(fav.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(fav.have), off, kWave);
      if ((slot + off) < 16 && ohave && key_better(og, oh, oc, fav)) {
        fav = BestState{og, oh, oc, true};
      }
    }
  }

  if (slot == 0) {
    // favored cluster for two-hop (label_propagation.h:517-535): stored only
    // while the vertex still sits in a light singleton cluster
    const bool store_favored = (u_w == cur_w) && (cur_w <= maxw_uniform / 2);
    if (store_favored) {
      favored[u] = fav.have ? fav.c : cur;
    }
    if (best.have && best.c != cur) {
      slots[sidx] = Prop{u, best.c, p - chunk_base, static_cast<u32>(u_w)};
    }
  }
}

// M path for clustering: one wave per position, per-wave LDS hash
// (kHashSlots key/value pairs), deg in (kSmallDeg, kClusterMidDeg].
template <bool kUnitWeights>
__global__ void k_phase_m_c(
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    i64 maxw_uniform,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const u32 *__restrict__ comm,
    const u64 *__restrict__ m_list,
    const u32 *__restrict__ m_count,
    u32 *__restrict__ favored,
    Prop *__restrict__ slots
) {
  extern __shared__ u32 ldsu[];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wave_in_wg = threadIdx.x >> 6;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 num_waves = (gridDim.x * blockDim.x) >> 6;
  u32 *hkeys = ldsu + wave_in_wg * 2 * kHashSlots;
  i32 *hvals = reinterpret_cast<i32 *>(hkeys + kHashSlots);

  const u32 count = *m_count;
  for (u32 vid = wave_id; vid < count; vid += num_waves) {
  const u64 rec = m_list[vid];
  const u32 p = static_cast<u32>(rec >> 32);
  const u32 u = static_cast<u32>(rec);
  const u64 row = xadj[u];
  const u32 deg = static_cast<u32>(xadj[u + 1] - row);

  for (u32 s = lane; s < kHashSlots; s += kWave) {
    hkeys[s] = kInvalid;
    hvals[s] = 0;
  }
  __threadfence_block();

  for (u32 e = lane; e < deg; e += kWave) {
    const u32 v = adjncy[row + e];
    const u32 c = labels[v];
    const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
    u32 s = hash_u32(c) & (kHashSlots - 1);
    while (true) {
      const u32 kcur = hkeys[s];
      if (kcur == c) {
        atomicAdd(&hvals[s], w);
        break;
      }
      if (kcur == kInvalid) {
        const u32 old = atomicCAS(&hkeys[s], kInvalid, c);
        if (old == kInvalid || old == c) {
          atomicAdd(&hvals[s], w);
          break;
        }
      }
      s = (s + 1) & (kHashSlots - 1);
    }
  }
  __threadfence_block();

  const u32 cur = labels[u];
  const i32 u_w = vwgt ? vwgt[u] : 1;
  const i64 cur_w = weights[cur];

  BestState best{0, 0, 0, false};
  BestState fav{0, 0, 0, false};
  for (u32 s = lane; s < kHashSlots; s += kWave) {
    const u32 c = hkeys[s];
    if (c == kInvalid) {
      continue;
    }
    const i32 g = hvals[s];
    if (g <= 0 || (comm != nullptr && comm[c] != comm[cur])) {
      continue;
    }
    const u64 h = tie_hash(iter_seed, u, c);
    if (key_better(g, h, c, fav)) {
      fav = BestState{g, h, c, true};
    }
    if (accept_cluster(c, cur, u_w, weights[c], maxw_uniform) && key_better(g, h, c, best)) {
      best = BestState{g, h, c, true};
    }
  }

  for (int off = 32; off > 0; off >>= 1) {
    {
      const i32 og = __shfl_down(best.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
      const u32 oc = __shfl_down(best.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
      if (ohave && key_better(og, oh, oc, best)) {
        best = BestState{og, oh, oc, true};
      }
    }
    {
      const i32 og = __shfl_down(fav.gain, off, kWave);
      const u64 oh = __shfl_down(static_cast<unsigned long long>(fav.h), off, kWave);
      const u32 oc = __shfl_down(fav.c, off, kWave);
      const int ohave = __shfl_down(static_cast<int>(fav.have), off, kWave);
      if (ohave && key_better(og, oh, oc, fav)) {
        fav = BestState{og, oh, oc, true};
      }
    }
  }

  if (lane == 0) {
    const bool store_favored = (u_w == cur_w) && (cur_w <= maxw_uniform / 2);
    if (store_favored) {
      favored[u] = fav.have ? fav.c : cur;
    }
    if (best.have && best.c != cur) {
      slots[p - pos_lo] = Prop{u, best.c, p - chunk_base, static_cast<u32>(u_w)};
    }
  }
  __threadfence_block(); // hash reuse across grid-stride iterations
  }
}

// M2 path for clustering: one 256-thread workgroup per vertex with a
// 4096-slot LDS hash -- absorbs the dense coarse-level rows
// (kClusterMidDeg < deg <= kClusterM2Deg) that otherwise take the pooled
// GLOBAL hash path (measured dominant in the multilevel pipeline).
template <bool kUnitWeights>
__global__ void k_phase_m2_c(
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    i64 maxw_uniform,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const u32 *__restrict__ comm,
    const u64 *__restrict__ m2_list,
    const u32 *__restrict__ m2_count,
    u32 *__restrict__ favored,
    Prop *__restrict__ slots
) {
  __shared__ u32 hkeys[kM2HashSlots];
  __shared__ i32 hvals[kM2HashSlots];
  __shared__ i64 red[24];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 count = *m2_count;
  for (u32 vid = blockIdx.x; vid < count; vid += gridDim.x) {
    const u64 rec = m2_list[vid];
    const u32 p = static_cast<u32>(rec >> 32);
    const u32 u = static_cast<u32>(rec);
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);

    for (u32 s = threadIdx.x; s < kM2HashSlots; s += blockDim.x) {
      hkeys[s] = kInvalid;
      hvals[s] = 0;
    }
    __syncthreads();
    for (u32 e = threadIdx.x; e < deg; e += blockDim.x) {
      const u32 v = adjncy[row + e];
      const u32 c = labels[v];
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      u32 s = hash_u32(c) & (kM2HashSlots - 1);
      while (true) {
        const u32 kcur = hkeys[s];
        if (kcur == c) {
          atomicAdd(&hvals[s], w);
          break;
        }
        if (kcur == kInvalid) {
          const u32 old = atomicCAS(&hkeys[s], kInvalid, c);
          if (old == kInvalid || old == c) {
            atomicAdd(&hvals[s], w);
            break;
          }
        }
        s = (s + 1) & (kM2HashSlots - 1);
      }
    }
    __syncthreads();

    const u32 cur = labels[u];
    const i32 u_w = vwgt ? vwgt[u] : 1;
    const i64 cur_w = weights[cur];

    BestState best{0, 0, 0, false};
    BestState fav{0, 0, 0, false};
    for (u32 s = threadIdx.x; s < kM2HashSlots; s += blockDim.x) {
      const u32 c = hkeys[s];
      if (c == kInvalid) {
        continue;
      }
      const i32 g = hvals[s];
      if (g <= 0 || (comm != nullptr && comm[c] != comm[cur])) {
        continue;
      }
      const u64 h = tie_hash(iter_seed, u, c);
      if (key_better(g, h, c, fav)) {
        fav = BestState{g, h, c, true};
      }
      if (accept_cluster(c, cur, u_w, weights[c], maxw_uniform) && key_better(g, h, c, best)) {
        best = BestState{g, h, c, true};
      }
    }
    for (int off = 32; off > 0; off >>= 1) {
      {
        const i32 og = __shfl_down(best.gain, off, kWave);
        const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
        const u32 oc = __shfl_down(best.c, off, kWave);
        const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
        if (ohave && key_better(og, oh, oc, best)) {
          best = BestState{og, oh, oc, true};
        }
      }
      {
        const i32 og = __shfl_down(fav.gain, off, kWave);
        const u64 oh = __shfl_down(static_cast<unsigned long long>(fav.h), off, kWave);
        const u32 oc = __shfl_down(fav.c, off, kWave);
        const int ohave = __shfl_down(static_cast<int>(fav.have), off, kWave);
        if (ohave && key_better(og, oh, oc, fav)) {
          fav = BestState{og, oh, oc, true};
        }
      }
    }
    const u32 wave_in_wg = threadIdx.x >> 6;
    if (lane == 0) {
      red[wave_in_wg * 3] = (static_cast<i64>(best.gain) << 2) | (best.have ? 1 : 0) |
                            (fav.have ? 2 : 0);
      red[wave_in_wg * 3 + 1] = static_cast<i64>(best.h);
      red[wave_in_wg * 3 + 2] = static_cast<i64>(fav.h);
      reinterpret_cast<u32 *>(red + 12)[wave_in_wg * 2] = best.c;
      reinterpret_cast<u32 *>(red + 12)[wave_in_wg * 2 + 1] = fav.c;
      reinterpret_cast<i32 *>(red + 16)[wave_in_wg] = fav.gain;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      BestState tb{0, 0, 0, false};
      BestState tf{0, 0, 0, false};
      const u32 waves = blockDim.x >> 6;
      for (u32 wv = 0; wv < waves; ++wv) {
        const i64 packed = red[wv * 3];
        const i32 bg = static_cast<i32>(packed >> 2);
        if (packed & 1) {
          const u64 h = static_cast<u64>(red[wv * 3 + 1]);
          const u32 c = reinterpret_cast<u32 *>(red + 12)[wv * 2];
          if (key_better(bg, h, c, tb)) {
            tb = BestState{bg, h, c, true};
          }
        }
        if (packed & 2) {
          const i32 fg = reinterpret_cast<i32 *>(red + 16)[wv];
          const u64 h = static_cast<u64>(red[wv * 3 + 2]);
          const u32 c = reinterpret_cast<u32 *>(red + 12)[wv * 2 + 1];
          if (key_better(fg, h, c, tf)) {
            tf = BestState{fg, h, c, true};
          }
        }
      }
      const bool store_favored = (u_w == cur_w) && (cur_w <= maxw_uniform / 2);
      if (store_favored) {
        favored[u] = tf.have ? tf.c : cur;
      }
      if (tb.have && tb.c != cur) {
        slots[p - pos_lo] = Prop{u, tb.c, p - chunk_base, static_cast<u32>(u_w)};
      }
    }
    __syncthreads();
  }
}

// L prep for clustering: per-vertex slice prefix AND pooled-hash region
// prefix (region size = next power of two >= 2*deg). Writes the total
// region demand to *hacc_out: when it exceeds the pool, the host processes
// the L list in pool-sized batches (k_phase_l_sel_c clears each region
// after reading it, so the pool can be reused within a chunk).
__global__ void k_l_prep_c(
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    const u64 *__restrict__ xadj,
    u32 l_cap,
    u32 *__restrict__ l_off,  // slice prefix
    u64 *__restrict__ l_hoff, // hash-region prefix
    u32 *__restrict__ l_hbits, // log2(region size)
    u32 *__restrict__ l_ccnt, // per-vertex claimed-slot counts (zeroed here)
    unsigned long long *__restrict__ hacc_out
) {
  __shared__ u64 red[17];
  const u32 count = *l_count < l_cap ? *l_count : l_cap;
  const u32 tid = threadIdx.x;
  const u32 lane = tid & (kWave - 1);
  u32 scarry = 0;
  u64 hcarry = 0;
  for (u32 base = 0; base < count; base += blockDim.x) {
    const u32 i = base + tid;
    u32 sc = 0;
    u64 hs = 0;
    if (i < count) {
      const u32 u = static_cast<u32>(l_list[i]);
      const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
      sc = (deg + kLSlice - 1) / kLSlice;
      u32 bits = 11; // >= 2048 slots
      while ((1u << bits) < 2 * deg) {
        ++bits;
      }
      l_hbits[i] = bits;
      l_ccnt[i] = 0;
      hs = 1ull << bits;
    }
    // joint block exclusive scan: region sizes in the high 32 bits, slice
    // counts in the low 32 (tile sums stay well under 2^32 each)
    u64 inc = (hs << 32) | sc;
    for (int off = 1; off < 64; off <<= 1) {
      const u64 o = __shfl_up(static_cast<unsigned long long>(inc), off, kWave);
      if (lane >= static_cast<u32>(off)) {
        inc += o;
      }
    }
    __syncthreads();
    if (lane == 63) {
      red[tid >> 6] = inc;
    }
    __syncthreads();
    u64 wbase = 0;
    for (u32 w = 0; w < (tid >> 6); ++w) {
      wbase += red[w];
    }
    const u64 ex = wbase + inc - ((hs << 32) | sc);
    if (i < count) {
      l_off[i] = scarry + static_cast<u32>(ex & 0xFFFFFFFFu);
      l_hoff[i] = hcarry + (ex >> 32);
    }
    u64 tsum = 0;
    for (u32 w = 0; w < blockDim.x / kWave; ++w) {
      tsum += red[w];
    }
    scarry += static_cast<u32>(tsum & 0xFFFFFFFFu);
    hcarry += tsum >> 32;
    __syncthreads();
  }
  if (tid == 0) {
    l_off[count] = scarry;
    l_hoff[count] = hcarry;
    *hacc_out = hcarry;
  }
}

// Slice-parallel accumulation into the pooled per-vertex global hash.
// Processes L vertices [vid_lo, vid_hi) with pool offsets rebased to
// l_hoff[vid_lo] (batched when a chunk's total region demand exceeds the
// pool; k_phase_l_sel_c clears each region so batches can reuse the pool).
template <bool kUnitWeights>
__global__ void k_phase_l_acc_c(
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const u64 *__restrict__ l_list,
    u32 vid_lo,
    u32 vid_hi,
    const u32 *__restrict__ l_off,
    const u64 *__restrict__ l_hoff,
    const u32 *__restrict__ l_hbits,
    u32 *__restrict__ pool_keys,
    i32 *__restrict__ pool_vals,
    u32 *__restrict__ l_clist, // claimed-slot lists (region/2 per vertex)
    u32 *__restrict__ l_ccnt   // per-vertex claimed counts
) {
  const u32 count = vid_hi;
  if (count <= vid_lo) {
    return;
  }
  const u64 hbase = l_hoff[vid_lo];
  const u32 s_lo = l_off[vid_lo];
  const u32 total = l_off[count];
  for (u32 s = s_lo + blockIdx.x; s < total; s += gridDim.x) {
    u32 lo = vid_lo, hi = count - 1;
    while (lo < hi) {
      const u32 mid = (lo + hi + 1) >> 1;
      if (l_off[mid] <= s) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    const u32 vid = lo;
    const u32 u = static_cast<u32>(l_list[vid]);
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    const u32 e_lo = (s - l_off[vid]) * kLSlice;
    const u32 e_hi = e_lo + kLSlice < deg ? e_lo + kLSlice : deg;
    const u64 roff = l_hoff[vid] - hbase;
    u32 *hk = pool_keys + roff;
    i32 *hv = pool_vals + roff;
    u32 *cl = l_clist + (roff >> 1); // region sizes are even powers of two
    const u32 mask = (1u << l_hbits[vid]) - 1;

    for (u32 e = e_lo + threadIdx.x; e < e_hi; e += blockDim.x) {
      const u32 v = adjncy[row + e];
      const u32 c = labels[v];
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      u32 slot = hash_u32(c) & mask;
      while (true) {
        const u32 kcur = hk[slot];
        if (kcur == c) {
          atomicAdd(&hv[slot], w);
          break;
        }
        if (kcur == kInvalid) {
          const u32 old = atomicCAS(&hk[slot], kInvalid, c);
          if (old == kInvalid || old == c) {
            if (old == kInvalid) {
              // record the claimed slot so selection/clearing is
              // O(distinct) instead of O(region)
              cl[atomicAdd(&l_ccnt[vid], 1u)] = slot;
            }
            atomicAdd(&hv[slot], w);
            break;
          }
        }
        slot = (slot + 1) & mask;
      }
    }
  }
}

// Selection per clustering L vertex: scan its hash region, clear it for the
// next chunk, write slot + favored.
__global__ void k_phase_l_sel_c(
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    u32 n,
    i64 maxw_uniform,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const u64 *__restrict__ l_list,
    u32 vid_lo,
    u32 vid_hi,
    const u64 *__restrict__ l_hoff,
    const u32 *__restrict__ l_hbits,
    u32 *__restrict__ pool_keys,
    i32 *__restrict__ pool_vals,
    const u32 *__restrict__ l_clist,
    const u32 *__restrict__ l_ccnt,
    const u32 *__restrict__ comm,
    u32 *__restrict__ favored,
    Prop *__restrict__ slots
) {
  __shared__ i64 red[24];
  const u64 hbase = vid_hi > vid_lo ? l_hoff[vid_lo] : 0;
  for (u32 vid = vid_lo + blockIdx.x; vid < vid_hi; vid += gridDim.x) {
    const u64 rec = l_list[vid];
    const u32 p = static_cast<u32>(rec >> 32);
    const u32 u = static_cast<u32>(rec);
    const u32 cur = labels[u];
    const i32 u_w = vwgt ? vwgt[u] : 1;
    const i64 cur_w = weights[cur];
    const u64 roff = l_hoff[vid] - hbase;
    u32 *hk = pool_keys + roff;
    i32 *hv = pool_vals + roff;
    const u32 *cl = l_clist + (roff >> 1);
    const u32 claimed = l_ccnt[vid];

    BestState best{0, 0, 0, false};
    BestState fav{0, 0, 0, false};
    for (u32 j = threadIdx.x; j < claimed; j += blockDim.x) {
      const u32 s = cl[j];
      const u32 c = hk[s];
      const i32 g = hv[s];
      hk[s] = kInvalid; // clear for the next chunk
      hv[s] = 0;
      if (g <= 0 || (comm != nullptr && comm[c] != comm[cur])) {
        continue;
      }
      const u64 h = tie_hash(iter_seed, u, c);
      if (key_better(g, h, c, fav)) {
        fav = BestState{g, h, c, true};
      }
      if (accept_cluster(c, cur, u_w, weights[c], maxw_uniform) && key_better(g, h, c, best)) {
        best = BestState{g, h, c, true};
      }
    }
    const u32 lane = threadIdx.x & (kWave - 1);
    for (int off = 32; off > 0; off >>= 1) {
      {
        const i32 og = __shfl_down(best.gain, off, kWave);
        const u64 oh = __shfl_down(static_cast<unsigned long long>(best.h), off, kWave);
        const u32 oc = __shfl_down(best.c, off, kWave);
        const int ohave = __shfl_down(static_cast<int>(best.have), off, kWave);
        if (ohave && key_better(og, oh, oc, best)) {
          best = BestState{og, oh, oc, true};
        }
      }
      {
        const i32 og = __shfl_down(fav.gain, off, kWave);
        const u64 oh = __shfl_down(static_cast<unsigned long long>(fav.h), off, kWave);
        const u32 oc = __shfl_down(fav.c, off, kWave);
        const int ohave = __shfl_down(static_cast<int>(fav.have), off, kWave);
        if (ohave && key_better(og, oh, oc, fav)) {
          fav = BestState{og, oh, oc, true};
        }
      }
    }
    const u32 wave_in_wg = threadIdx.x >> 6;
    if (lane == 0) {
      red[wave_in_wg * 3] = (static_cast<i64>(best.gain) << 2) | (best.have ? 1 : 0) |
                            (fav.have ? 2 : 0);
      red[wave_in_wg * 3 + 1] = static_cast<i64>(best.h);
      red[wave_in_wg * 3 + 2] = static_cast<i64>(fav.h);
      reinterpret_cast<u32 *>(red + 12)[wave_in_wg * 2] = best.c;
      reinterpret_cast<u32 *>(red + 12)[wave_in_wg * 2 + 1] = fav.c;
      reinterpret_cast<i32 *>(red + 16)[wave_in_wg] = fav.gain;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      BestState tb{0, 0, 0, false};
      BestState tf{0, 0, 0, false};
      const u32 waves = blockDim.x >> 6;
      for (u32 wv = 0; wv < waves; ++wv) {
        const i64 packed = red[wv * 3];
        const i32 bg = static_cast<i32>(packed >> 2);
        if (packed & 1) {
          const u64 h = static_cast<u64>(red[wv * 3 + 1]);
          const u32 c = reinterpret_cast<u32 *>(red + 12)[wv * 2];
          if (key_better(bg, h, c, tb)) {
            tb = BestState{bg, h, c, true};
          }
        }
        if (packed & 2) {
          const i32 fg = reinterpret_cast<i32 *>(red + 16)[wv];
          const u64 h = static_cast<u64>(red[wv * 3 + 2]);
          const u32 c = reinterpret_cast<u32 *>(red + 12)[wv * 2 + 1];
          if (key_better(fg, h, c, tf)) {
            tf = BestState{fg, h, c, true};
          }
        }
      }
      const bool store_favored = (u_w == cur_w) && (cur_w <= maxw_uniform / 2);
      if (store_favored) {
        favored[u] = tf.have ? tf.c : cur;
      }
      if (tb.have && tb.c != cur) {
        slots[p - pos_lo] = Prop{u, tb.c, p - chunk_base, static_cast<u32>(u_w)};
      }
    }
    __syncthreads();
  }
}

// -------- big-k commit variants (cluster space = n) --------

// Reset dep entries touched by this chunk's proposals (idempotent).
__global__ void k_dep_reset_touched(
    const Prop *__restrict__ props,
    u32 count,
    const u32 *__restrict__ labels,
    unsigned long long *__restrict__ dep
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    dep[labels[props[i].u]] = 0;
  }
}

// Departures via direct global atomics (sources spread over many clusters).
__global__ void k_dep_direct(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ labels,
    unsigned long long *__restrict__ dep
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 to = sto[i];
  if (i - seg_begin[to] < prefix_len[to]) {
    const Prop pr = props[order[i]];
    atomicAdd(&dep[labels[pr.u]], static_cast<unsigned long long>(pr.w));
  }
}

// Head-driven cutoff: one thread per SEGMENT HEAD in the sorted proposals
// (avoids an O(num_clusters) sweep per fixpoint round).
__global__ void k_cutoff_heads(
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ seg_end,
    u32 *__restrict__ prefix_len,
    const i64 *__restrict__ pw,
    const i64 *__restrict__ weights,
    i64 maxw_uniform,
    const unsigned long long *__restrict__ dep,
    int *__restrict__ changed
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 c = sto[i];
  const u32 b = seg_begin[c];
  if (i != b) {
    return; // not the segment head
  }
  const i64 capacity = maxw_uniform - weights[c] + static_cast<i64>(dep[c]);
  const u32 old_len = prefix_len[c];
  u32 lo = 0, hi = old_len;
  while (lo < hi) {
    const u32 mid = (lo + hi + 1) >> 1;
    if (pw[b + mid - 1] <= capacity) {
      lo = mid;
    } else {
      hi = mid - 1;
    }
  }
  if (lo < old_len) {
    prefix_len[c] = lo;
    atomicExch(changed, 1);
  }
}

__global__ void k_seg_len_heads(
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ seg_end,
    u32 *__restrict__ prefix_len
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 c = sto[i];
  if (i == seg_begin[c]) {
    prefix_len[c] = seg_end[c] - seg_begin[c];
  }
}

// Weight updates for big-k: arrivals head-driven, departures via a
// once-per-cluster atomic exchange of the final dep value.
__global__ void k_weights_update_big(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ prefix_len,
    const i64 *__restrict__ pw,
    const u32 *__restrict__ labels,
    unsigned long long *__restrict__ dep,
    i64 *__restrict__ weights
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 to = sto[i];
  const u32 b = seg_begin[to];
  if (i == b && prefix_len[to] > 0) {
    atomicAdd(reinterpret_cast<unsigned long long *>(&weights[to]),
              static_cast<unsigned long long>(pw[b + prefix_len[to] - 1]));
  }
  // departures: first toucher per source cluster applies the whole dep
  const u32 from = labels[props[order[i]].u];
  const unsigned long long d = atomicExch(&dep[from], 0ull);
  if (d) {
    atomicAdd(reinterpret_cast<unsigned long long *>(&weights[from]),
              static_cast<unsigned long long>(-static_cast<i64>(d)));
  }
}

// Count clusters emptied by this chunk (distinct sources with final weight
// zero); eflag entries are reset by k_emptied_reset.
__global__ void k_emptied_count(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    uint8_t *__restrict__ eflag,
    unsigned long long *__restrict__ emptied
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 to = sto[i];
  if (i - seg_begin[to] >= prefix_len[to]) {
    return;
  }
  const u32 from = labels[props[order[i]].u];
  if (weights[from] == 0) {
    // no 8-bit atomicExch: use 32-bit CAS on the aligned word
    u32 *word = reinterpret_cast<u32 *>(reinterpret_cast<uintptr_t>(&eflag[from]) & ~3ull);
    const u32 shift = (from & 3u) * 8;
    const u32 bit = 1u << shift;
    const u32 old = atomicOr(word, bit);
    if ((old & bit) == 0) {
      atomicAdd(emptied, 1ull);
    }
  }
}

__global__ void k_emptied_reset(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    u32 count,
    const u32 *__restrict__ labels,
    uint8_t *__restrict__ eflag
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    eflag[labels[props[i].u]] = 0;
  }
}

// -------- post passes (clusterer) --------

__global__ void k_iota(u32 n, u32 *__restrict__ a) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    a[i] = i;
  }
}

__global__ void k_init_cluster_weights(
    u32 n, const i32 *__restrict__ vwgt, i64 *__restrict__ weights
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    weights[i] = vwgt ? vwgt[i] : 1;
  }
}

// Two-hop candidate collection (label_propagation.h:939-975 threadwise
// considered-set): deg > 0, still in its own singleton cluster at its
// initial weight, light. Slot-write + stable select keeps ascending-u order.
__global__ void k_twohop_cand(
    u32 n,
    i64 maxw_uniform,
    const u64 *__restrict__ xadj,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const u32 *__restrict__ favored,
    u64 *__restrict__ cand_slots // (favored << 32) | u, or ~0 invalid
) {
  const u32 u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= n) {
    return;
  }
  u64 out = ~0ull;
  const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
  if (deg > 0 && labels[u] == u) {
    const i64 w = weights[u];
    const i64 vw = vwgt ? vwgt[u] : 1;
    if (w == vw && w <= maxw_uniform / 2) {
      out = (static_cast<u64>(favored[u]) << 32) | u;
    }
  }
  cand_slots[u] = out;
}

struct CandValid {
  __host__ __device__ bool operator()(const u64 &v) const { return v != ~0ull; }
};

// Pair consecutive candidates with equal favored cluster: even rank joins
// the preceding candidate's (singleton) cluster. rank = 1-based rank within
// the favored segment (scan_by_key output). Always fits: both weights
// <= max/2 (label_propagation.h:977-1002 match semantics).
__global__ void k_twohop_pair(
    const u64 *__restrict__ cand_sorted,
    const u32 *__restrict__ rank, // 1-based within favored segment
    u32 count,
    u32 *__restrict__ labels,
    i64 *__restrict__ weights,
    unsigned long long *__restrict__ merged
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  u64 local = 0;
  if (i < count && (rank[i] & 1u) == 0) {
    const u32 u = static_cast<u32>(cand_sorted[i]);
    const u32 rep = static_cast<u32>(cand_sorted[i - 1]);
    labels[u] = rep;
    // no atomics needed: each candidate appears in exactly one pair
    weights[rep] += weights[u];
    weights[u] = 0;
    local = 1;
  }
  for (int off = 32; off > 0; off >>= 1) {
    local += __shfl_down(static_cast<unsigned long long>(local), off, kWave);
  }
  if ((threadIdx.x & (kWave - 1)) == 0 && local) {
    atomicAdd(merged, static_cast<unsigned long long>(local));
  }
}

__global__ void k_extract_fav(const u64 *__restrict__ cand, u32 count, u32 *__restrict__ fav) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    fav[i] = static_cast<u32>(cand[i] >> 32);
  }
}

__global__ void k_ones(u32 count, u32 *__restrict__ a) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    a[i] = 1;
  }
}

// Apply host-computed isolated-node matches: pairs (u -> rep).
__global__ void k_apply_pairs(
    const u64 *__restrict__ pairs, // (rep << 32) | u
    u32 count,
    u32 *__restrict__ labels,
    i64 *__restrict__ weights
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    const u32 u = static_cast<u32>(pairs[i]);
    const u32 rep = static_cast<u32>(pairs[i] >> 32);
    labels[u] = rep;
    weights[rep] += weights[u];
    weights[u] = 0;
  }
}

// Count non-empty clusters.
__global__ void k_count_nonempty(
    u32 n, const i64 *__restrict__ weights, unsigned long long *__restrict__ out
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  u64 local = (i < n && weights[i] != 0) ? 1 : 0;
  for (int off = 32; off > 0; off >>= 1) {
    local += __shfl_down(static_cast<unsigned long long>(local), off, kWave);
  }
  if ((threadIdx.x & (kWave - 1)) == 0 && local) {
    atomicAdd(out, static_cast<unsigned long long>(local));
  }
}

// -------------------------------------------------------------- commit
__global__ void k_make_keys(
    const Prop *__restrict__ props, u32 count, u32 *__restrict__ keys, u32 *__restrict__ vals
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    keys[i] = props[i].to;
    vals[i] = i;
  }
}

__global__ void k_extract_w(
    const u32 *__restrict__ order, const Prop *__restrict__ props, u32 count, i64 *__restrict__ sw
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < count) {
    sw[i] = static_cast<i64>(static_cast<i32>(props[order[i]].w));
  }
}

__global__ void k_seg_bounds(
    const u32 *__restrict__ sto, u32 count, u32 *__restrict__ seg_begin, u32 *__restrict__ seg_end
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 to = sto[i];
  if (i == 0 || sto[i - 1] != to) {
    seg_begin[to] = i;
  }
  if (i == count - 1 || sto[i + 1] != to) {
    seg_end[to] = i + 1;
  }
}

// Per-WG LDS histogram of departures (one global atomic per cluster per WG;
// a per-proposal global atomic on k addresses serializes badly). Dynamic
// LDS: k x u64. Grid-stride so the WG count stays bounded.
__global__ void k_dep(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    const u32 *__restrict__ sto,
    u32 count,
    u32 k,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ labels,
    unsigned long long *__restrict__ dep
) {
  extern __shared__ unsigned long long hist[];
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    hist[c] = 0;
  }
  __syncthreads();
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < count; i += stride) {
    const u32 to = sto[i];
    if (i - seg_begin[to] < prefix_len[to]) {
      const Prop pr = props[order[i]];
      atomicAdd(&hist[labels[pr.u]], static_cast<unsigned long long>(pr.w));
    }
  }
  __syncthreads();
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    if (hist[c]) {
      atomicAdd(&dep[c], hist[c]);
    }
  }
}

__global__ void k_cutoff(
    u32 k_or_n,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ seg_end,
    u32 *__restrict__ prefix_len,
    const i64 *__restrict__ pw, // within-segment inclusive prefix weights
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const unsigned long long *__restrict__ dep,
    int *__restrict__ changed
) {
  const u32 c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= k_or_n) {
    return;
  }
  const u32 b = seg_begin[c], e = seg_end[c];
  if (e <= b) {
    return;
  }
  const i64 capacity = maxw[c] - weights[c] + static_cast<i64>(dep[c]);
  const u32 old_len = prefix_len[c];
  u32 lo = 0, hi = old_len;
  while (lo < hi) {
    const u32 mid = (lo + hi + 1) >> 1;
    if (pw[b + mid - 1] <= capacity) {
      lo = mid;
    } else {
      hi = mid - 1;
    }
  }
  if (lo < old_len) {
    prefix_len[c] = lo;
    atomicExch(changed, 1);
  }
}

__global__ void k_weights_update(
    u32 k_or_n,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ seg_end,
    const u32 *__restrict__ prefix_len,
    const i64 *__restrict__ pw,
    const unsigned long long *__restrict__ dep,
    i64 *__restrict__ weights
) {
  const u32 c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= k_or_n) {
    return;
  }
  i64 arr = 0;
  const u32 b = seg_begin[c], e = seg_end[c];
  if (e > b && prefix_len[c] > 0) {
    arr = pw[b + prefix_len[c] - 1];
  }
  const i64 delta = arr - static_cast<i64>(dep[c]);
  if (delta != 0) {
    weights[c] += delta;
  }
}

__global__ void k_apply(
    const u32 *__restrict__ order,
    const Prop *__restrict__ props,
    const u32 *__restrict__ sto,
    u32 count,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ prefix_len,
    u32 *__restrict__ labels,
    uint16_t *__restrict__ labels16, // refine shadow; null for clustering
    uint8_t *__restrict__ labels8,   // u8 shadow; null unless k <= 256
    u32 *__restrict__ admitted_flags, // per sorted index
    unsigned long long *__restrict__ moves
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  u64 local = 0;
  if (i < count) {
    const u32 to = sto[i];
    const bool admitted = (i - seg_begin[to]) < prefix_len[to];
    admitted_flags[i] = admitted ? 1u : 0u;
    if (admitted) {
      const u32 u = props[order[i]].u;
      labels[u] = to;
      if (labels16 != nullptr) {
        labels16[u] = static_cast<uint16_t>(to);
      }
      if (labels8 != nullptr) {
        labels8[u] = static_cast<uint8_t>(to);
      }
      local = 1;
    }
  }
  // wave-aggregated move count
  for (int off = 32; off > 0; off >>= 1) {
    local += __shfl_down(static_cast<unsigned long long>(local), off, kWave);
  }
  if ((threadIdx.x & (kWave - 1)) == 0 && local) {
    atomicAdd(moves, static_cast<unsigned long long>(local));
  }
}

// Clear active flags for processed vertices of the WHOLE chunk (identical on
// every rank) and count scanned arcs (WG-aggregated).
__global__ void k_clear_active(
    u32 chunk_lo,
    u32 chunk_hi,
    u32 n,
    u64 iter_seed,
    u32 max_degree,
    const u64 *__restrict__ xadj,
    uint8_t *__restrict__ active,
    uint8_t *__restrict__ unit_active,
    unsigned long long *__restrict__ arcs
) {
  __shared__ unsigned long long wg_sum[4];
  const u32 tid = blockIdx.x * blockDim.x + threadIdx.x;
  const u32 p = chunk_lo + tid;
  const BlockPerm perm(n, iter_seed);
  const u32 lane0 = threadIdx.x & (kWave - 1);

  // one wave covers exactly one 64-vertex unit (chunks are 64-aligned):
  // skip fully-inactive units, clear the unit bit after processing (the
  // activation kernel re-sets it for units that received movers)
  u32 vb = 0;
  bool unit_on = false;
  if (p < chunk_hi) {
    vb = perm.fp(p / kmp::kUnit);
    unit_on = unit_active[vb] != 0;
  }

  u64 my_deg = 0;
  if (p < chunk_hi && unit_on) {
    const u32 u = vb * kmp::kUnit + (p % kmp::kUnit);
    if (u < n) {
      const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
      if (deg <= max_degree && active[u]) {
        my_deg = deg;
        active[u] = 0;
      }
    }
    if (lane0 == 0) {
      unit_active[vb] = 0;
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    my_deg += __shfl_down(static_cast<unsigned long long>(my_deg), off, kWave);
  }
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wave_in_wg = threadIdx.x >> 6;
  if (lane == 0) {
    wg_sum[wave_in_wg] = my_deg;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned long long total = wg_sum[0] + wg_sum[1] + wg_sum[2] + wg_sum[3];
    if (total) {
      atomicAdd(arcs, total);
    }
  }
}

__global__ void k_activate(
    const u32 *__restrict__ order,
    const u32 *__restrict__ admitted_flags,
    const Prop *__restrict__ props,
    u32 count,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    uint8_t *__restrict__ active,
    uint8_t *__restrict__ unit_active
) {
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 lane = threadIdx.x & (kWave - 1);
  if (wave_id >= count || !admitted_flags[wave_id]) {
    return;
  }
  const u32 u = props[order ? order[wave_id] : wave_id].u;
  const u64 row = xadj[u];
  const u32 deg = static_cast<u32>(xadj[u + 1] - row);
  for (u32 e = lane; e < deg; e += kWave) {
    const u32 v = adjncy[row + e];
    active[v] = 1;
    unit_active[v >> 6] = 1;
  }
}

// Reset the touched per-cluster segment entries (reads the UNSORTED props'
// target fields directly; zeroing per cluster is idempotent).
__global__ void k_reset_segs(
    const Prop *__restrict__ props,
    u32 count,
    u32 *__restrict__ seg_begin,
    u32 *__restrict__ seg_end,
    u32 *__restrict__ prefix_len
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= count) {
    return;
  }
  const u32 to = props[i].to;
  seg_begin[to] = 0;
  seg_end[to] = 0;
  prefix_len[to] = 0;
}

__global__ void k_dep_reset_all(u32 k_or_n, unsigned long long *__restrict__ dep) {
  const u32 c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c < k_or_n) {
    dep[c] = 0;
  }
}

// Grid-stride with a per-workgroup LDS histogram (one global atomic per
// cluster per WG instead of one per vertex -- same-address global atomics
// serialize at ~11ns).
__global__ void k_sync_labels16(
    u32 n, const u32 *__restrict__ labels, uint16_t *__restrict__ labels16,
    uint8_t *__restrict__ labels8
) {
  const u32 u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u < n) {
    const u32 l = labels[u];
    labels16[u] = static_cast<uint16_t>(l);
    labels8[u] = static_cast<uint8_t>(l); // meaningful only for k <= 256
  }
}

__global__ void k_init_weights(
    u32 n,
    u32 k,
    const u32 *__restrict__ labels,
    const i32 *__restrict__ vwgt,
    unsigned long long *__restrict__ weights
) {
  extern __shared__ unsigned long long hist[];
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    hist[c] = 0;
  }
  __syncthreads();
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 u = blockIdx.x * blockDim.x + threadIdx.x; u < n; u += stride) {
    atomicAdd(&hist[labels[u]], static_cast<unsigned long long>(vwgt ? vwgt[u] : 1));
  }
  __syncthreads();
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    if (hist[c]) {
      atomicAdd(&weights[c], hist[c]);
    }
  }
}

// Grid-stride, 16-lane subgroups (4 vertices per wave) so low-degree rows
// keep lanes busy; per-wave partial sums, one atomic per wave.
__global__ void k_edge_cut(
    u32 n,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    unsigned long long *__restrict__ cut
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 sub = lane >> 4;
  const u32 slot = lane & 15;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 num_waves = (gridDim.x * blockDim.x) >> 6;
  u64 local = 0;
  for (u32 base = wave_id * 4; base < n; base += num_waves * 4) {
    const u32 u = base + sub;
    if (u >= n) {
      continue;
    }
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    const u32 lu = labels[u];
    for (u32 e = slot; e < deg; e += 16) {
      if (labels[adjncy[row + e]] != lu) {
        local += adjwgt ? adjwgt[row + e] : 1;
      }
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    local += __shfl_down(static_cast<unsigned long long>(local), off, kWave);
  }
  if (lane == 0 && local) {
    atomicAdd(cut, static_cast<unsigned long long>(local));
  }
}

// ==================== commit v2 (refine, k <= 256) ====================
// The round-1 commit issued ~30 launches + 3 rocprim calls + 2-3 host syncs
// per chunk (~85 ms/step control floor). v2 replaces it
// with 4 fixed-shape launches and ZERO host syncs: a stable counting sort
// by target block (k <= 256 distinct keys -- radix sort is overkill), then
// ONE kernel holding the whole admission fixpoint + weight update + label
// apply + active-set maintenance, using a hand-rolled resident-grid barrier
// (all blocks co-resident; hipLaunchCooperativeKernel semantics without the
// API so the launch stays graph-capturable). Admission semantics unchanged:
// per target, proposals are admitted in rank-prefix order under the
// greatest-fixpoint rollback of kaminpar-dist lp_refiner.cc:296-333.

// Refine L-path prep: per-vertex slice counts + exclusive prefix in one
// single-workgroup kernel (replaces k_l_sizes + a rocprim scan over l_cap).
__global__ void k_l_prep_r(
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    const u64 *__restrict__ xadj,
    u32 l_cap,
    u32 *__restrict__ l_off
) {
  __shared__ u32 red[17];
  const u32 count = *l_count < l_cap ? *l_count : l_cap;
  const u32 tid = threadIdx.x;
  u32 carry = 0;
  for (u32 base = 0; base < count; base += blockDim.x) {
    const u32 i = base + tid;
    u32 v = 0;
    if (i < count) {
      const u32 u = static_cast<u32>(l_list[i]);
      const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
      v = (deg + kLSlice - 1) / kLSlice;
    }
    u32 inc = v;
    for (int off = 1; off < 64; off <<= 1) {
      const u32 o = __shfl_up(inc, off, kWave);
      if ((tid & 63) >= static_cast<u32>(off)) {
        inc += o;
      }
    }
    __syncthreads();
    if ((tid & 63) == 63) {
      red[tid >> 6] = inc;
    }
    __syncthreads();
    u32 wbase = 0;
    for (u32 w = 0; w < (tid >> 6); ++w) {
      wbase += red[w];
    }
    if (i < count) {
      l_off[i] = carry + wbase + inc - v; // exclusive
    }
    u32 tsum = 0;
    for (u32 w = 0; w < blockDim.x / kWave; ++w) {
      tsum += red[w];
    }
    carry += tsum;
    __syncthreads();
  }
  if (tid == 0) {
    l_off[count] = carry;
  }
}

// Build the M and L work lists for a chunk straight from xadj/active under
// the permutation (no S-kernel involvement): each wave owns a run of
// 64-position tiles; pass 1 counts its M/L entries, ONE reservation atomic
// per wave per list, pass 2 writes the records. List order is wave-row
// order -- irrelevant for determinism since the M/L kernels write
// per-position slots.
__global__ void k_build_lists(
    u32 pos_lo,
    u32 pos_hi, // may be an unaligned sub-range (sharded path)
    u32 n,
    u64 iter_seed,
    u32 rows,
    u32 tpw,
    u32 mid_lo, // kSmallDeg
    u32 mid_hi, // kMidDeg (refine) / kClusterMidDeg (cluster)
    u32 m2_hi,  // kClusterM2Deg (cluster; == mid_hi disables the M2 tier)
    u32 max_degree,
    const u64 *__restrict__ xadj,
    const uint8_t *__restrict__ active,
    const uint8_t *__restrict__ unit_active,
    u64 *__restrict__ m_list,
    u32 *__restrict__ m_count,
    u64 *__restrict__ m2_list,
    u32 *__restrict__ m2_count,
    u64 *__restrict__ l_list,
    u32 *__restrict__ l_count,
    unsigned long long *__restrict__ arcs // null: no tally (legacy path)
) {
  __shared__ unsigned long long arc_red[4];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 ta0 = pos_lo >> 6;                 // absolute unit-tile range
  const u32 T = ((pos_hi + 63) >> 6) - ta0;
  unsigned long long my_arcs = 0;
  const kmp::FeistelPerm fp(kmp::num_units(n), iter_seed);
  const u32 t0 = row * tpw;
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  u32 m_cnt = 0, m2_cnt = 0, l_cnt = 0;
  for (u32 pass = 0; pass < 2; ++pass) {
    u32 m_base = 0, m2_base = 0, l_base = 0;
    if (pass == 1) {
      // flush the arcs tally (one WG-aggregated atomic)
      if (arcs != nullptr) {
        for (int off = 32; off > 0; off >>= 1) {
          my_arcs += __shfl_down(my_arcs, off, kWave);
        }
        __syncthreads();
        if (lane == 0) {
          arc_red[threadIdx.x >> 6] = my_arcs;
        }
        __syncthreads();
        if (threadIdx.x == 0) {
          const unsigned long long t2 =
              arc_red[0] + arc_red[1] + arc_red[2] + arc_red[3];
          if (t2) {
            atomicAdd(arcs, t2);
          }
        }
        my_arcs = 0;
      }
      // reserve (lane 0; contention = one atomic per wave-row per list)
      if (lane == 0) {
        if (m_cnt) {
          m_base = atomicAdd(m_count, m_cnt);
        }
        if (m2_cnt) {
          m2_base = atomicAdd(m2_count, m2_cnt);
        }
        if (l_cnt) {
          l_base = atomicAdd(l_count, l_cnt);
        }
      }
      m_base = __shfl(m_base, 0, kWave);
      m2_base = __shfl(m2_base, 0, kWave);
      l_base = __shfl(l_base, 0, kWave);
      if (m_cnt == 0 && m2_cnt == 0 && l_cnt == 0) {
        return;
      }
    }
    for (u32 t = t0; t < t1; ++t) {
      u32 vb = 0;
      if (lane == 0) {
        vb = fp(ta0 + t);
      }
      vb = __shfl(vb, 0, kWave);
      if (!unit_active[vb]) {
        continue;
      }
      const u32 p = ((ta0 + t) << 6) + lane;
      const u32 u = vb * kmp::kUnit + lane;
      bool is_m = false, is_m2 = false, is_l = false;
      if (p >= pos_lo && p < pos_hi && u < n && active[u]) {
        const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
        if (deg <= max_degree) {
          is_m = deg > mid_lo && deg <= mid_hi;
          is_m2 = deg > mid_hi && deg <= m2_hi;
          is_l = deg > m2_hi;
          if (pass == 0 && arcs != nullptr) {
            my_arcs += deg; // arcs-scanned tally (was k_clear_active's job)
          }
        }
      }
      const unsigned long long mm = __ballot(is_m);
      const unsigned long long mm2 = __ballot(is_m2);
      const unsigned long long ll = __ballot(is_l);
      if (pass == 0) {
        m_cnt += __popcll(mm);
        m2_cnt += __popcll(mm2);
        l_cnt += __popcll(ll);
      } else {
        if (is_m) {
          m_list[m_base + __popcll(mm & ((1ull << lane) - 1))] =
              (static_cast<u64>(p) << 32) | u;
        }
        if (is_m2) {
          m2_list[m2_base + __popcll(mm2 & ((1ull << lane) - 1))] =
              (static_cast<u64>(p) << 32) | u;
        }
        if (is_l) {
          l_list[l_base + __popcll(ll & ((1ull << lane) - 1))] =
              (static_cast<u64>(p) << 32) | u;
        }
        m_base += __popcll(mm);
        m2_base += __popcll(mm2);
        l_base += __popcll(ll);
      }
    }
  }
}

// Resident-grid exclusive scan of the k x rows histogram (entries up to
// 65536; a single-workgroup scan is latency-bound at that size) + segment
// offsets + prefix_len/dep/changed init. Uses the same coop_bar as the
// fixpoint kernel.
__global__ void k_scan_coop(
    u32 k,
    u32 rows,
    u32 nblk,
    const u32 *__restrict__ histT,
    u32 *__restrict__ offT,
    u32 *__restrict__ seg_off, // k+1
    u32 *__restrict__ prefix_len,
    unsigned long long *__restrict__ dep, // 2*k
    int *__restrict__ changed2,           // int[2]
    i64 *__restrict__ blocksums,
    u32 *__restrict__ bar
);

// Per-wave-row histogram of proposal targets over the raw slot array.
// Each wave owns a contiguous run of 64-position tiles (= permutation
// units); fully-inactive units are skipped without reading their (stale)
// slots -- the phase-A kernels freshly write every slot of an ACTIVE unit.
__global__ void k_hist_v2(
    u32 span,
    u32 pos_lo,
    u32 n,
    u64 iter_seed,
    u32 k,
    u32 rows,
    u32 tpw,
    const Prop *__restrict__ slots,
    const uint8_t *__restrict__ unit_active,
    u32 *__restrict__ histT // k x rows, column-major by target
) {
  extern __shared__ u32 cnt_lds[];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * k;
  for (u32 c = lane; c < k; c += kWave) {
    cnt[c] = 0;
  }
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 T = span >> 6;
  const kmp::FeistelPerm fp(kmp::num_units(n), iter_seed);
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    u32 vb = 0;
    if (lane == 0) {
      vb = fp(pos_lo / kmp::kUnit + t);
    }
    vb = __shfl(vb, 0, kWave);
    if (!unit_active[vb]) {
      continue;
    }
    const Prop pr = slots[(t << 6) + lane];
    if (pr.to != kInvalid) {
      atomicAdd(&cnt[pr.to], 1u);
    }
  }
  for (u32 c = lane; c < k; c += kWave) {
    histT[c * rows + row] = cnt[c];
  }
}

// Stable scatter into target-sorted order. Each wave replays its tile run
// in position order; within a 64-slot tile, stable per-target ranks come
// from a 64-step shuffle waterfall (no LDS read-after-write ordering
// assumptions: the leader's base is broadcast by shuffle). Also freezes the
// source block of every proposal (labels16 gather, coalesced within the
// unit) so the fixpoint never gathers labels again.
__global__ void k_scatter_v2(
    u32 span,
    u32 pos_lo,
    u32 n,
    u64 iter_seed,
    u32 k,
    u32 rows,
    u32 tpw,
    const Prop *__restrict__ slots,
    const uint8_t *__restrict__ unit_active,
    const u32 *__restrict__ offT,
    const uint16_t *__restrict__ labels16,
    u32 *__restrict__ s_u,
    u32 *__restrict__ s_w,
    uint16_t *__restrict__ s_to,
    uint16_t *__restrict__ s_b,
    unsigned long long *__restrict__ dep // [0..k) departures, [k..2k) arrivals
) {
  // dynamic LDS: per-wave cnt (k u32 each) then one per-WG pair of u64
  // hists: full-admission departure weights by source block and incoming
  // weights by target (the fixpoint kernel starts from these totals).
  extern __shared__ u32 cnt_lds[];
  const u32 waves_per_wg = blockDim.x >> 6;
  unsigned long long *h_out =
      reinterpret_cast<unsigned long long *>(cnt_lds + waves_per_wg * k);
  unsigned long long *h_in = h_out + k;
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * k;
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  for (u32 c = lane; c < k; c += kWave) {
    cnt[c] = offT[c * rows + row];
  }
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    h_out[c] = 0;
    h_in[c] = 0;
  }
  __syncthreads();
  const u32 T = span >> 6;
  const kmp::FeistelPerm fp(kmp::num_units(n), iter_seed);
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    u32 vb = 0;
    if (lane == 0) {
      vb = fp(pos_lo / kmp::kUnit + t);
    }
    vb = __shfl(vb, 0, kWave);
    if (!unit_active[vb]) {
      continue;
    }
    const Prop pr = slots[(t << 6) + lane];
    const u32 c = pr.to;
    const bool valid = c != kInvalid;
    u32 rank = 0, total = 0, leader = 0;
    bool seen = false;
    for (u32 j = 0; j < kWave; ++j) {
      const u32 cj = __shfl(c, j, kWave);
      if (valid && cj == c) {
        ++total;
        if (j < lane) {
          ++rank;
        }
        if (!seen) {
          leader = j;
          seen = true;
        }
      }
    }
    u32 base0 = 0;
    if (valid && rank == 0) {
      base0 = cnt[c];
      cnt[c] = base0 + total;
    }
    const u32 base = __shfl(base0, leader, kWave);
    if (valid) {
      const u32 dst = base + rank;
      const uint16_t src = labels16[pr.u];
      s_u[dst] = pr.u;
      s_w[dst] = pr.w;
      s_to[dst] = static_cast<uint16_t>(c);
      s_b[dst] = src;
      atomicAdd(&h_out[src], static_cast<unsigned long long>(pr.w));
      atomicAdd(&h_in[c], static_cast<unsigned long long>(pr.w));
    }
  }
  __syncthreads();
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    if (h_out[c]) {
      atomicAdd(&dep[c], h_out[c]);
    }
    if (h_in[c]) {
      atomicAdd(&dep[k + c], h_in[c]);
    }
  }
}

// ==================== degree-bucket rearrangement (on-GPU) ====================
// The reference's default NodeOrdering::DEGREE_BUCKETS preprocessing
// (graphutils/permutator.cc:36-110): stable counting sort of vertices into
// 33 log2-degree buckets (deg-0 last) + CSR re-gather with neighbour
// relabeling. Device-resident so the drop-in path needs no host pass
// (bit-identical to the host kmp_rearrange_degree_buckets).
constexpr u32 kDbBuckets = 33;

__device__ inline u32 db_bucket_of(u32 deg) {
  return deg == 0 ? kDbBuckets - 1 : (31 - __clz(deg) + 1);
}

__global__ void k_db_hist(
    u32 n,
    u32 rows,
    u32 tpw,
    const u64 *__restrict__ xadj,
    u32 *__restrict__ histT // kDbBuckets x rows
) {
  __shared__ u32 cnt_lds[4 * kDbBuckets];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * kDbBuckets;
  for (u32 b = lane; b < kDbBuckets; b += kWave) {
    cnt[b] = 0;
  }
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 T = (n + 63) >> 6;
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    const u32 u = (t << 6) + lane;
    if (u < n) {
      atomicAdd(&cnt[db_bucket_of(static_cast<u32>(xadj[u + 1] - xadj[u]))], 1u);
    }
  }
  for (u32 b = lane; b < kDbBuckets; b += kWave) {
    histT[b * rows + row] = cnt[b];
  }
}

__global__ void k_db_scatter(
    u32 n,
    u32 rows,
    u32 tpw,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ offT,
    u32 *__restrict__ perm, // perm[u_old] = u_new
    u32 *__restrict__ inv   // inv[u_new] = u_old
) {
  __shared__ u32 cnt_lds[4 * kDbBuckets];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * kDbBuckets;
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  for (u32 b = lane; b < kDbBuckets; b += kWave) {
    cnt[b] = offT[b * rows + row];
  }
  const u32 T = (n + 63) >> 6;
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    const u32 u = (t << 6) + lane;
    const bool valid = u < n;
    const u32 b = valid ? db_bucket_of(static_cast<u32>(xadj[u + 1] - xadj[u]))
                        : 0xFFFFFFFFu;
    u32 rank = 0, total = 0, leader = 0;
    bool seen = false;
    for (u32 j = 0; j < kWave; ++j) {
      const u32 bj = __shfl(b, j, kWave);
      if (valid && bj == b) {
        ++total;
        if (j < lane) {
          ++rank;
        }
        if (!seen) {
          leader = j;
          seen = true;
        }
      }
    }
    u32 base0 = 0;
    if (valid && rank == 0) {
      base0 = cnt[b];
      cnt[b] = base0 + total;
    }
    const u32 base = __shfl(base0, leader, kWave);
    if (valid) {
      const u32 pos = base + rank;
      perm[u] = pos;
      inv[pos] = u;
    }
  }
}

__global__ void k_db_degrees(
    u32 n,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ inv,
    u64 *__restrict__ new_deg // n+1 entries ([0] = 0 written by host memset)
) {
  const u32 v = blockIdx.x * blockDim.x + threadIdx.x;
  if (v < n) {
    const u32 u = inv[v];
    new_deg[v + 1] = xadj[u + 1] - xadj[u];
  }
}

// 16-lane subgroups, 4 vertices/wave: copy each new row from its old
// position, relabeling neighbours through perm.
__global__ void k_db_gather(
    u32 n,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ perm,
    const u32 *__restrict__ inv,
    const u64 *__restrict__ new_xadj,
    u32 *__restrict__ new_adjncy,
    i32 *__restrict__ new_adjwgt,
    i32 *__restrict__ new_vwgt
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 sub = lane >> 4;
  const u32 slot = lane & 15;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 num_waves = (gridDim.x * blockDim.x) >> 6;
  for (u32 base = wave_id * 4; base < n; base += num_waves * 4) {
    const u32 v = base + sub;
    if (v >= n) {
      continue;
    }
    const u32 u = inv[v];
    const u64 src = xadj[u];
    const u64 dst = new_xadj[v];
    const u32 deg = static_cast<u32>(xadj[u + 1] - src);
    for (u32 i = slot; i < deg; i += 16) {
      new_adjncy[dst + i] = perm[adjncy[src + i]];
      if (adjwgt != nullptr) {
        new_adjwgt[dst + i] = adjwgt[src + i];
      }
    }
    if (slot == 0 && vwgt != nullptr) {
      new_vwgt[v] = vwgt[u];
    }
  }
}

// ==================== sharded commit (multi-GPU) ====================
// Rank-sharded deterministic admission over the ALL-GATHERED proposal list
// (dense Prop records in global rank order): each rank sorts and fixpoints
// only its own target range [c_lo, c_hi); the k-sized de-admission delta is
// allreduced per round (exactly the block-weight reconciliation of
// kaminpar-dist/refinement/lp/lp_refiner.cc:296-333) and the final
// per-target rank-cutoffs are exchanged so every rank applies the identical
// admitted set. Bit-identical to the single-GPU commit: the per-round
// global departure sums match the monolithic fixpoint's rounds.

// Histogram over the dense proposal list, filtered to the rank's targets.
__global__ void k_hist_props(
    const u32 *__restrict__ count_dev,
    u32 k,
    u32 rows,
    u32 tpw,
    u32 c_lo,
    u32 c_hi,
    const Prop *__restrict__ props,
    u32 *__restrict__ histT
) {
  extern __shared__ u32 cnt_lds[];
  const u32 count = *count_dev;
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * k;
  for (u32 c = lane; c < k; c += kWave) {
    cnt[c] = 0;
  }
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 T = (count + 63) >> 6;
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    const u32 i = (t << 6) + lane;
    if (i < count) {
      const u32 to = props[i].to;
      if (to >= c_lo && to < c_hi) {
        atomicAdd(&cnt[to], 1u);
      }
    }
  }
  for (u32 c = lane; c < k; c += kWave) {
    histT[c * rows + row] = cnt[c];
  }
}

// Stable scatter of the rank's targets + full-admission dep/arr totals:
// dep contributions (by SOURCE block, over the rank's own-target proposals)
// go to the caller's allreduce buffer; arrival totals stay local.
__global__ void k_scatter_props(
    const u32 *__restrict__ count_dev,
    u32 k,
    u32 rows,
    u32 tpw,
    u32 c_lo,
    u32 c_hi,
    const Prop *__restrict__ props,
    const u32 *__restrict__ offT,
    const u32 *__restrict__ labels,
    u32 *__restrict__ s_u,
    u32 *__restrict__ s_w,
    u32 *__restrict__ s_r,
    uint16_t *__restrict__ s_to,
    uint16_t *__restrict__ s_b,
    long long *__restrict__ dep_out,          // k+1 (caller buffer, zeroed)
    unsigned long long *__restrict__ arr_loc  // e->d_dep + k
) {
  extern __shared__ u32 cnt_lds[];
  const u32 waves_per_wg = blockDim.x >> 6;
  unsigned long long *h_out =
      reinterpret_cast<unsigned long long *>(cnt_lds + waves_per_wg * k);
  unsigned long long *h_in = h_out + k;
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = threadIdx.x >> 6;
  u32 *cnt = cnt_lds + wv * k;
  const u32 row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  for (u32 c = lane; c < k; c += kWave) {
    cnt[c] = offT[c * rows + row];
  }
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    h_out[c] = 0;
    h_in[c] = 0;
  }
  __syncthreads();
  const u32 count = *count_dev;
  const u32 T = (count + 63) >> 6;
  const u32 t1 = (row + 1) * tpw < T ? (row + 1) * tpw : T;
  for (u32 t = row * tpw; t < t1; ++t) {
    const u32 i = (t << 6) + lane;
    Prop pr{0u, kInvalid, 0u, 0u};
    if (i < count) {
      pr = props[i];
    }
    const u32 c = pr.to;
    const bool valid = c != kInvalid && c >= c_lo && c < c_hi;
    u32 rank = 0, total = 0, leader = 0;
    bool seen = false;
    for (u32 j = 0; j < kWave; ++j) {
      const u32 cj = __shfl(c, j, kWave);
      const int vj = __shfl(static_cast<int>(valid), j, kWave);
      if (valid && vj && cj == c) {
        ++total;
        if (j < lane) {
          ++rank;
        }
        if (!seen) {
          leader = j;
          seen = true;
        }
      }
    }
    u32 base0 = 0;
    if (valid && rank == 0) {
      base0 = cnt[c];
      cnt[c] = base0 + total;
    }
    const u32 base = __shfl(base0, leader, kWave);
    if (valid) {
      const u32 dst = base + rank;
      const uint16_t srcb = static_cast<uint16_t>(labels[pr.u]);
      s_u[dst] = pr.u;
      s_w[dst] = pr.w;
      s_r[dst] = pr.rank;
      s_to[dst] = static_cast<uint16_t>(c);
      s_b[dst] = srcb;
      atomicAdd(&h_out[srcb], static_cast<unsigned long long>(pr.w));
      atomicAdd(&h_in[c], static_cast<unsigned long long>(pr.w));
    }
  }
  __syncthreads();
  for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
    if (h_out[c]) {
      atomicAdd(reinterpret_cast<unsigned long long *>(&dep_out[c]), h_out[c]);
    }
    if (h_in[c]) {
      atomicAdd(&arr_loc[c], h_in[c]);
    }
  }
}

// Copy the device proposal count into the send buffer's trailing row so a
// single fixed-size allgather moves payload + counts with no host sync.
__global__ void k_set_u32(u32 *__restrict__ dst, u32 v) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    *dst = v;
  }
}

__global__ void k_pack_count(const u32 *__restrict__ count_dev, Prop *__restrict__ trailing) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    trailing->u = *count_dev;
  }
}

// Dense-pack the gathered per-rank segments (cap+1 rows each, trailing row
// carries the rank's count) into e->d_props, in rank order (= global rank
// order since rank position slices ascend); writes the total count.
__global__ void k_compact_gathered(
    u32 world,
    u32 cap,
    const Prop *__restrict__ recv, // world * (cap+1) rows
    Prop *__restrict__ dense,
    u32 *__restrict__ total_out
) {
  __shared__ u32 cnt_s[8], off_s[9];
  const u32 tid = blockIdx.x * blockDim.x + threadIdx.x;
  if (threadIdx.x < world) {
    cnt_s[threadIdx.x] = recv[(static_cast<size_t>(threadIdx.x) * (cap + 1) + cap)].u;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    u32 acc = 0;
    for (u32 r = 0; r < world; ++r) {
      off_s[r] = acc;
      acc += cnt_s[r];
    }
    off_s[world] = acc;
    if (blockIdx.x == 0) {
      *total_out = acc;
    }
  }
  __syncthreads();
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 r = 0; r < world; ++r) {
    const u32 n_r = cnt_s[r];
    const Prop *src_r = recv + static_cast<size_t>(r) * (cap + 1);
    Prop *dst_r = dense + off_s[r];
    for (u32 i = tid; i < n_r; i += stride) {
      dst_r[i] = src_r[i];
    }
  }
}

__global__ void k_vec_sub_i64(u32 n, long long *__restrict__ a, const long long *__restrict__ b) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    a[i] -= b[i];
  }
}

// Activation over the dense proposal list with a device count (wave per
// admitted entry; the C++ dist driver's analogue of k_activate).
__global__ void k_activate_props_dev(
    const u32 *__restrict__ count_dev,
    const u32 *__restrict__ admitted_flags,
    const Prop *__restrict__ props,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    uint8_t *__restrict__ active,
    uint8_t *__restrict__ unit_active
) {
  const u32 count = *count_dev;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 waves = (gridDim.x * blockDim.x) >> 6;
  const u32 lane = threadIdx.x & (kWave - 1);
  for (u32 i = wave_id; i < count; i += waves) {
    if (!admitted_flags[i]) {
      continue;
    }
    const u32 u = props[i].u;
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    for (u32 e2 = lane; e2 < deg; e2 += kWave) {
      const u32 v = adjncy[row + e2];
      active[v] = 1;
      unit_active[v >> 6] = 1;
    }
  }
}

// Segmented prefix weights for the rank's targets (weighted graphs only;
// wave per segment, tile scans).
__global__ void k_shard_pw(
    u32 k,
    u32 c_lo,
    u32 c_hi,
    const u32 *__restrict__ seg_off,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ s_w,
    i64 *__restrict__ pw
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wv = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 waves = (gridDim.x * blockDim.x) >> 6;
  for (u32 c = c_lo + wv; c < c_hi; c += waves) {
    const u32 b = seg_off[c];
    const u32 len = prefix_len[c];
    i64 carry = 0;
    for (u32 base = 0; base < len; base += kWave) {
      const u32 i = base + lane;
      i64 v = (i < len) ? static_cast<i64>(s_w[b + i]) : 0;
      i64 inc = v;
      for (int off = 1; off < 64; off <<= 1) {
        const i64 o = __shfl_up(inc, off, kWave);
        if (lane >= static_cast<u32>(off)) {
          inc += o;
        }
      }
      if (i < len) {
        pw[b + i] = carry + inc;
      }
      carry += __shfl(inc, kWave - 1, kWave);
    }
  }
}

// One synchronous fixpoint round over the rank's targets: capacities frozen
// from the GLOBAL departure vector, de-admissions accumulated into the
// caller's delta buffer ([k] = changed flag). plen/arr persist in
// d_prefix_len / d_dep[k..2k).
__global__ void k_shard_round(
    u32 k,
    u32 c_lo,
    u32 c_hi,
    u32 has_vwgt,
    const u32 *__restrict__ seg_off,
    u32 *__restrict__ prefix_len,
    unsigned long long *__restrict__ arr_loc, // d_dep + k (live)
    const long long *__restrict__ dep_global, // k (allreduced)
    const u32 *__restrict__ s_w,
    const uint16_t *__restrict__ s_b,
    const i64 *__restrict__ pw, // segmented prefix weights (weighted only)
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    long long *__restrict__ delta_out // k+1 (caller buffer, zeroed here)
) {
  __shared__ unsigned long long ddelta[256];
  __shared__ u32 rlo_s[256], rhi_s[256];
  __shared__ int chg;
  const u32 tid = threadIdx.x;
  if (tid < k) {
    ddelta[tid] = 0;
  }
  if (tid == 0) {
    chg = 0;
  }
  __syncthreads();
  if (tid < k) {
    const u32 c = tid;
    u32 nlo = 0, nhi = 0;
    if (c >= c_lo && c < c_hi && prefix_len[c] > 0) {
      const i64 cap = maxw[c] - weights[c] + static_cast<i64>(dep_global[c]);
      const long long a = static_cast<long long>(arr_loc[c]);
      if (a > cap) {
        const u32 old = prefix_len[c];
        const u32 b = seg_off[c];
        u32 nl;
        if (!has_vwgt) {
          nl = cap <= 0 ? 0u : (cap >= static_cast<i64>(old) ? old : static_cast<u32>(cap));
          arr_loc[c] = nl;
        } else {
          u32 lo2 = 0, hi2 = old;
          while (lo2 < hi2) {
            const u32 mid = (lo2 + hi2 + 1) >> 1;
            if (pw[b + mid - 1] <= cap) {
              lo2 = mid;
            } else {
              hi2 = mid - 1;
            }
          }
          nl = lo2;
          arr_loc[c] = nl ? static_cast<unsigned long long>(pw[b + nl - 1]) : 0ull;
        }
        if (nl < old) {
          prefix_len[c] = nl;
          nlo = b + nl;
          nhi = b + old;
          chg = 1;
        }
      }
    }
    rlo_s[c] = nlo;
    rhi_s[c] = nhi;
  }
  __syncthreads();
  for (u32 c = 0; c < k; ++c) {
    const u32 lo2 = rlo_s[c], hi2 = rhi_s[c];
    for (u32 i = lo2 + tid; i < hi2; i += blockDim.x) {
      atomicAdd(&ddelta[s_b[i]], has_vwgt ? static_cast<unsigned long long>(s_w[i]) : 1ull);
    }
  }
  __syncthreads();
  if (tid < k) {
    delta_out[tid] = static_cast<long long>(ddelta[tid]);
  }
  if (tid == 0) {
    delta_out[k] = chg;
  }
}

// Export the rank's final per-target cutoff ranks + admitted arrival
// weights into caller buffers (zero elsewhere; an allreduce-sum doubles as
// the allgather).
__global__ void k_shard_finish_meta(
    u32 k,
    u32 c_lo,
    u32 c_hi,
    const u32 *__restrict__ seg_off,
    const u32 *__restrict__ prefix_len,
    const unsigned long long *__restrict__ arr_loc,
    const u32 *__restrict__ s_r,
    unsigned long long *__restrict__ cutoff_out, // k i64 (zeroed by caller)
    long long *__restrict__ arr_out              // k i64 (zeroed by caller)
) {
  const u32 c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c < c_lo || c >= c_hi) {
    return;
  }
  const u32 len = prefix_len[c];
  const u32 seg_len = seg_off[c + 1] - seg_off[c];
  // cutoff = rank of the first NON-admitted proposal (all ranks below it
  // are admitted: the sorted order is rank order within the target); i64
  // cells so a plain allreduce-sum doubles as the allgather
  cutoff_out[c] = (len >= seg_len) ? ~0ull : s_r[seg_off[c] + len];
  arr_out[c] = static_cast<long long>(arr_loc[c]);
}

// Apply the globally-agreed admission on the full proposal list + update
// all block weights; every rank executes this identically.
__global__ void k_shard_apply(
    const u32 *__restrict__ count_dev,
    u32 k,
    const Prop *__restrict__ props,
    const unsigned long long *__restrict__ cutoff_all, // k (allreduced)
    const long long *__restrict__ arr_all,   // k (allreduced)
    const long long *__restrict__ dep_global, // k (final)
    i64 *__restrict__ weights,
    u32 *__restrict__ labels,
    uint16_t *__restrict__ labels16,
    uint8_t *__restrict__ labels8, // null unless k <= 256
    u32 *__restrict__ admitted_flags,
    unsigned long long *__restrict__ moves
) {
  const u32 count = *count_dev;
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < k) {
    const i64 delta = static_cast<i64>(arr_all[i]) - static_cast<i64>(dep_global[i]);
    if (delta) {
      weights[i] += delta;
    }
  }
  unsigned long long local = 0;
  if (i < count) {
    const Prop pr = props[i];
    const bool adm = pr.rank < cutoff_all[pr.to];
    admitted_flags[i] = adm ? 1u : 0u;
    if (adm) {
      labels[pr.u] = pr.to;
      labels16[pr.u] = static_cast<uint16_t>(pr.to);
      if (labels8 != nullptr) {
        labels8[pr.u] = static_cast<uint8_t>(pr.to);
      }
      local = 1;
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    local += __shfl_down(static_cast<unsigned long long>(local), off, kWave);
  }
  if ((threadIdx.x & (kWave - 1)) == 0 && local) {
    atomicAdd(moves, local);
  }
}

// Resident-grid barrier: flat arrival counter + generation release, relaxed
// agent-scope atomics bracketed by __threadfence (measured the fastest
// variant on gfx950: ~3 us at 64 blocks vs ~20 us for acq_rel two-level --
// tools/bar_bench.hip). Requires every block co-resident (grid sized from
// occupancy, capped small since the commit stages are latency- not
// bandwidth-bound).
__device__ inline void coop_bar(u32 *bar, u32 nblk) {
  __syncthreads();
  if (threadIdx.x == 0) {
    u32 *root = bar + 8;
    u32 *gen = bar + 9;
    __threadfence(); // publish this block's writes before arrival
    const u32 g = __hip_atomic_load(gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (__hip_atomic_fetch_add(root, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) ==
        nblk - 1) {
      __hip_atomic_store(root, 0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_fetch_add(gen, 1u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
    } else {
      // deadman: if residency was ever violated this spin would hang the
      // device -- trap instead so the failure is a visible abort
      u32 spins = 0;
      while (__hip_atomic_load(gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) == g) {
        __builtin_amdgcn_s_sleep(1);
        if (++spins > 400000000u) {
          __builtin_trap();
        }
      }
    }
    __threadfence(); // see every other block's published writes
  }
  __syncthreads();
}

__global__ void k_scan_coop(
    u32 k,
    u32 rows,
    u32 nblk,
    const u32 *__restrict__ histT,
    u32 *__restrict__ offT,
    u32 *__restrict__ seg_off,
    u32 *__restrict__ prefix_len,
    unsigned long long *__restrict__ dep,
    int *__restrict__ changed2,
    i64 *__restrict__ blocksums,
    u32 *__restrict__ bar
) {
  __shared__ u32 red[5];
  const u32 tid = threadIdx.x;
  const u32 lane = tid & (kWave - 1);
  const u32 entries = k * rows;
  const u32 rlen = (entries + nblk - 1) / nblk;
  const u32 lo = blockIdx.x * rlen < entries ? blockIdx.x * rlen : entries;
  const u32 hi = lo + rlen < entries ? lo + rlen : entries;

  // stage 1: block-local exclusive scan of [lo, hi) into offT + block total
  u32 carry = 0;
  for (u32 base = lo; base < hi; base += blockDim.x) {
    const u32 i = base + tid;
    u32 v = i < hi ? histT[i] : 0;
    u32 inc = v;
    for (int off = 1; off < 64; off <<= 1) {
      const u32 o = __shfl_up(inc, off, kWave);
      if (lane >= static_cast<u32>(off)) {
        inc += o;
      }
    }
    __syncthreads();
    if (lane == 63) {
      red[tid >> 6] = inc;
    }
    __syncthreads();
    u32 wbase = 0;
    for (u32 w = 0; w < (tid >> 6); ++w) {
      wbase += red[w];
    }
    if (i < hi) {
      offT[i] = carry + wbase + inc - v; // local exclusive
    }
    u32 tsum = 0;
    for (u32 w = 0; w < blockDim.x / kWave; ++w) {
      tsum += red[w];
    }
    carry += tsum;
    __syncthreads();
  }
  if (tid == 0) {
    blocksums[blockIdx.x] = static_cast<i64>(carry);
  }
  coop_bar(bar, nblk);
  // stage 2: block 0 exclusive-scans the block totals; stashes the grand
  // total at blocksums[nblk]
  if (blockIdx.x == 0 && tid == 0) {
    i64 run = 0;
    for (u32 b = 0; b < nblk; ++b) {
      const i64 v = blocksums[b];
      blocksums[b] = run;
      run += v;
    }
    blocksums[nblk] = run;
  }
  coop_bar(bar, nblk);
  // stage 3: add carries
  {
    const u32 bc = static_cast<u32>(blocksums[blockIdx.x]);
    if (bc) {
      for (u32 i = lo + tid; i < hi; i += blockDim.x) {
        offT[i] += bc;
      }
    }
  }
  coop_bar(bar, nblk);
  // stage 4: segment offsets + fixpoint init
  const u32 total = static_cast<u32>(blocksums[nblk]);
  const u32 gid = blockIdx.x * blockDim.x + tid;
  const u32 gsz = nblk * blockDim.x;
  for (u32 c = gid; c < k; c += gsz) {
    const u32 b = offT[c * rows];
    const u32 e2 = (c + 1 < k) ? offT[(c + 1) * rows] : total;
    seg_off[c] = b;
    prefix_len[c] = e2 - b;
    dep[c] = 0;
    dep[k + c] = 0;
  }
  if (gid == 0) {
    seg_off[k] = total;
    changed2[0] = 0;
    changed2[1] = 0;
  }
}

// The admission fixpoint in ONE single-workgroup launch (no grid
// barriers, trivially stream-ordered). Starts from the full-admission
// state the scatter kernel accumulated (dep = departures by source block,
// arr = incoming weight by target) and walks each target's rank-cutoff
// BACKWARD per round under capacities frozen at round start -- exactly the
// synchronous rollback rounds of kaminpar-dist lp_refiner.cc:296-333, with
// O(de-admitted) work per round instead of a full rescan. Bit-identical to
// the legacy commit (same per-round cutoffs, greatest fixpoint).
__global__ void k_fixpoint_v2(
    u32 k,
    u32 has_vwgt,
    const u32 *__restrict__ seg_off,
    u32 *__restrict__ prefix_len,
    const unsigned long long *__restrict__ dep, // [0..k) out, [k..2k) in
    const u32 *__restrict__ s_w,
    const uint16_t *__restrict__ s_b,
    i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    unsigned long long *__restrict__ moves,
    i64 *__restrict__ pw // scratch: segmented prefix weights (weighted only)
) {
  __shared__ unsigned long long ddep[256];   // current departures
  __shared__ unsigned long long ddelta[256]; // this round's de-admissions
  __shared__ long long arr_s[256];           // admitted incoming weight
  __shared__ u32 plen_s[256];
  __shared__ u32 soff_s[256];
  __shared__ u32 rlo_s[256], rhi_s[256]; // this round's de-admitted ranges
  __shared__ unsigned long long red[4];
  __shared__ int chg;
  const u32 tid = threadIdx.x;
  if (tid < k) {
    ddep[tid] = dep[tid];
    arr_s[tid] = static_cast<long long>(dep[k + tid]);
    plen_s[tid] = prefix_len[tid];
    soff_s[tid] = seg_off[tid];
    ddelta[tid] = 0;
    rlo_s[tid] = 0;
    rhi_s[tid] = 0;
  }
  __syncthreads();
  if (has_vwgt) {
    // build segmented prefix weights once (wave per target, tile scans) so
    // per-round cutoffs are binary searches instead of serial walks
    const u32 lane = tid & (kWave - 1);
    const u32 wv = tid >> 6;
    for (u32 c = wv; c < k; c += blockDim.x / kWave) {
      const u32 b = soff_s[c];
      const u32 len = plen_s[c];
      i64 carry = 0;
      for (u32 base = 0; base < len; base += kWave) {
        const u32 i = base + lane;
        i64 v = (i < len) ? static_cast<i64>(s_w[b + i]) : 0;
        i64 inc = v;
        for (int off = 1; off < 64; off <<= 1) {
          const i64 o = __shfl_up(inc, off, kWave);
          if (lane >= static_cast<u32>(off)) {
            inc += o;
          }
        }
        if (i < len) {
          pw[b + i] = carry + inc;
        }
        carry += __shfl(inc, kWave - 1, kWave);
      }
    }
    __syncthreads();
  }
  for (;;) {
    if (tid == 0) {
      chg = 0;
    }
    i64 cap = 0;
    if (tid < k) {
      // capacity frozen at round start (synchronous rounds)
      cap = maxw[tid] - weights[tid] + static_cast<i64>(ddep[tid]);
    }
    __syncthreads();
    if (tid < k && plen_s[tid] > 0 && arr_s[tid] > cap) {
      const u32 old = plen_s[tid];
      const u32 b = soff_s[tid];
      u32 nl;
      if (!has_vwgt) {
        // unit weights: admitted weight of a prefix IS its length
        nl = cap <= 0 ? 0u : (cap >= static_cast<i64>(old) ? old : static_cast<u32>(cap));
        arr_s[tid] = static_cast<long long>(nl);
      } else {
        // weighted: binary search the segmented prefix weights
        u32 lo2 = 0, hi2 = old;
        while (lo2 < hi2) {
          const u32 mid = (lo2 + hi2 + 1) >> 1;
          if (pw[b + mid - 1] <= cap) {
            lo2 = mid;
          } else {
            hi2 = mid - 1;
          }
        }
        nl = lo2;
        arr_s[tid] = nl ? pw[b + nl - 1] : 0;
      }
      plen_s[tid] = nl;
      rlo_s[tid] = b + nl;
      rhi_s[tid] = b + old;
      chg = 1; // benign same-value race
    }
    __syncthreads();
    if (!chg) {
      break;
    }
    // block-parallel de-admission deltas (coalesced; O(total de-admitted))
    for (u32 c = 0; c < k; ++c) {
      const u32 lo2 = rlo_s[c], hi2 = rhi_s[c];
      for (u32 i = lo2 + tid; i < hi2; i += blockDim.x) {
        atomicAdd(&ddelta[s_b[i]], has_vwgt ? static_cast<unsigned long long>(s_w[i]) : 1ull);
      }
    }
    __syncthreads();
    if (tid < k) {
      if (ddelta[tid]) {
        ddep[tid] -= ddelta[tid];
        ddelta[tid] = 0;
      }
      rlo_s[tid] = 0;
      rhi_s[tid] = 0;
    }
    __syncthreads();
  }

  // weights + writeback + move count
  unsigned long long mv = 0;
  if (tid < k) {
    const long long delta = arr_s[tid] - static_cast<long long>(ddep[tid]);
    if (delta) {
      weights[tid] += delta;
    }
    prefix_len[tid] = plen_s[tid];
    mv = plen_s[tid];
  }
  for (int off = 32; off > 0; off >>= 1) {
    mv += __shfl_down(mv, off, kWave);
  }
  if ((tid & (kWave - 1)) == 0) {
    red[tid >> 6] = mv;
  }
  __syncthreads();
  if (tid == 0) {
    unsigned long long t = 0;
    for (u32 w = 0; w < blockDim.x / kWave; ++w) {
      t += red[w];
    }
    if (t) {
      atomicAdd(moves, t);
    }
  }
}

// Single-workgroup variant of the histogram scan for small chunks (entries
// <= a few thousand; the resident-grid version's barriers dominate there).
__global__ void k_scan_small(
    u32 k,
    u32 rows,
    const u32 *__restrict__ histT,
    u32 *__restrict__ offT,
    u32 *__restrict__ seg_off,
    u32 *__restrict__ prefix_len,
    unsigned long long *__restrict__ dep,
    int *__restrict__ changed2
) {
  __shared__ u32 red[5];
  const u32 entries = k * rows;
  const u32 tid = threadIdx.x;
  const u32 lane = tid & (kWave - 1);
  u32 carry = 0;
  for (u32 base = 0; base < entries; base += blockDim.x) {
    const u32 i = base + tid;
    u32 v = i < entries ? histT[i] : 0;
    u32 inc = v;
    for (int off = 1; off < 64; off <<= 1) {
      const u32 o = __shfl_up(inc, off, kWave);
      if (lane >= static_cast<u32>(off)) {
        inc += o;
      }
    }
    __syncthreads();
    if (lane == 63) {
      red[tid >> 6] = inc;
    }
    __syncthreads();
    u32 wbase = 0;
    for (u32 w = 0; w < (tid >> 6); ++w) {
      wbase += red[w];
    }
    if (i < entries) {
      offT[i] = carry + wbase + inc - v; // exclusive
    }
    u32 tsum = 0;
    for (u32 w = 0; w < blockDim.x / kWave; ++w) {
      tsum += red[w];
    }
    carry += tsum;
    __syncthreads();
  }
  __threadfence_block();
  __syncthreads();
  for (u32 c = tid; c < k; c += blockDim.x) {
    const u32 b = offT[c * rows];
    const u32 e2 = (c + 1 < k) ? offT[(c + 1) * rows] : carry;
    seg_off[c] = b;
    prefix_len[c] = e2 - b;
    dep[c] = 0;
    dep[k + c] = 0;
  }
  if (tid == 0) {
    seg_off[k] = carry;
    changed2[0] = 0;
    changed2[1] = 0;
  }
}

// Serial rank-order admission for underload-balancer mode: both per-block
// minima (source side) and maxima (target side) are re-checked against live
// weights per move, in deterministic rank order -- the batch analogue of
// underload_balancer.cc's locked per-move checks. Proposal counts in this
// mode are small (only vertices adjacent to underloaded blocks propose).
__global__ void k_commit_underload(
    const Prop *__restrict__ props,
    const u32 *__restrict__ count_dev,
    const i64 *__restrict__ maxw,
    const i64 *__restrict__ minw,
    i64 *__restrict__ weights,
    u32 *__restrict__ labels,
    uint16_t *__restrict__ labels16,
    uint8_t *__restrict__ labels8, // null unless k <= 256
    u32 *__restrict__ admitted_flags,
    unsigned long long *__restrict__ moves
) {
  if (threadIdx.x != 0 || blockIdx.x != 0) {
    return;
  }
  const u32 count = *count_dev;
  unsigned long long mv = 0;
  for (u32 i = 0; i < count; ++i) {
    const Prop pr = props[i];
    const u32 from = labels[pr.u];
    const i64 w = static_cast<i64>(static_cast<i32>(pr.w));
    const i64 mn_from = minw[from];
    const i64 mn_to = minw[pr.to];
    const bool ok = weights[from] >= mn_from && weights[from] - w >= mn_from &&
                    weights[pr.to] < mn_to && weights[pr.to] + w <= maxw[pr.to];
    admitted_flags[i] = ok ? 1u : 0u;
    if (ok) {
      weights[from] -= w;
      weights[pr.to] += w;
      labels[pr.u] = pr.to;
      labels16[pr.u] = static_cast<uint16_t>(pr.to);
      if (labels8 != nullptr) {
        labels8[pr.u] = static_cast<uint8_t>(pr.to);
      }
      ++mv;
    }
  }
  if (mv) {
    atomicAdd(moves, mv);
  }
}

// Apply admitted labels + clear the chunk's unit-active bits (full grid;
// the per-vertex flags were cleared inline by k_phase_s, and activation --
// which re-marks receiving units -- runs in the NEXT kernel, so stream
// order preserves the clear-then-activate semantics).
__global__ void k_apply_v2(
    u32 k,
    u32 pos_lo,
    u32 pos_hi,
    u32 n,
    u64 iter_seed,
    const u32 *__restrict__ seg_off,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ s_u,
    const uint16_t *__restrict__ s_to,
    u32 *__restrict__ labels,
    uint16_t *__restrict__ labels16,
    uint8_t *__restrict__ labels8, // null unless k <= 256
    uint8_t *__restrict__ unit_active
) {
  const u32 count = seg_off[k];
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < count; i += stride) {
    const u32 c = s_to[i];
    if (i - seg_off[c] < prefix_len[c]) {
      const u32 u = s_u[i];
      labels[u] = c;
      labels16[u] = static_cast<uint16_t>(c);
      if (labels8 != nullptr) {
        labels8[u] = static_cast<uint8_t>(c);
      }
    }
  }
  {
    const kmp::FeistelPerm fp(kmp::num_units(n), iter_seed);
    const u32 units = (pos_hi - pos_lo) >> 6;
    const u32 u0 = pos_lo >> 6;
    for (u32 uix = blockIdx.x * blockDim.x + threadIdx.x; uix < units; uix += stride) {
      const u32 vb = fp(u0 + uix);
      if (unit_active[vb]) {
        unit_active[vb] = 0;
      }
    }
  }
}

// Activate neighbours of admitted movers (wave per admitted entry).
__global__ void k_activate_v2(
    u32 k,
    const u32 *__restrict__ seg_off,
    const u32 *__restrict__ prefix_len,
    const u32 *__restrict__ s_u,
    const uint16_t *__restrict__ s_to,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    uint8_t *__restrict__ active,
    uint8_t *__restrict__ unit_active
) {
  const u32 count = seg_off[k];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 waves_total = (gridDim.x * blockDim.x) >> 6;
  for (u32 i = (blockIdx.x * blockDim.x + threadIdx.x) >> 6; i < count; i += waves_total) {
    const u32 c = s_to[i];
    if (i - seg_off[c] >= prefix_len[c]) {
      continue;
    }
    const u32 u = s_u[i];
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    for (u32 e2 = lane; e2 < deg; e2 += kWave) {
      const u32 v = adjncy[row + e2];
      // blind stores: measured FASTER than read-test-write (the reads add
      // serialized gather latency; the dirty-line cost is smaller)
      active[v] = 1;
      unit_active[v >> 6] = 1;
    }
  }
}

} // namespace

// ================================================================ engine
struct kmp_lp_t {
  u32 n = 0;
  u64 m = 0;
  u32 k = 0;
  u64 seed = 1;
  u32 C = 0; // positions per chunk
  u32 P = 0; // total positions (pos_count)
  bool has_vwgt = false, has_adjwgt = false;

  // device graph (xadj is u64 device-side: EdgeID-64 everywhere removes the
  // 2^32-arc ceiling at ~zero cost -- xadj reads are per-vertex + coalesced)
  u64 *d_xadj = nullptr;
  u32 *d_adjncy = nullptr;
  i32 *d_vwgt = nullptr;
  i32 *d_adjwgt = nullptr;

  // device LP state
  u32 *d_labels = nullptr;
  u32 *d_labels0 = nullptr;   // initial labels (for kmp_lp_reset)
  uint8_t *d_labels8 = nullptr;   // u8 shadow for refine gathers (k <= 256:
                                  // 64 MB at scale 26, Infinity-Cache-resident)
  uint16_t *d_labels16 = nullptr; // u16 shadow for refine gathers (k <= 2048
                                  // fits; halves the gather cache footprint:
                                  // scale-26 labels 128 MB -> L3-resident)
  i64 *d_weights = nullptr;
  i64 *d_maxw = nullptr;
  i64 *d_minw = nullptr; // per-block minimums (underload mode; else null)
  u32 *d_comm = nullptr; // per-vertex communities (clusterer; else null)
  std::vector<u32> comm_host; // host copy (isolated-chain community breaks)
  uint8_t *d_active = nullptr;
  uint8_t *d_unit_active = nullptr; // one byte per 64-vertex unit

  // phase buffers
  Prop *d_slots = nullptr; // C
  Prop *d_props = nullptr; // C (compacted; legacy/sharded commit input)
  u64 *d_m_list = nullptr;  // C (built by k_build_lists)
  u32 *d_m_count = nullptr; // base of a 4-u32 alloc; [1]=l, [2]=m2
  u64 *d_l_list = nullptr; // C
  u32 *d_l_count = nullptr; // = d_m_count + 1
  u64 *d_m2_list = nullptr; // C (clustering per-WG hash tier)
  u32 *d_m2_count = nullptr; // = d_m_count + 2
  u32 *d_l_off = nullptr;   // l_cap + 1 (slice prefix)
  u32 *d_l_sizes = nullptr; // l_cap + 1 (slice counts, scan input)
  void *d_lscan_temp = nullptr;
  size_t lscan_temp_bytes = 0;
  i32 *d_l_gains = nullptr; // l_cap x k (allocated at refine_begin)
  u32 l_cap = 0;
  u32 *d_prop_count = nullptr;
  unsigned long long *d_arcs = nullptr;
  unsigned long long *d_moves = nullptr;

  // commit buffers
  u32 *d_sort_keys[2] = {nullptr, nullptr};
  u32 *d_sort_vals[2] = {nullptr, nullptr};
  void *d_sort_temp = nullptr;
  size_t sort_temp_bytes = 0;
  void *d_select_temp = nullptr;
  size_t select_temp_bytes = 0;
  i64 *d_sw = nullptr;
  i64 *d_pw = nullptr;
  void *d_scan_temp = nullptr;
  size_t scan_temp_bytes = 0;
  u32 *d_seg_begin = nullptr, *d_seg_end = nullptr, *d_prefix_len = nullptr;
  unsigned long long *d_dep = nullptr;
  int *d_changed = nullptr;
  u32 *d_admitted_flags = nullptr;
  unsigned long long *d_cut = nullptr;

  // commit v2 (refine, k <= 256): counting sort + cooperative fixpoint.
  // Replaces the per-chunk radix sort / scan-by-key / host fixpoint loop
  // with 4 fixed-shape launches and ZERO host syncs per chunk.
  u32 *d_s_u = nullptr;       // C (sorted-by-target source vertices)
  u32 *d_s_w = nullptr;       // C (their node weights)
  u32 *d_s_r = nullptr;       // C (global ranks; sharded-commit cutoffs)
  uint16_t *d_s_to = nullptr; // C (target block per sorted index)
  uint16_t *d_s_b = nullptr;  // C (source block, frozen at scatter)
  u32 *d_histT = nullptr;     // k x rows column-major histogram
  u32 *d_offT = nullptr;      // its exclusive scan
  u32 *d_seg_off = nullptr;   // k+1 segment offsets
  u32 *d_bar = nullptr;       // grid barrier state (sub[8], root, gen)
  i64 *d_blocksums = nullptr; // coop pw-scan block partials
  u32 coop_nblk = 0;          // resident-guaranteed grid for k_commit_coop
  u32 rows_v2 = 0;            // histogram rows for current k
  u32 max_deg = 0;            // max degree (gates the L-path launches)

  // clustering state
  u32 *d_favored = nullptr;      // n (two-hop favored clusters)
  uint8_t *d_eflag = nullptr;    // n (emptied-cluster flags)
  u32 *d_pool_keys = nullptr;    // pooled L hash
  i32 *d_pool_vals = nullptr;
  u32 *d_l_clist = nullptr;      // claimed-slot lists (pool_slots/2)
  u32 *d_l_ccnt = nullptr;       // C (per-L-vertex claimed counts)
  u64 pool_slots = 0;
  u64 *d_l_hoff = nullptr;       // l_cap+1
  u32 *d_l_hbits = nullptr;      // l_cap
  unsigned long long *d_l_hacc = nullptr; // total region demand of the chunk
  u64 *d_cand = nullptr;         // n (two-hop candidate slots)
  u64 *d_cand2 = nullptr;        // n (sorted)
  u32 *d_cfav = nullptr;         // n
  u32 *d_crank = nullptr;        // n
  u32 *d_cones = nullptr;        // n
  void *d_cand_select_temp = nullptr;
  size_t cand_select_temp_bytes = 0;
  void *d_cand_sort_temp = nullptr;
  size_t cand_sort_temp_bytes = 0;
  void *d_cand_scan_temp = nullptr;
  size_t cand_scan_temp_bytes = 0;
  unsigned long long *d_emptied = nullptr;
  std::vector<u32> isolated;   // host: vertices with degree 0
  std::vector<i32> iso_weights; // their node weights
  bool clusterer = false;
  int balance = 0; // balance mode: overloaded vertices lose "stay" (select)
  std::vector<i64> maxw_host; // refine caps (host copy, for the fallback)
  i64 maxw_uniform = 0;
  int mp_count = 0; // CUs (for the resident-grid commit kernel)

  // pinned host mirrors
  u32 *h_count = nullptr;
  int *h_changed = nullptr;
  unsigned long long *h_hacc = nullptr;
  unsigned long long *h_moves = nullptr; // [0..1]=moves before/after [2..3]=emptied
  u64 last_emptied = 0;

  hipStream_t stream = nullptr;
  hipStream_t own_stream = nullptr; // saved when an external stream is adopted

  // run bookkeeping
  double phase_a_ms = 0.0;
  double commit_ms = 0.0;
  std::vector<hipEvent_t> ev_pool;
  size_t ev_used = 0;

  hipEvent_t sync_ev = nullptr;

  // captured per-sweep launch trains (v2 refine): graph[i] replays sweep i
  // (all 64 chunk trains + timing events); rebuilt lazily after every
  // refine_begin (buffer pointers change). Replay cost ~launch-free.
  static constexpr int kMaxSweepGraphs = 16;
  hipGraphExec_t sweep_graph[kMaxSweepGraphs] = {};
  uint8_t sweep_seen[kMaxSweepGraphs] = {}; // capture only on the 2nd run
  bool graphs_ok = true; // capture failed once -> plain launches

  void ev_pair(hipEvent_t &a, hipEvent_t &b) {
    if (ev_used + 2 > ev_pool.size()) {
      hipEvent_t x, y;
      HIP_CHECK(hipEventCreate(&x));
      HIP_CHECK(hipEventCreate(&y));
      ev_pool.push_back(x);
      ev_pool.push_back(y);
    }
    a = ev_pool[ev_used];
    b = ev_pool[ev_used + 1];
    ev_used += 2;
  }
};

namespace {

u32 ceil_div(u64 a, u64 b) { return static_cast<u32>((a + b - 1) / b); }

// Blocking hipStreamSynchronize waits cost ~100+ us each on a busy queue
// (yield/interrupt based); the per-chunk control syncs are latency-critical,
// so spin on an event instead (~5 us).
void sync_spin(kmp_lp_t *e);

void engine_alloc_k_buffers(kmp_lp_t *e, u32 k_or_n) {
  // dep is double-buffered by the v2 commit's fixpoint (round parity);
  // legacy paths use the first k_or_n entries only
  const u64 dep_n = static_cast<u64>(k_or_n) * (k_or_n <= 256 ? 2 : 1);
  HIP_CHECK(hipMalloc(&e->d_seg_begin, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_seg_end, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_prefix_len, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_dep, sizeof(unsigned long long) * dep_n));
  HIP_CHECK(hipMemsetAsync(e->d_seg_begin, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_seg_end, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_prefix_len, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_dep, 0, sizeof(unsigned long long) * dep_n, e->stream));
}

void engine_destroy_sweep_graphs(kmp_lp_t *e) {
  for (int i = 0; i < kmp_lp_t::kMaxSweepGraphs; ++i) {
    if (e->sweep_graph[i]) {
      (void)hipGraphExecDestroy(e->sweep_graph[i]);
      e->sweep_graph[i] = nullptr;
    }
    e->sweep_seen[i] = 0;
  }
  e->graphs_ok = true;
}

void engine_free_k_buffers(kmp_lp_t *e) {
  if (e->d_seg_begin) {
    HIP_CHECK(hipFree(e->d_seg_begin));
    e->d_seg_begin = nullptr;
  }
  if (e->d_seg_end) {
    HIP_CHECK(hipFree(e->d_seg_end));
    e->d_seg_end = nullptr;
  }
  if (e->d_prefix_len) {
    HIP_CHECK(hipFree(e->d_prefix_len));
    e->d_prefix_len = nullptr;
  }
  if (e->d_dep) {
    HIP_CHECK(hipFree(e->d_dep));
    e->d_dep = nullptr;
  }
}

void sync_spin(kmp_lp_t *e) {
  HIP_CHECK(hipEventRecord(e->sync_ev, e->stream));
  while (true) {
    const hipError_t st = hipEventQuery(e->sync_ev);
    if (st == hipSuccess) {
      return;
    }
    if (st != hipErrorNotReady) {
      HIP_CHECK(st);
    }
  }
}

// Everything an engine allocates beyond the graph arrays themselves
// (LP state, phase/commit buffers, rocprim temps, pinned mirrors).
// Requires e->n, e->m, e->C, e->stream to be set.
void engine_alloc_common(kmp_lp_t *e) {
  HIP_CHECK(hipMalloc(&e->d_labels, sizeof(u32) * e->n));
  HIP_CHECK(hipMalloc(&e->d_labels0, sizeof(u32) * e->n));
  HIP_CHECK(hipMalloc(&e->d_labels16, sizeof(uint16_t) * e->n));
  HIP_CHECK(hipMalloc(&e->d_labels8, e->n));
  HIP_CHECK(hipMalloc(&e->d_active, e->n));
  HIP_CHECK(hipMalloc(&e->d_unit_active, kmp::num_units(e->n)));

  const u32 C = e->C;
  HIP_CHECK(hipMalloc(&e->d_slots, sizeof(Prop) * C));
  HIP_CHECK(hipMalloc(&e->d_props, sizeof(Prop) * C));
  HIP_CHECK(hipMalloc(&e->d_m_list, sizeof(u64) * C));
  HIP_CHECK(hipMalloc(&e->d_m_count, sizeof(u32) * 4)); // [0]=m, [1]=l, [2]=m2
  e->d_l_count = e->d_m_count + 1;
  e->d_m2_count = e->d_m_count + 2;
  HIP_CHECK(hipMalloc(&e->d_l_list, sizeof(u64) * C));
  HIP_CHECK(hipMalloc(&e->d_m2_list, sizeof(u64) * C));

  // commit v2 buffers (sized for the largest supported k = 256 rows layout)
  HIP_CHECK(hipMalloc(&e->d_s_u, sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_s_w, sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_s_to, sizeof(uint16_t) * C));
  HIP_CHECK(hipMalloc(&e->d_s_r, sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_s_b, sizeof(uint16_t) * C));
  HIP_CHECK(hipMalloc(&e->d_histT, sizeof(u32) * 262144));
  HIP_CHECK(hipMalloc(&e->d_offT, sizeof(u32) * 262144));
  HIP_CHECK(hipMalloc(&e->d_seg_off, sizeof(u32) * 257));
  HIP_CHECK(hipMalloc(&e->d_bar, sizeof(u32) * 16));
  HIP_CHECK(hipMemsetAsync(e->d_bar, 0, sizeof(u32) * 16, e->stream));
  HIP_CHECK(hipMalloc(&e->d_blocksums, sizeof(i64) * 512));
  HIP_CHECK(hipMalloc(&e->d_prop_count, sizeof(u32)));
  HIP_CHECK(hipMalloc(&e->d_arcs, sizeof(unsigned long long)));
  HIP_CHECK(hipMalloc(&e->d_moves, sizeof(unsigned long long)));

  HIP_CHECK(hipMalloc(&e->d_sort_keys[0], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_keys[1], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_vals[0], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_vals[1], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sw, sizeof(i64) * C));
  HIP_CHECK(hipMalloc(&e->d_pw, sizeof(i64) * C));
  HIP_CHECK(hipMalloc(&e->d_changed, sizeof(int) * 2)); // [round-parity] for coop
  HIP_CHECK(hipMalloc(&e->d_admitted_flags, sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_cut, sizeof(unsigned long long)));

  rocprim::double_buffer<u32> keys(e->d_sort_keys[0], e->d_sort_keys[1]);
  rocprim::double_buffer<u32> vals(e->d_sort_vals[0], e->d_sort_vals[1]);
  HIP_CHECK(rocprim::radix_sort_pairs(nullptr, e->sort_temp_bytes, keys, vals, C, 0, 32));
  HIP_CHECK(hipMalloc(&e->d_sort_temp, e->sort_temp_bytes));
  HIP_CHECK(rocprim::inclusive_scan_by_key(
      nullptr, e->scan_temp_bytes, e->d_sort_keys[0], e->d_sw, e->d_pw, C, rocprim::plus<i64>(),
      rocprim::equal_to<u32>()
  ));
  HIP_CHECK(hipMalloc(&e->d_scan_temp, e->scan_temp_bytes));
  HIP_CHECK(rocprim::select(
      nullptr, e->select_temp_bytes, e->d_slots, e->d_props, e->d_prop_count, C, PropValid()
  ));
  HIP_CHECK(hipMalloc(&e->d_select_temp, e->select_temp_bytes));

  HIP_CHECK(hipHostMalloc(&e->h_count, sizeof(u32) * 2));
  HIP_CHECK(hipHostMalloc(&e->h_changed, sizeof(int)));
  HIP_CHECK(hipHostMalloc(&e->h_moves, sizeof(unsigned long long) * 4));
  HIP_CHECK(hipHostMalloc(&e->h_hacc, sizeof(unsigned long long)));
}

// Isolated-vertex scan used by the clusterer's isolated-node handling
// (lp_clusterer.cc cluster_isolated_nodes semantics).
template <typename XT>
void engine_scan_isolated(kmp_lp_t *e, const XT *xadj, const i32 *vwgt) {
  u32 maxd = 0;
  for (u32 u = 0; u < e->n; ++u) {
    const u32 deg = static_cast<u32>(xadj[u + 1] - xadj[u]);
    if (deg > maxd) {
      maxd = deg;
    }
    if (deg == 0) {
      e->isolated.push_back(u);
      e->iso_weights.push_back(vwgt ? vwgt[u] : 1);
    }
  }
  e->max_deg = maxd;
}

hipEvent_t ev_one(kmp_lp_t *e) {
  if (e->ev_used >= e->ev_pool.size()) {
    hipEvent_t x;
    HIP_CHECK(hipEventCreate(&x));
    e->ev_pool.push_back(x);
  }
  return e->ev_pool[e->ev_used++];
}

// ---- commit v2 driver (refine, k <= 256; zero host syncs per chunk) ----

bool v2_eligible(const kmp_lp_t *e) {
  return !e->clusterer && !e->balance && e->k <= 256 && e->coop_nblk >= 8;
}

// Phase A without slot memsets: S/M/L freshly write every slot of an
// active unit; inactive units are gated downstream by unit_active.
void phase_a_v2(kmp_lp_t *e, int iter, u32 pos_lo, u32 pos_hi, u32 chunk_base) {
  const u64 iseed = iter_seed_of(e->seed, iter);
  const u32 span = pos_hi - pos_lo;
  const u32 threads = 256;
  HIP_CHECK(hipMemsetAsync(e->d_m_count, 0, sizeof(u32) * 3, e->stream));
  {
    const u32 rows_bl = 4096;
    const u32 T_bl = ((pos_hi + 63) >> 6) - (pos_lo >> 6);
    const u32 tpw = (T_bl + rows_bl - 1) / rows_bl;
    hipLaunchKernelGGL(
        k_build_lists, dim3(rows_bl / 4), dim3(threads), 0, e->stream, pos_lo, pos_hi, e->n,
        iseed, rows_bl, tpw, kSmallDeg, kMidDeg, kMidDeg, 0xFFFFFFFFu, e->d_xadj, e->d_active,
        e->d_unit_active, e->d_m_list, e->d_m_count, e->d_m2_list, e->d_m2_count, e->d_l_list,
        e->d_l_count, e->d_arcs
    );
    LAUNCH_CHECK();
  }
  hipLaunchKernelGGL(
      k_phase_s<uint8_t>, dim3(ceil_div(static_cast<u64>(ceil_div(span, 4)) * kWave, threads)),
      dim3(threads), 0, e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed, 0u, kInvalid,
      0xFFFFFFFFu, 1u, e->d_xadj, e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights,
      e->d_maxw, e->d_minw, e->d_labels8, e->d_active, e->d_unit_active, e->d_slots
  );
  LAUNCH_CHECK();
  {
    const size_t lds =
        static_cast<size_t>(threads / kWave) * e->k * gain_replicas(e->k) * sizeof(i32);
    auto *kern = e->has_adjwgt ? k_phase_m<false, uint8_t> : k_phase_m<true, uint8_t>;
    hipLaunchKernelGGL(
        kern, dim3(4096), dim3(threads), lds, e->stream, pos_lo, chunk_base, e->k, iseed, 0u,
        kInvalid, e->d_xadj, e->d_adjncy, e->d_adjwgt, e->d_vwgt, e->d_labels, e->d_labels8,
        e->d_weights, e->d_maxw, e->d_minw, e->d_m_list, e->d_m_count, e->d_slots
    );
    LAUNCH_CHECK();
  }
  if (e->max_deg > kMidDeg) {
    hipLaunchKernelGGL(
        k_l_prep_r, dim3(1), dim3(256), 0, e->stream, e->d_l_list, e->d_l_count, e->d_xadj,
        e->l_cap, e->d_l_off
    );
    LAUNCH_CHECK();
    const size_t hist_lds = static_cast<size_t>(e->k) * gain_replicas(e->k) * sizeof(i32);
    {
      auto *kern = e->has_adjwgt ? k_phase_l_acc<false, uint8_t> : k_phase_l_acc<true, uint8_t>;
      hipLaunchKernelGGL(
          kern, dim3(2048), dim3(256), hist_lds, e->stream, e->k, e->d_xadj, e->d_adjncy,
          e->d_adjwgt, e->d_labels8, e->d_l_list, e->d_l_count, e->l_cap, e->d_l_off, e->d_l_gains
      );
      LAUNCH_CHECK();
    }
    hipLaunchKernelGGL(
        k_phase_l_sel, dim3(2048), dim3(256), 0, e->stream, 0u, kInvalid, pos_lo, chunk_base,
        iseed, e->k, e->d_xadj, e->d_vwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_minw,
        e->d_l_list, e->d_l_count, e->l_cap, e->d_l_gains, e->d_slots
    );
    LAUNCH_CHECK();
    {
      const size_t lds =
          ((static_cast<size_t>(e->k) * gain_replicas(e->k) + 1) & ~1ull) * sizeof(i32) +
          16 * sizeof(i64);
      auto *kern =
          e->has_adjwgt ? k_phase_l_direct<false, uint8_t> : k_phase_l_direct<true, uint8_t>;
      hipLaunchKernelGGL(
          kern, dim3(512), dim3(256), lds, e->stream, 0u, kInvalid, pos_lo, chunk_base, iseed,
          e->k, e->d_xadj, e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_labels8,
          e->d_weights, e->d_maxw, e->d_minw, e->d_l_list, e->d_l_count, e->l_cap, e->d_slots
      );
      LAUNCH_CHECK();
    }
  }
}

void commit_v2(kmp_lp_t *e, int iter, u32 pos_lo, u32 pos_hi) {
  const u64 iseed = iter_seed_of(e->seed, iter);
  const u32 span = pos_hi - pos_lo;
  const u32 T = span >> 6;
  // clamp the histogram rows to the chunk's tile count (small chunks need
  // neither the full table nor the resident-grid scan)
  u32 rows = (T + 3) & ~3u;
  if (rows < 4) {
    rows = 4;
  }
  if (rows > e->rows_v2) {
    rows = e->rows_v2;
  }
  const u32 tpw = (T + rows - 1) / rows;
  const u32 threads = 256;
  const size_t lds_h = static_cast<size_t>(threads / kWave) * e->k * sizeof(u32);
  hipLaunchKernelGGL(
      k_hist_v2, dim3(rows / 4), dim3(threads), lds_h, e->stream, span, pos_lo, e->n, iseed,
      e->k, rows, tpw, e->d_slots, e->d_unit_active, e->d_histT
  );
  LAUNCH_CHECK();
  if (e->k * rows <= 8192) {
    hipLaunchKernelGGL(
        k_scan_small, dim3(1), dim3(threads), 0, e->stream, e->k, rows, e->d_histT, e->d_offT,
        e->d_seg_off, e->d_prefix_len, e->d_dep, e->d_changed
    );
  } else {
    hipLaunchKernelGGL(
        k_scan_coop, dim3(e->coop_nblk), dim3(threads), 0, e->stream, e->k, rows, e->coop_nblk,
        e->d_histT, e->d_offT, e->d_seg_off, e->d_prefix_len, e->d_dep, e->d_changed,
        e->d_blocksums, e->d_bar
    );
  }
  LAUNCH_CHECK();
  const size_t lds_sc = lds_h + static_cast<size_t>(2 * e->k) * sizeof(unsigned long long);
  hipLaunchKernelGGL(
      k_scatter_v2, dim3(rows / 4), dim3(threads), lds_sc, e->stream, span, pos_lo, e->n, iseed,
      e->k, rows, tpw, e->d_slots, e->d_unit_active, e->d_offT, e->d_labels16, e->d_s_u,
      e->d_s_w, e->d_s_to, e->d_s_b, e->d_dep
  );
  LAUNCH_CHECK();
  hipLaunchKernelGGL(
      k_fixpoint_v2, dim3(1), dim3(threads), 0, e->stream, e->k,
      static_cast<u32>(e->has_vwgt ? 1 : 0), e->d_seg_off, e->d_prefix_len, e->d_dep,
      e->d_s_w, e->d_s_b, e->d_weights, e->d_maxw, e->d_moves, e->d_pw
  );
  LAUNCH_CHECK();
  hipLaunchKernelGGL(
      k_apply_v2, dim3(2048), dim3(threads), 0, e->stream, e->k, pos_lo, pos_hi, e->n, iseed,
      e->d_seg_off, e->d_prefix_len, e->d_s_u, e->d_s_to, e->d_labels, e->d_labels16,
      e->d_labels8, e->d_unit_active
  );
  LAUNCH_CHECK();
  hipLaunchKernelGGL(
      k_activate_v2, dim3(2048), dim3(threads), 0, e->stream, e->k, e->d_seg_off,
      e->d_prefix_len, e->d_s_u, e->d_s_to, e->d_xadj, e->d_adjncy, e->d_active,
      e->d_unit_active
  );
  LAUNCH_CHECK();
}

// One sweep = 64 chunk trains enqueued back-to-back with no host sync; the
// single sync per sweep reads the device move counter (early-exit check)
// and settles the per-chunk timing events.
// Enqueue one full sweep's chunk trains (the capture body; zero host syncs
// inside). Events use a per-sweep slice of the pool so each captured graph
// owns distinct event objects across replays.
void enqueue_sweep_v2(kmp_lp_t *e, int iter, size_t *ev_base_io) {
  e->ev_used = *ev_base_io;
  for (u32 chunk = 0; chunk < kmp::kNumChunks; ++chunk) {
    const u32 pos_lo = chunk * e->C;
    const u32 pos_hi = pos_lo + e->C > e->P ? e->P : pos_lo + e->C;
    if (pos_lo >= pos_hi) {
      continue;
    }
    hipEvent_t a0 = ev_one(e), a1 = ev_one(e), c1 = ev_one(e);
    HIP_CHECK(hipEventRecord(a0, e->stream));
    phase_a_v2(e, iter, pos_lo, pos_hi, pos_lo);
    HIP_CHECK(hipEventRecord(a1, e->stream));
    commit_v2(e, iter, pos_lo, pos_hi);
    HIP_CHECK(hipEventRecord(c1, e->stream));
  }
  *ev_base_io = e->ev_used;
}

i64 run_sweeps_v2(kmp_lp_t *e, int iters) {
  HIP_CHECK(hipMemcpyAsync(&e->h_moves[0], e->d_moves, sizeof(unsigned long long),
                           hipMemcpyDeviceToHost, e->stream));
  sync_spin(e);
  unsigned long long last = e->h_moves[0];
  u64 total = 0;
  // per-sweep event slice size (3 events per non-empty chunk)
  u32 nonempty = 0;
  for (u32 chunk = 0; chunk < kmp::kNumChunks; ++chunk) {
    const u32 pos_lo = chunk * e->C;
    if (pos_lo < e->P) {
      ++nonempty;
    }
  }
  const size_t slice = static_cast<size_t>(nonempty) * 3;
  for (int iter = 0; iter < iters; ++iter) {
    size_t ev_base = (iter < kmp_lp_t::kMaxSweepGraphs ? iter : 0) * slice;
    const size_t ev_lo = ev_base;
    // capture only on the sweep's SECOND execution: one-shot callers (the
    // multilevel levels) skip the instantiate cost, repeated callers (the
    // bench steps) replay launch-overhead-free from step 2
    const bool can_graph = e->graphs_ok && iter < kmp_lp_t::kMaxSweepGraphs &&
                           e->sweep_seen[iter] != 0;
    if (iter < kmp_lp_t::kMaxSweepGraphs && e->sweep_seen[iter] == 0) {
      e->sweep_seen[iter] = 1;
    }
    if (can_graph && e->sweep_graph[iter] == nullptr) {
      // capture this sweep's train once; replays are launch-overhead-free
      // (the train is fixed-shape: counts are read on device, no host
      // syncs, and the per-chunk args depend only on (iter, chunk))
      hipError_t st = hipStreamBeginCapture(e->stream, hipStreamCaptureModeRelaxed);
      if (st == hipSuccess) {
        enqueue_sweep_v2(e, iter, &ev_base);
        hipGraph_t graph = nullptr;
        st = hipStreamEndCapture(e->stream, &graph);
        if (st == hipSuccess && graph != nullptr) {
          st = hipGraphInstantiate(&e->sweep_graph[iter], graph, nullptr, nullptr, 0);
          (void)hipGraphDestroy(graph);
        }
      }
      if (st != hipSuccess || e->sweep_graph[iter] == nullptr) {
        (void)hipGetLastError();
        e->graphs_ok = false; // fall back to plain launches for good
        e->sweep_graph[iter] = nullptr;
      }
    }
    if (can_graph && e->sweep_graph[iter] != nullptr) {
      HIP_CHECK(hipGraphLaunch(e->sweep_graph[iter], e->stream));
      ev_base = ev_lo + slice;
    } else {
      enqueue_sweep_v2(e, iter, &ev_base);
    }
    HIP_CHECK(hipMemcpyAsync(&e->h_moves[1], e->d_moves, sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, e->stream));
    sync_spin(e);
    for (size_t i = ev_lo; i + 2 < ev_base; i += 3) {
      float ms = 0;
      HIP_CHECK(hipEventElapsedTime(&ms, e->ev_pool[i], e->ev_pool[i + 1]));
      e->phase_a_ms += ms;
      HIP_CHECK(hipEventElapsedTime(&ms, e->ev_pool[i + 1], e->ev_pool[i + 2]));
      e->commit_ms += ms;
    }
    const unsigned long long cur = e->h_moves[1];
    const u64 sweep_moves = cur - last;
    last = cur;
    total += sweep_moves;
    if (sweep_moves == 0) {
      break;
    }
  }
  return static_cast<i64>(total);
}

} // namespace

extern "C" {

kmp_lp_t *kmp_lp_create(const kmp_graph_t *g) {
  int ndev = 0;
  if (hipGetDeviceCount(&ndev) != hipSuccess || ndev == 0) {
    fprintf(stderr, "kaminpar_amd: no HIP device available -- the LP engine requires a GPU\n");
    return nullptr;
  }

  auto *e = new kmp_lp_t();
  (void)hipSetDeviceFlags(hipDeviceScheduleSpin); // ignore if context exists
  e->n = kmp_graph_n(g);
  e->m = kmp_graph_m(g);
  e->C = kmp::chunk_size_for(e->n);
  e->P = kmp::pos_count(e->n);
  e->has_vwgt = kmp_graph_vwgt(g) != nullptr;
  e->has_adjwgt = kmp_graph_adjwgt(g) != nullptr;
  HIP_CHECK(hipStreamCreate(&e->stream));
  HIP_CHECK(hipEventCreateWithFlags(&e->sync_ev, hipEventDisableTiming));
  {
    int dev = 0;
    HIP_CHECK(hipGetDevice(&dev));
    hipDeviceProp_t props;
    HIP_CHECK(hipGetDeviceProperties(&props, dev));
    e->mp_count = props.multiProcessorCount;
  }

  HIP_CHECK(hipMalloc(&e->d_xadj, sizeof(u64) * (e->n + 1)));
  HIP_CHECK(hipMalloc(&e->d_adjncy, sizeof(u32) * e->m));
  // device xadj is u64 (EdgeID-64); widen u32 host offsets on upload
  const u64 *xadj64 = kmp_graph_xadj64(g);
  if (xadj64 != nullptr) {
    HIP_CHECK(
        hipMemcpy(e->d_xadj, xadj64, sizeof(u64) * (e->n + 1), hipMemcpyHostToDevice)
    );
  } else {
    const u32 *x32 = kmp_graph_xadj(g);
    std::vector<u64> wide(static_cast<size_t>(e->n) + 1);
    for (u32 i = 0; i <= e->n; ++i) {
      wide[i] = x32[i];
    }
    HIP_CHECK(
        hipMemcpy(e->d_xadj, wide.data(), sizeof(u64) * (e->n + 1), hipMemcpyHostToDevice)
    );
  }
  HIP_CHECK(hipMemcpy(e->d_adjncy, kmp_graph_adjncy(g), sizeof(u32) * e->m, hipMemcpyHostToDevice));
  if (e->has_vwgt) {
    HIP_CHECK(hipMalloc(&e->d_vwgt, sizeof(i32) * e->n));
    HIP_CHECK(hipMemcpy(e->d_vwgt, kmp_graph_vwgt(g), sizeof(i32) * e->n, hipMemcpyHostToDevice));
  }
  if (e->has_adjwgt) {
    HIP_CHECK(hipMalloc(&e->d_adjwgt, sizeof(i32) * e->m));
    HIP_CHECK(
        hipMemcpy(e->d_adjwgt, kmp_graph_adjwgt(g), sizeof(i32) * e->m, hipMemcpyHostToDevice)
    );
  }

  engine_alloc_common(e);
  if (kmp_graph_xadj64(g) != nullptr) {
    engine_scan_isolated(e, kmp_graph_xadj64(g), kmp_graph_vwgt(g));
  } else {
    engine_scan_isolated(e, kmp_graph_xadj(g), kmp_graph_vwgt(g));
  }
  return e;
}

void kmp_lp_free(kmp_lp_t *e) {
  if (!e) {
    return;
  }
  HIP_CHECK(hipDeviceSynchronize());
  engine_destroy_sweep_graphs(e);
  for (hipEvent_t ev : e->ev_pool) {
    (void)hipEventDestroy(ev);
  }
  if (e->sync_ev) {
    (void)hipEventDestroy(e->sync_ev);
  }
  engine_free_k_buffers(e);
  for (void *p : {(void *)e->d_xadj, (void *)e->d_adjncy, (void *)e->d_vwgt, (void *)e->d_adjwgt,
                  (void *)e->d_labels, (void *)e->d_labels0, (void *)e->d_labels16, (void *)e->d_labels8, (void *)e->d_weights, (void *)e->d_maxw, (void *)e->d_minw, (void *)e->d_comm, (void *)e->d_active, (void *)e->d_unit_active,
                  (void *)e->d_slots, (void *)e->d_props,
                  (void *)e->d_m_list, (void *)e->d_m_count, (void *)e->d_m2_list,
                  (void *)e->d_l_list, (void *)e->d_l_off, (void *)e->d_l_sizes, (void *)e->d_lscan_temp, (void *)e->d_l_gains, (void *)e->d_prop_count, (void *)e->d_arcs,
                  (void *)e->d_moves, (void *)e->d_sort_keys[0], (void *)e->d_sort_keys[1],
                  (void *)e->d_sort_vals[0], (void *)e->d_sort_vals[1], (void *)e->d_sort_temp,
                  (void *)e->d_select_temp, (void *)e->d_sw, (void *)e->d_pw,
                  (void *)e->d_scan_temp, (void *)e->d_changed, (void *)e->d_admitted_flags,
                  (void *)e->d_cut, (void *)e->d_s_u, (void *)e->d_s_w, (void *)e->d_s_r, (void *)e->d_s_to,
                  (void *)e->d_s_b, (void *)e->d_histT, (void *)e->d_offT, (void *)e->d_seg_off,
                  (void *)e->d_bar, (void *)e->d_blocksums}) {
    if (p) {
      (void)hipFree(p);
    }
  }
  for (void *p : {(void *)e->d_favored, (void *)e->d_eflag, (void *)e->d_pool_keys,
                  (void *)e->d_pool_vals, (void *)e->d_l_clist, (void *)e->d_l_ccnt,
                  (void *)e->d_l_hoff, (void *)e->d_l_hbits,
                  (void *)e->d_l_hacc, (void *)e->d_cand, (void *)e->d_cand2,
                  (void *)e->d_cfav, (void *)e->d_crank, (void *)e->d_cones,
                  (void *)e->d_cand_select_temp, (void *)e->d_cand_sort_temp,
                  (void *)e->d_cand_scan_temp, (void *)e->d_emptied}) {
    if (p) {
      (void)hipFree(p);
    }
  }
  if (e->h_count) {
    (void)hipHostFree(e->h_count);
  }
  if (e->h_changed) {
    (void)hipHostFree(e->h_changed);
  }
  if (e->h_moves) {
    (void)hipHostFree(e->h_moves);
  }
  if (e->h_hacc) {
    (void)hipHostFree(e->h_hacc);
  }
  (void)hipStreamDestroy(e->own_stream != nullptr ? e->own_stream : e->stream);
  delete e;
}

u32 kmp_lp_num_chunks(const kmp_lp_t *) { return kmp::kNumChunks; }

// Adopt an external HIP stream (e.g. torch's current stream) so the
// multi-GPU driver's collectives and the engine's kernels share ONE stream
// and the per-call cross-stream hipStreamSynchronize bridges disappear.
// Pass 0 to restore the engine's own stream. The engine must be idle.
int kmp_lp_set_stream(kmp_lp_t *e, void *external_stream) {
  HIP_CHECK(hipStreamSynchronize(e->stream));
  engine_destroy_sweep_graphs(e); // graphs are bound to the old stream
  if (external_stream != nullptr) {
    if (e->own_stream == nullptr) {
      e->own_stream = e->stream;
    }
    e->stream = static_cast<hipStream_t>(external_stream);
  } else if (e->own_stream != nullptr) {
    e->stream = e->own_stream;
    e->own_stream = nullptr;
  }
  return 0;
}

int kmp_lp_refine_begin(
    kmp_lp_t *e, u32 k, const i64 *max_block_weights, const u32 *partition, u64 seed
) {
  if (k > kMaxDenseK) {
    fprintf(stderr, "kaminpar_amd: refine currently supports k <= %u (got %u)\n", kMaxDenseK, k);
    return -1;
  }
  e->k = k;
  e->seed = seed;
  e->clusterer = false;
  e->maxw_host.assign(max_block_weights, max_block_weights + k);
  e->phase_a_ms = 0.0;
  e->commit_ms = 0.0;
  e->ev_used = 0;

  engine_destroy_sweep_graphs(e);
  engine_free_k_buffers(e);
  engine_alloc_k_buffers(e, k);

  if (e->d_weights) {
    HIP_CHECK(hipFree(e->d_weights));
  }
  if (e->d_maxw) {
    HIP_CHECK(hipFree(e->d_maxw));
  }
  HIP_CHECK(hipMalloc(&e->d_weights, sizeof(i64) * k));
  HIP_CHECK(hipMalloc(&e->d_maxw, sizeof(i64) * k));
  if (e->d_l_off) {
    HIP_CHECK(hipFree(e->d_l_off));
  }
  if (e->d_l_gains) {
    HIP_CHECK(hipFree(e->d_l_gains));
  }
  e->l_cap = (1u << 22) / k;
  if (e->l_cap > e->C) {
    e->l_cap = e->C;
  }
  if (e->l_cap < 1024) {
    e->l_cap = 1024;
  }
  HIP_CHECK(hipMalloc(&e->d_l_off, sizeof(u32) * (e->l_cap + 1)));
  if (e->d_l_sizes) {
    HIP_CHECK(hipFree(e->d_l_sizes));
  }
  if (e->d_lscan_temp) {
    HIP_CHECK(hipFree(e->d_lscan_temp));
  }
  HIP_CHECK(hipMalloc(&e->d_l_sizes, sizeof(u32) * (e->l_cap + 1)));
  HIP_CHECK(rocprim::exclusive_scan(
      nullptr, e->lscan_temp_bytes, e->d_l_sizes, e->d_l_off, 0u, e->l_cap + 1
  ));
  HIP_CHECK(hipMalloc(&e->d_lscan_temp, e->lscan_temp_bytes));
  HIP_CHECK(hipMalloc(&e->d_l_gains, sizeof(i32) * e->l_cap * k));
  HIP_CHECK(hipMemsetAsync(e->d_l_gains, 0, sizeof(i32) * e->l_cap * k, e->stream));
  HIP_CHECK(hipMemcpy(e->d_maxw, max_block_weights, sizeof(i64) * k, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(e->d_labels, partition, sizeof(u32) * e->n, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpyAsync(
      e->d_labels0, e->d_labels, sizeof(u32) * e->n, hipMemcpyDeviceToDevice, e->stream
  ));
  HIP_CHECK(hipMemsetAsync(e->d_active, 1, e->n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_unit_active, 1, kmp::num_units(e->n), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_arcs, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_moves, 0, sizeof(unsigned long long), e->stream));

  HIP_CHECK(hipMemsetAsync(e->d_weights, 0, sizeof(i64) * k, e->stream));
  {
    const u32 threads = 256;
    hipLaunchKernelGGL(
        k_sync_labels16, dim3(ceil_div(e->n, threads)), dim3(threads), 0, e->stream, e->n,
        e->d_labels, e->d_labels16, e->d_labels8
    );
    LAUNCH_CHECK();
    const size_t lds = static_cast<size_t>(k) * sizeof(unsigned long long);
    hipLaunchKernelGGL(
        k_init_weights, dim3(2048), dim3(threads), lds, e->stream, e->n, k, e->d_labels, e->d_vwgt,
        reinterpret_cast<unsigned long long *>(e->d_weights)
    );
    LAUNCH_CHECK();
  }
  // commit v2 setup: histogram row count + resident-guaranteed grid for the
  // single-launch commit kernel (falls back to the legacy commit if the
  // occupancy query says the grid cannot be made co-resident). The grid is
  // deliberately SMALL (64 blocks): the commit stages are latency-bound and
  // the grid barrier cost grows with the block count (tools/bar_bench.hip).
  {
    // histogram rows: enough waves to hide tile-load latency even at
    // k = 256 (a 256-row table serialized 256 tiles per wave at scale 28 --
    // measured regression); table <= 262144 entries = 1 MB
    u32 rows = (262144u / k) & ~3u;
    if (rows < 256) {
      rows = 256;
    }
    if (rows > 4096) {
      rows = 4096;
    }
    e->rows_v2 = rows;
  }
  e->coop_nblk = 0;
  if (k <= 256) {
    int occ_scan = 0;
    if (hipOccupancyMaxActiveBlocksPerMultiprocessor(&occ_scan, k_scan_coop, 256, 0) ==
            hipSuccess &&
        occ_scan > 0 && e->mp_count > 0) {
      u64 nb = static_cast<u64>(occ_scan) * static_cast<u64>(e->mp_count);
      if (nb > 64) {
        nb = 64;
      }
      nb &= ~7ull;
      e->coop_nblk = static_cast<u32>(nb);
    }
  }
  HIP_CHECK(hipStreamSynchronize(e->stream));
  return 0;
}

i64 kmp_lp_phase_a(
    kmp_lp_t *e, int iter, u32 chunk, u32 pos_lo, u32 pos_hi, void *d_out, u32 cap
) {
  const u64 iseed = iter_seed_of(e->seed, iter);
  const u32 chunk_base = chunk * e->C;
  Prop *out = static_cast<Prop *>(d_out);
  const u32 span = pos_hi - pos_lo;
  if (span == 0) {
    return 0; // empty rank slice
  }
  const u32 threads = 256;
  const u32 max_degree = 0xFFFFFFFFu;

  HIP_CHECK(hipMemsetAsync(e->d_m_count, 0, sizeof(u32) * 3, e->stream)); // m+l counts
  // pre-mark every slot invalid (the legacy/sharded/clusterer commit reads
  // compacted proposals; the clusterer S kernel does not always-write)
  HIP_CHECK(hipMemsetAsync(e->d_slots, 0xFF, sizeof(Prop) * span, e->stream));

  hipEvent_t ev0, ev1;
  e->ev_pair(ev0, ev1);
  HIP_CHECK(hipEventRecord(ev0, e->stream));

  if (!e->clusterer) {
    // balance mode: per-chunk fallback target = lightest block with room
    // (for overloaded vertices with no admissible adjacent candidate)
    u32 fallback = kInvalid;
    if (e->balance == 1) {
      std::vector<i64> w(e->k);
      HIP_CHECK(hipMemcpyAsync(w.data(), e->d_weights, sizeof(i64) * e->k,
                               hipMemcpyDeviceToHost, e->stream));
      sync_spin(e);
      i64 bw = -1;
      for (u32 c = 0; c < e->k; ++c) {
        if (w[c] < e->maxw_host[c] && (bw < 0 || w[c] < bw)) {
          bw = w[c];
          fallback = c;
        }
      }
    }
    {
      const u32 rows_bl = 4096;
      const u32 T_bl = ((pos_hi + 63) >> 6) - (pos_lo >> 6);
      const u32 tpw = (T_bl + rows_bl - 1) / rows_bl;
      hipLaunchKernelGGL(
          k_build_lists, dim3(rows_bl / 4), dim3(threads), 0, e->stream, pos_lo, pos_hi, e->n,
          iseed, rows_bl, tpw, kSmallDeg, kMidDeg, kMidDeg, max_degree, e->d_xadj, e->d_active,
          e->d_unit_active, e->d_m_list, e->d_m_count, e->d_m2_list, e->d_m2_count,
          e->d_l_list, e->d_l_count, static_cast<unsigned long long *>(nullptr)
      );
      LAUNCH_CHECK();
    }
    const bool k8 = e->k <= 256;
    // S: 4 positions/wave (unit-gated; slots pre-marked invalid)
    if (k8) {
      hipLaunchKernelGGL(
          k_phase_s<uint8_t>, dim3(ceil_div(static_cast<u64>(ceil_div(span, 4)) * kWave, threads)),
          dim3(threads), 0, e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed,
          static_cast<u32>(e->balance), fallback, max_degree, 0u,
          e->d_xadj, e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights, e->d_maxw,
          e->d_minw, e->d_labels8, e->d_active, e->d_unit_active, e->d_slots
      );
    } else {
      hipLaunchKernelGGL(
          k_phase_s<uint16_t>, dim3(ceil_div(static_cast<u64>(ceil_div(span, 4)) * kWave, threads)),
          dim3(threads), 0, e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed,
          static_cast<u32>(e->balance), fallback, max_degree, 0u,
          e->d_xadj, e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights, e->d_maxw,
          e->d_minw, e->d_labels16, e->d_active, e->d_unit_active, e->d_slots
      );
    }
    LAUNCH_CHECK();
    // one wave per M-listed vertex (grid-stride; list built by k_build_lists)
    {
      const size_t lds =
          static_cast<size_t>(threads / kWave) * e->k * gain_replicas(e->k) * sizeof(i32);
      if (k8) {
        auto *kern = e->has_adjwgt ? k_phase_m<false, uint8_t> : k_phase_m<true, uint8_t>;
        hipLaunchKernelGGL(
            kern, dim3(4096), dim3(threads), lds, e->stream, pos_lo, chunk_base, e->k, iseed,
            static_cast<u32>(e->balance), fallback,
            e->d_xadj, e->d_adjncy, e->d_adjwgt, e->d_vwgt, e->d_labels, e->d_labels8,
            e->d_weights, e->d_maxw, e->d_minw, e->d_m_list, e->d_m_count, e->d_slots
        );
      } else {
        auto *kern = e->has_adjwgt ? k_phase_m<false, uint16_t> : k_phase_m<true, uint16_t>;
        hipLaunchKernelGGL(
            kern, dim3(4096), dim3(threads), lds, e->stream, pos_lo, chunk_base, e->k, iseed,
            static_cast<u32>(e->balance), fallback,
            e->d_xadj, e->d_adjncy, e->d_adjwgt, e->d_vwgt, e->d_labels, e->d_labels16,
            e->d_weights, e->d_maxw, e->d_minw, e->d_m_list, e->d_m_count, e->d_slots
        );
      }
      LAUNCH_CHECK();
    }
    // L: slice-parallel accumulation over the (rare) high-degree list
    {
      HIP_CHECK(hipMemsetAsync(e->d_l_sizes, 0, sizeof(u32) * (e->l_cap + 1), e->stream));
      hipLaunchKernelGGL(
          k_l_sizes, dim3(ceil_div(e->l_cap, 256)), dim3(256), 0, e->stream, e->d_l_list,
          e->d_l_count, e->d_xadj, e->l_cap, e->d_l_sizes
      );
      LAUNCH_CHECK();
      size_t ltb = e->lscan_temp_bytes;
      HIP_CHECK(rocprim::exclusive_scan(
          e->d_lscan_temp, ltb, e->d_l_sizes, e->d_l_off, 0u, e->l_cap + 1,
          rocprim::plus<u32>(), e->stream
      ));
      const size_t hist_lds = static_cast<size_t>(e->k) * gain_replicas(e->k) * sizeof(i32);
      if (k8) {
        auto *kern = e->has_adjwgt ? k_phase_l_acc<false, uint8_t> : k_phase_l_acc<true, uint8_t>;
        hipLaunchKernelGGL(
            kern, dim3(2048), dim3(256), hist_lds, e->stream, e->k, e->d_xadj, e->d_adjncy,
            e->d_adjwgt, e->d_labels8, e->d_l_list, e->d_l_count, e->l_cap, e->d_l_off,
            e->d_l_gains
        );
        LAUNCH_CHECK();
      } else {
        auto *kern =
            e->has_adjwgt ? k_phase_l_acc<false, uint16_t> : k_phase_l_acc<true, uint16_t>;
        hipLaunchKernelGGL(
            kern, dim3(2048), dim3(256), hist_lds, e->stream, e->k, e->d_xadj, e->d_adjncy,
            e->d_adjwgt, e->d_labels16, e->d_l_list, e->d_l_count, e->l_cap, e->d_l_off,
            e->d_l_gains
        );
        LAUNCH_CHECK();
      }
      hipLaunchKernelGGL(
          k_phase_l_sel, dim3(2048), dim3(256), 0, e->stream,
          static_cast<u32>(e->balance), fallback, pos_lo, chunk_base, iseed, e->k,
          e->d_xadj, e->d_vwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_minw, e->d_l_list,
          e->d_l_count, e->l_cap, e->d_l_gains, e->d_slots
      );
      LAUNCH_CHECK();
      // pathological overflow beyond l_cap: direct per-vertex workgroups
      {
        const size_t lds =
            ((static_cast<size_t>(e->k) * gain_replicas(e->k) + 1) & ~1ull) * sizeof(i32) +
            16 * sizeof(i64);
        if (k8) {
          auto *kern =
              e->has_adjwgt ? k_phase_l_direct<false, uint8_t> : k_phase_l_direct<true, uint8_t>;
          hipLaunchKernelGGL(
              kern, dim3(512), dim3(256), lds, e->stream,
              static_cast<u32>(e->balance), fallback, pos_lo, chunk_base, iseed, e->k, e->d_xadj,
              e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_labels8, e->d_weights,
              e->d_maxw, e->d_minw, e->d_l_list, e->d_l_count, e->l_cap, e->d_slots
          );
        } else {
          auto *kern =
              e->has_adjwgt ? k_phase_l_direct<false, uint16_t> : k_phase_l_direct<true, uint16_t>;
          hipLaunchKernelGGL(
              kern, dim3(512), dim3(256), lds, e->stream,
              static_cast<u32>(e->balance), fallback, pos_lo, chunk_base, iseed, e->k, e->d_xadj,
              e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_labels16, e->d_weights,
              e->d_maxw, e->d_minw, e->d_l_list, e->d_l_count, e->l_cap, e->d_slots
          );
        }
        LAUNCH_CHECK();
      }
    }
  } else {
    // clustering: hash-based gain maps, favored-cluster tracking
    {
      const u32 rows_bl = 4096;
      const u32 T_bl = ((pos_hi + 63) >> 6) - (pos_lo >> 6);
      const u32 tpw = (T_bl + rows_bl - 1) / rows_bl;
      hipLaunchKernelGGL(
          k_build_lists, dim3(rows_bl / 4), dim3(threads), 0, e->stream, pos_lo, pos_hi, e->n,
          iseed, rows_bl, tpw, kSmallDeg, kClusterMidDeg, kClusterM2Deg, max_degree, e->d_xadj,
          e->d_active, e->d_unit_active, e->d_m_list, e->d_m_count, e->d_m2_list,
          e->d_m2_count, e->d_l_list, e->d_l_count,
          static_cast<unsigned long long *>(nullptr)
      );
      LAUNCH_CHECK();
    }
    hipLaunchKernelGGL(
        k_phase_s_c, dim3(ceil_div(static_cast<u64>(ceil_div(span, 4)) * kWave, threads)),
        dim3(threads), 0, e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed, max_degree,
        e->maxw_uniform, e->d_xadj, e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights,
        e->d_comm, e->d_active, e->d_unit_active, e->d_favored, e->d_slots
    );
    LAUNCH_CHECK();
    {
      const size_t lds = static_cast<size_t>(threads / kWave) * 2 * kHashSlots * sizeof(u32);
      auto *kern = e->has_adjwgt ? k_phase_m_c<false> : k_phase_m_c<true>;
      hipLaunchKernelGGL(
          kern, dim3(4096), dim3(threads), lds, e->stream, pos_lo, chunk_base, iseed,
          e->maxw_uniform, e->d_xadj, e->d_adjncy, e->d_adjwgt, e->d_vwgt, e->d_labels,
          e->d_weights, e->d_comm, e->d_m_list, e->d_m_count, e->d_favored, e->d_slots
      );
      LAUNCH_CHECK();
      auto *kern2 = e->has_adjwgt ? k_phase_m2_c<false> : k_phase_m2_c<true>;
      hipLaunchKernelGGL(
          kern2, dim3(1024), dim3(threads), 0, e->stream, pos_lo, chunk_base, iseed,
          e->maxw_uniform, e->d_xadj, e->d_adjncy, e->d_adjwgt, e->d_vwgt, e->d_labels,
          e->d_weights, e->d_comm, e->d_m2_list, e->d_m2_count, e->d_favored, e->d_slots
      );
      LAUNCH_CHECK();
    }
    {
      hipLaunchKernelGGL(
          k_l_prep_c, dim3(1), dim3(1024), 0, e->stream, e->d_l_list, e->d_l_count, e->d_xadj,
          e->l_cap, e->d_l_off, e->d_l_hoff, e->d_l_hbits, e->d_l_ccnt, e->d_l_hacc
      );
      LAUNCH_CHECK();
      // read back the L count and the chunk's total hash-region demand:
      // when it exceeds the pool (hub-dense chunks, e.g. after degree-bucket
      // rearrangement packs 64 hubs per permutation unit), process the L
      // list in pool-sized batches -- k_phase_l_sel_c clears each region
      // after reading it, so batches can reuse the pool safely.
      HIP_CHECK(hipMemcpyAsync(
          e->h_count + 1, e->d_l_count, sizeof(u32), hipMemcpyDeviceToHost, e->stream
      ));
      HIP_CHECK(hipMemcpyAsync(
          e->h_hacc, e->d_l_hacc, sizeof(unsigned long long), hipMemcpyDeviceToHost, e->stream
      ));
      sync_spin(e);
      const u32 lcount = e->h_count[1] < e->l_cap ? e->h_count[1] : e->l_cap;
      const u64 hacc = *e->h_hacc;
      auto *kern = e->has_adjwgt ? k_phase_l_acc_c<false> : k_phase_l_acc_c<true>;
      auto launch_batch = [&](u32 lo, u32 hi) {
        hipLaunchKernelGGL(
            kern, dim3(2048), dim3(256), 0, e->stream, e->d_xadj, e->d_adjncy, e->d_adjwgt,
            e->d_labels, e->d_l_list, lo, hi, e->d_l_off, e->d_l_hoff, e->d_l_hbits,
            e->d_pool_keys, e->d_pool_vals, e->d_l_clist, e->d_l_ccnt
        );
        LAUNCH_CHECK();
        hipLaunchKernelGGL(
            k_phase_l_sel_c, dim3(2048), dim3(256), 0, e->stream, pos_lo, chunk_base, iseed,
            e->n, e->maxw_uniform, e->d_vwgt, e->d_labels, e->d_weights, e->d_l_list, lo, hi,
            e->d_l_hoff, e->d_l_hbits, e->d_pool_keys, e->d_pool_vals, e->d_l_clist,
            e->d_l_ccnt, e->d_comm, e->d_favored, e->d_slots
        );
        LAUNCH_CHECK();
      };
      if (lcount > 0 && hacc <= e->pool_slots) {
        launch_batch(0, lcount);
      } else if (lcount > 0) {
        std::vector<u64> hoff(lcount + 1);
        HIP_CHECK(hipMemcpyAsync(
            hoff.data(), e->d_l_hoff, sizeof(u64) * (lcount + 1), hipMemcpyDeviceToHost,
            e->stream
        ));
        sync_spin(e);
        u32 lo = 0;
        while (lo < lcount) {
          const u64 need_one = hoff[lo + 1] - hoff[lo];
          if (need_one > e->pool_slots) {
            // a single region larger than the pool: grow it
            HIP_CHECK(hipFree(e->d_pool_keys));
            HIP_CHECK(hipFree(e->d_pool_vals));
            HIP_CHECK(hipFree(e->d_l_clist));
            e->pool_slots = need_one;
            HIP_CHECK(hipMalloc(&e->d_pool_keys, sizeof(u32) * e->pool_slots));
            HIP_CHECK(hipMalloc(&e->d_pool_vals, sizeof(i32) * e->pool_slots));
            HIP_CHECK(hipMalloc(&e->d_l_clist, sizeof(u32) * (e->pool_slots / 2)));
            HIP_CHECK(hipMemsetAsync(e->d_pool_keys, 0xFF, sizeof(u32) * e->pool_slots,
                                     e->stream));
            HIP_CHECK(hipMemsetAsync(e->d_pool_vals, 0, sizeof(i32) * e->pool_slots,
                                     e->stream));
          }
          u32 hi = lo + 1;
          while (hi < lcount && hoff[hi + 1] - hoff[lo] <= e->pool_slots) {
            ++hi;
          }
          launch_batch(lo, hi);
          lo = hi;
        }
      }
    }
  }
  // compact valid slots in position order (stable select)
  {
    size_t tb = e->select_temp_bytes;
    HIP_CHECK(rocprim::select(
        e->d_select_temp, tb, e->d_slots, out, e->d_prop_count, span, PropValid(), e->stream
    ));
  }
  HIP_CHECK(hipEventRecord(ev1, e->stream));

  HIP_CHECK(
      hipMemcpyAsync(e->h_count, e->d_prop_count, sizeof(u32), hipMemcpyDeviceToHost, e->stream)
  );
  sync_spin(e);
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
  e->phase_a_ms += ms;

  if (*e->h_count > cap) {
    fprintf(stderr, "kaminpar_amd: proposal buffer overflow (%u > %u)\n", *e->h_count, cap);
    return -1;
  }
  return static_cast<i64>(*e->h_count);
}

i64 kmp_lp_commit(kmp_lp_t *e, int iter, u32 chunk, const void *d_props, u32 count) {
  const u64 iseed = iter_seed_of(e->seed, iter);
  const u32 threads = 256;
  const u32 chunk_lo = chunk * e->C;
  const u32 chunk_hi = chunk_lo + e->C > e->P ? e->P : chunk_lo + e->C;
  const bool big_k = e->clusterer; // cluster space = n

  hipEvent_t cev0, cev1;
  e->ev_pair(cev0, cev1);
  HIP_CHECK(hipEventRecord(cev0, e->stream));
  HIP_CHECK(hipMemcpyAsync(&e->h_moves[0], e->d_moves, sizeof(unsigned long long),
                           hipMemcpyDeviceToHost, e->stream));
  if (big_k) {
    HIP_CHECK(hipMemcpyAsync(&e->h_moves[2], e->d_emptied, sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, e->stream));
  }

  const Prop *props = static_cast<const Prop *>(d_props);
  u32 *order = nullptr;
  u32 *sto = nullptr;

  if (count > 0) {
    const u32 grid = ceil_div(count, threads);
    hipLaunchKernelGGL(
        k_make_keys, dim3(grid), dim3(threads), 0, e->stream, props, count, e->d_sort_keys[0],
        e->d_sort_vals[0]
    );
    LAUNCH_CHECK();
    rocprim::double_buffer<u32> keys(e->d_sort_keys[0], e->d_sort_keys[1]);
    rocprim::double_buffer<u32> vals(e->d_sort_vals[0], e->d_sort_vals[1]);
    size_t tb = e->sort_temp_bytes;
    HIP_CHECK(rocprim::radix_sort_pairs(e->d_sort_temp, tb, keys, vals, count, 0, 32, e->stream));
    order = vals.current();
    sto = keys.current();

    hipLaunchKernelGGL(
        k_extract_w, dim3(grid), dim3(threads), 0, e->stream, order, props, count, e->d_sw
    );
    LAUNCH_CHECK();
    size_t sb = e->scan_temp_bytes;
    HIP_CHECK(rocprim::inclusive_scan_by_key(
        e->d_scan_temp, sb, sto, e->d_sw, e->d_pw, count, rocprim::plus<i64>(),
        rocprim::equal_to<u32>(), e->stream
    ));
    hipLaunchKernelGGL(
        k_seg_bounds, dim3(grid), dim3(threads), 0, e->stream, sto, count, e->d_seg_begin,
        e->d_seg_end
    );
    LAUNCH_CHECK();
    hipLaunchKernelGGL(
        k_seg_len_heads, dim3(grid), dim3(threads), 0, e->stream, sto, count, e->d_seg_begin,
        e->d_seg_end, e->d_prefix_len
    );
    LAUNCH_CHECK();

    // greatest-fixpoint rollback (kaminpar-dist lp_refiner.cc:296-333).
    // Two rounds per host sync; the second round's changed flag decides.
    const u32 kgrid = big_k ? 0 : ceil_div(e->k, threads);
    while (true) {
      for (int half = 0; half < 2; ++half) {
        if (big_k) {
          hipLaunchKernelGGL(
              k_dep_reset_touched, dim3(grid), dim3(threads), 0, e->stream, props, count,
              e->d_labels, e->d_dep
          );
          LAUNCH_CHECK();
          hipLaunchKernelGGL(
              k_dep_direct, dim3(grid), dim3(threads), 0, e->stream, order, props, sto, count,
              e->d_seg_begin, e->d_prefix_len, e->d_labels, e->d_dep
          );
          LAUNCH_CHECK();
          HIP_CHECK(hipMemsetAsync(e->d_changed, 0, sizeof(int), e->stream));
          hipLaunchKernelGGL(
              k_cutoff_heads, dim3(grid), dim3(threads), 0, e->stream, sto, count, e->d_seg_begin,
              e->d_seg_end, e->d_prefix_len, e->d_pw, e->d_weights, e->maxw_uniform, e->d_dep,
              e->d_changed
          );
          LAUNCH_CHECK();
        } else {
          hipLaunchKernelGGL(
              k_dep_reset_all, dim3(kgrid), dim3(threads), 0, e->stream, e->k, e->d_dep
          );
          LAUNCH_CHECK();
          hipLaunchKernelGGL(
              k_dep, dim3(grid > 2048 ? 2048 : grid), dim3(threads),
              static_cast<size_t>(e->k) * sizeof(unsigned long long), e->stream, order, props, sto,
              count, e->k, e->d_seg_begin, e->d_prefix_len, e->d_labels, e->d_dep
          );
          LAUNCH_CHECK();
          HIP_CHECK(hipMemsetAsync(e->d_changed, 0, sizeof(int), e->stream));
          hipLaunchKernelGGL(
              k_cutoff, dim3(kgrid), dim3(threads), 0, e->stream, e->k, e->d_seg_begin,
              e->d_seg_end, e->d_prefix_len, e->d_pw, e->d_weights, e->d_maxw, e->d_dep,
              e->d_changed
          );
          LAUNCH_CHECK();
        }
      }
      HIP_CHECK(
          hipMemcpyAsync(e->h_changed, e->d_changed, sizeof(int), hipMemcpyDeviceToHost, e->stream)
      );
      sync_spin(e);
      if (!*e->h_changed) {
        break;
      }
    }

    if (big_k) {
      hipLaunchKernelGGL(
          k_weights_update_big, dim3(grid), dim3(threads), 0, e->stream, order, props, sto, count,
          e->d_seg_begin, e->d_prefix_len, e->d_pw, e->d_labels, e->d_dep, e->d_weights
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_emptied_count, dim3(grid), dim3(threads), 0, e->stream, order, props, sto, count,
          e->d_seg_begin, e->d_prefix_len, e->d_labels, e->d_weights, e->d_eflag, e->d_emptied
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_emptied_reset, dim3(grid), dim3(threads), 0, e->stream, order, props, count,
          e->d_labels, e->d_eflag
      );
      LAUNCH_CHECK();
    } else {
      hipLaunchKernelGGL(
          k_weights_update, dim3(ceil_div(e->k, threads)), dim3(threads), 0, e->stream, e->k,
          e->d_seg_begin, e->d_seg_end, e->d_prefix_len, e->d_pw, e->d_dep, e->d_weights
      );
      LAUNCH_CHECK();
    }
    hipLaunchKernelGGL(
        k_apply, dim3(grid), dim3(threads), 0, e->stream, order, props, sto, count, e->d_seg_begin,
        e->d_prefix_len, e->d_labels, big_k ? nullptr : e->d_labels16,
        (big_k || e->k > 256) ? nullptr : e->d_labels8, e->d_admitted_flags,
        e->d_moves
    );
    LAUNCH_CHECK();
  }

  // clear active for the WHOLE chunk's processed set (identical on all
  // ranks) and tally scanned arcs
  hipLaunchKernelGGL(
      k_clear_active, dim3(ceil_div(chunk_hi - chunk_lo, threads)), dim3(threads), 0, e->stream,
      chunk_lo, chunk_hi, e->n, iseed, 0xFFFFFFFFu, e->d_xadj, e->d_active, e->d_unit_active,
      e->d_arcs
  );
  LAUNCH_CHECK();
  if (count > 0) {
    hipLaunchKernelGGL(
        k_activate, dim3(ceil_div(static_cast<u64>(count) * kWave, threads)), dim3(threads), 0,
        e->stream, order, e->d_admitted_flags, props, count, e->d_xadj, e->d_adjncy, e->d_active,
        e->d_unit_active
    );
    LAUNCH_CHECK();
    hipLaunchKernelGGL(
        k_reset_segs, dim3(ceil_div(count, threads)), dim3(threads), 0, e->stream, props, count,
        e->d_seg_begin, e->d_seg_end, e->d_prefix_len
    );
    LAUNCH_CHECK();
  }

  HIP_CHECK(hipEventRecord(cev1, e->stream));
  HIP_CHECK(hipMemcpyAsync(&e->h_moves[1], e->d_moves, sizeof(unsigned long long),
                           hipMemcpyDeviceToHost, e->stream));
  if (big_k) {
    HIP_CHECK(hipMemcpyAsync(&e->h_moves[3], e->d_emptied, sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, e->stream));
  }
  sync_spin(e);
  float cms = 0;
  HIP_CHECK(hipEventElapsedTime(&cms, cev0, cev1));
  e->commit_ms += cms;
  e->last_emptied = big_k ? (e->h_moves[3] - e->h_moves[2]) : 0;
  return static_cast<i64>(e->h_moves[1] - e->h_moves[0]);
}

// ---- sharded commit ABI (multi-GPU; see kernel block comment) ----
// d_* arguments are DEVICE pointers (e.g. torch tensors' data_ptr()).

int kmp_lp_shard_begin(
    kmp_lp_t *e, u32 c_lo, u32 c_hi, const void *d_props, u32 count,
    long long *d_dep_out /* k+1 i64, pre-zeroed by caller */
) {
  const u32 threads = 256;
  const u32 rows = 1024;
  const u32 T = (count + 63) >> 6;
  const u32 tpw = (T + rows - 1) / rows;
  const Prop *props = static_cast<const Prop *>(d_props);
  const size_t lds_h = static_cast<size_t>(threads / kWave) * e->k * sizeof(u32);
  // stage the (host-known) count into the device scalar the kernels read
  // (kernel arg by value: no pinned-buffer reuse races)
  hipLaunchKernelGGL(k_set_u32, dim3(1), dim3(1), 0, e->stream, e->d_prop_count, count);
  LAUNCH_CHECK();
  // reset local arr slot
  HIP_CHECK(hipMemsetAsync(e->d_dep, 0, sizeof(unsigned long long) * 2 * e->k, e->stream));
  if (count > 0) {
    hipLaunchKernelGGL(
        k_hist_props, dim3(rows / 4), dim3(threads), lds_h, e->stream, e->d_prop_count, e->k,
        rows, tpw, c_lo, c_hi, props, e->d_histT
    );
    LAUNCH_CHECK();
  } else {
    HIP_CHECK(hipMemsetAsync(e->d_histT, 0, sizeof(u32) * e->k * rows, e->stream));
  }
  hipLaunchKernelGGL(
      k_scan_small, dim3(1), dim3(threads), 0, e->stream, e->k, rows, e->d_histT, e->d_offT,
      e->d_seg_off, e->d_prefix_len, e->d_dep, e->d_changed
  );
  LAUNCH_CHECK();
  if (count > 0) {
    const size_t lds_sc = lds_h + static_cast<size_t>(2 * e->k) * sizeof(unsigned long long);
    hipLaunchKernelGGL(
        k_scatter_props, dim3(rows / 4), dim3(threads), lds_sc, e->stream, e->d_prop_count,
        e->k, rows, tpw, c_lo, c_hi, props, e->d_offT, e->d_labels, e->d_s_u, e->d_s_w,
        e->d_s_r, e->d_s_to, e->d_s_b, d_dep_out,
        reinterpret_cast<unsigned long long *>(e->d_dep) + e->k
    );
    LAUNCH_CHECK();
    if (e->has_vwgt) {
      hipLaunchKernelGGL(
          k_shard_pw, dim3(64), dim3(threads), 0, e->stream, e->k, c_lo, c_hi, e->d_seg_off,
          e->d_prefix_len, e->d_s_w, e->d_pw
      );
      LAUNCH_CHECK();
    }
  }
  if (e->own_stream == nullptr) {
    HIP_CHECK(hipStreamSynchronize(e->stream)); // cross-stream caller
  }
  return 0;
}

int kmp_lp_shard_round(
    kmp_lp_t *e, u32 c_lo, u32 c_hi, const long long *d_dep_global,
    long long *d_delta_out /* k+1 */
) {
  hipLaunchKernelGGL(
      k_shard_round, dim3(1), dim3(256), 0, e->stream, e->k, c_lo, c_hi,
      static_cast<u32>(e->has_vwgt ? 1 : 0), e->d_seg_off, e->d_prefix_len,
      reinterpret_cast<unsigned long long *>(e->d_dep) + e->k, d_dep_global, e->d_s_w,
      e->d_s_b, e->d_pw, e->d_weights, e->d_maxw, d_delta_out
  );
  LAUNCH_CHECK();
  if (e->own_stream == nullptr) {
    HIP_CHECK(hipStreamSynchronize(e->stream));
  }
  return 0;
}

int kmp_lp_shard_finish_meta(
    kmp_lp_t *e, u32 c_lo, u32 c_hi,
    unsigned long long *d_cutoff_out /* k i64, pre-zeroed */,
    long long *d_arr_out /* k i64, pre-zeroed */
) {
  hipLaunchKernelGGL(
      k_shard_finish_meta, dim3(ceil_div(e->k, 256u)), dim3(256), 0, e->stream, e->k, c_lo,
      c_hi, e->d_seg_off, e->d_prefix_len,
      reinterpret_cast<unsigned long long *>(e->d_dep) + e->k, e->d_s_r, d_cutoff_out,
      d_arr_out
  );
  LAUNCH_CHECK();
  if (e->own_stream == nullptr) {
    HIP_CHECK(hipStreamSynchronize(e->stream));
  }
  return 0;
}

i64 kmp_lp_shard_apply(
    kmp_lp_t *e, int iter, u32 chunk, const void *d_props, u32 count,
    const unsigned long long *d_cutoff_all, const long long *d_arr_all,
    const long long *d_dep_global
) {
  const u32 threads = 256;
  const u64 iseed = iter_seed_of(e->seed, iter);
  const u32 chunk_lo = chunk * e->C;
  const u32 chunk_hi = chunk_lo + e->C > e->P ? e->P : chunk_lo + e->C;
  const Prop *props = static_cast<const Prop *>(d_props);
  hipLaunchKernelGGL(k_set_u32, dim3(1), dim3(1), 0, e->stream, e->d_prop_count, count);
  LAUNCH_CHECK();
  {
    const u32 span = count > e->k ? count : e->k;
    hipLaunchKernelGGL(
        k_shard_apply, dim3(ceil_div(span, threads)), dim3(threads), 0, e->stream,
        e->d_prop_count, e->k, props, d_cutoff_all, d_arr_all, d_dep_global, e->d_weights,
        e->d_labels, e->d_labels16, e->k <= 256 ? e->d_labels8 : nullptr, e->d_admitted_flags,
        e->d_moves
    );
    LAUNCH_CHECK();
  }
  hipLaunchKernelGGL(
      k_clear_active, dim3(ceil_div(chunk_hi - chunk_lo, threads)), dim3(threads), 0, e->stream,
      chunk_lo, chunk_hi, e->n, iseed, 0xFFFFFFFFu, e->d_xadj, e->d_active, e->d_unit_active,
      e->d_arcs
  );
  LAUNCH_CHECK();
  if (count > 0) {
    hipLaunchKernelGGL(
        k_activate, dim3(ceil_div(static_cast<u64>(count) * kWave, threads)), dim3(threads), 0,
        e->stream, static_cast<const u32 *>(nullptr), e->d_admitted_flags, props, count,
        e->d_xadj, e->d_adjncy, e->d_active, e->d_unit_active
    );
    LAUNCH_CHECK();
  }
  // no per-call moves readback: callers poll the cumulative counter per
  // sweep via kmp_lp_get_stats (one sync per sweep instead of per chunk)
  return 0;
}

i64 kmp_lp_refine_end(kmp_lp_t *e, u32 *partition, kmp_lp_stats_t *stats) {
  HIP_CHECK(hipMemsetAsync(e->d_cut, 0, sizeof(unsigned long long), e->stream));
  hipLaunchKernelGGL(
      k_edge_cut, dim3(32768), dim3(256), 0, e->stream, e->n, e->d_xadj, e->d_adjncy, e->d_adjwgt,
      e->d_labels, e->d_cut
  );
  LAUNCH_CHECK();
  unsigned long long cut2 = 0, arcs = 0, moves = 0;
  HIP_CHECK(hipMemcpyAsync(&cut2, e->d_cut, sizeof(cut2), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpyAsync(&arcs, e->d_arcs, sizeof(arcs), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpyAsync(&moves, e->d_moves, sizeof(moves), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpy(partition, e->d_labels, sizeof(u32) * e->n, hipMemcpyDeviceToHost));
  HIP_CHECK(hipStreamSynchronize(e->stream));

  if (stats) {
    stats->arcs_scanned = arcs;
    stats->moves = moves;
    stats->phase_a_ns = static_cast<u64>(e->phase_a_ms * 1e6);
    stats->total_ns = static_cast<u64>(e->commit_ms * 1e6); // commit-region time
    stats->num_clusters = 0;
    stats->edge_cut = static_cast<i64>(cut2 / 2);
  }
  return static_cast<i64>(cut2 / 2);
}

// Reset the engine to the initial partition of the last refine_begin, fully
// on-device (no host transfers): labels, weights, active flags, counters.
int kmp_lp_reset(kmp_lp_t *e) {
  HIP_CHECK(hipMemcpyAsync(
      e->d_labels, e->d_labels0, sizeof(u32) * e->n, hipMemcpyDeviceToDevice, e->stream
  ));
  HIP_CHECK(hipMemsetAsync(e->d_active, 1, e->n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_unit_active, 1, kmp::num_units(e->n), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_arcs, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_moves, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_weights, 0, sizeof(i64) * e->k, e->stream));
  e->phase_a_ms = 0.0;
  e->commit_ms = 0.0;
  e->ev_used = 0;
  {
    const u32 threads = 256;
    hipLaunchKernelGGL(
        k_sync_labels16, dim3(ceil_div(e->n, threads)), dim3(threads), 0, e->stream, e->n,
        e->d_labels, e->d_labels16, e->d_labels8
    );
    LAUNCH_CHECK();
    const size_t lds = static_cast<size_t>(e->k) * sizeof(unsigned long long);
    hipLaunchKernelGGL(
        k_init_weights, dim3(2048), dim3(threads), lds, e->stream, e->n, e->k, e->d_labels,
        e->d_vwgt, reinterpret_cast<unsigned long long *>(e->d_weights)
    );
    LAUNCH_CHECK();
  }
  return 0;
}

// Run the LP sweeps on the current device state (the timed region: all data
// resident in HBM; the v2 path does ONE host sync per sweep, the legacy
// path syncs per chunk). Returns total committed moves.
i64 kmp_lp_run_sweeps(kmp_lp_t *e, int iters) {
  if (v2_eligible(e)) {
    return run_sweeps_v2(e, iters);
  }
  u64 total_moves = 0;
  for (int iter = 0; iter < iters; ++iter) {
    u64 sweep_moves = 0;
    for (u32 chunk = 0; chunk < kmp::kNumChunks; ++chunk) {
      const u32 pos_lo = chunk * e->C;
      const u32 pos_hi = pos_lo + e->C > e->P ? e->P : pos_lo + e->C;
      if (pos_lo >= pos_hi) {
        continue;
      }
      const i64 cnt = kmp_lp_phase_a(e, iter, chunk, pos_lo, pos_hi, e->d_props, e->C);
      if (cnt < 0) {
        return -1;
      }
      const i64 mv = kmp_lp_commit(e, iter, chunk, e->d_props, static_cast<u32>(cnt));
      if (mv < 0) {
        return -1;
      }
      sweep_moves += mv;
    }
    total_moves += sweep_moves;
    if (sweep_moves == 0) {
      break;
    }
  }
  return static_cast<i64>(total_moves);
}

// Light stats (no cut kernel, no label download).
int kmp_lp_get_stats(kmp_lp_t *e, kmp_lp_stats_t *stats) {
  unsigned long long arcs = 0, moves = 0;
  HIP_CHECK(hipMemcpyAsync(&arcs, e->d_arcs, sizeof(arcs), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpyAsync(&moves, e->d_moves, sizeof(moves), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipStreamSynchronize(e->stream));
  stats->arcs_scanned = arcs;
  stats->moves = moves;
  stats->phase_a_ns = static_cast<u64>(e->phase_a_ms * 1e6);
  stats->total_ns = static_cast<u64>(e->commit_ms * 1e6);
  stats->num_clusters = 0;
  stats->edge_cut = -1;
  return 0;
}

i64 kmp_lp_refine(
    kmp_lp_t *e,
    u32 k,
    const i64 *max_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    kmp_lp_stats_t *stats
) {
  if (kmp_lp_refine_begin(e, k, max_block_weights, partition, seed) != 0) {
    return -1;
  }
  if (kmp_lp_run_sweeps(e, iters) < 0) {
    return -1;
  }
  return kmp_lp_refine_end(e, partition, stats);
}

// Overload-balancer mode (the role of the reference's OVERLOAD_BALANCER
// bracketing LP in the default refiner chain, presets.cc:332-338): same
// deterministic schedule and commit as kmp_lp_refine, but a vertex whose
// block exceeds its cap loses "stay" as a candidate, so overloaded blocks
// shed their boundary vertices to the best admissible targets even at
// negative gain (interior vertices follow over subsequent sweeps as the
// boundary peels). Feasible partitions are left untouched up to normal LP
// moves. Returns the resulting edge cut, or -1.
i64 kmp_lp_balance(
    kmp_lp_t *e,
    u32 k,
    const i64 *max_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    kmp_lp_stats_t *stats
) {
  // Isolated vertices have no LP candidates and can never move through the
  // sweeps: assign those sitting in over-cap blocks to the lightest block
  // with room first (deterministic; zero cut impact -- they have no edges).
  if (!e->isolated.empty()) {
    if (kmp_lp_refine_begin(e, k, max_block_weights, partition, seed) != 0) {
      return -1;
    }
    std::vector<i64> w(k);
    HIP_CHECK(hipMemcpyAsync(w.data(), e->d_weights, sizeof(i64) * k,
                             hipMemcpyDeviceToHost, e->stream));
    sync_spin(e);
    for (size_t i = 0; i < e->isolated.size(); ++i) {
      const u32 u = e->isolated[i];
      const i32 uw = e->iso_weights[i];
      const u32 b = partition[u];
      if (w[b] <= max_block_weights[b]) {
        continue;
      }
      i64 best = -1;
      u32 t = b;
      for (u32 c = 0; c < k; ++c) {
        if (c != b && w[c] + uw <= max_block_weights[c] &&
            (best < 0 || w[c] < best)) {
          best = w[c];
          t = c;
        }
      }
      if (t != b) {
        w[b] -= uw;
        w[t] += uw;
        partition[u] = t;
      }
    }
  }
  e->balance = 1;
  const i64 cut = kmp_lp_refine(e, k, max_block_weights, partition, seed, iters, stats);
  e->balance = 0;
  return cut;
}

// Clusterer::set_communities (coarsening/clusterer.h:35,
// lp_clusterer.cc:61-66,193-194): when set, clustering never merges across
// community boundaries (cluster ids are vertex ids, so the per-vertex array
// indexes both sides of the check). Pass null to clear.
int kmp_lp_set_communities(kmp_lp_t *e, const u32 *communities) {
  if (e->d_comm) {
    HIP_CHECK(hipFree(e->d_comm));
    e->d_comm = nullptr;
  }
  e->comm_host.clear();
  if (communities != nullptr) {
    HIP_CHECK(hipMalloc(&e->d_comm, sizeof(u32) * e->n));
    HIP_CHECK(
        hipMemcpy(e->d_comm, communities, sizeof(u32) * e->n, hipMemcpyHostToDevice)
    );
    e->comm_host.assign(communities, communities + e->n);
  }
  return 0;
}

// Underload-balancer mode (the reference's UNDERLOAD_BALANCER closing the
// default refiner chain, presets.cc:332-338; semantics restated from
// refinement/balancer/underload_balancer.cc): fill blocks below their
// minimum weight with best-gain admissible vertices under the deterministic
// chunk schedule; admission is serial in rank order with BOTH per-block
// minima and maxima re-checked per move (the batch analogue of the
// reference's locked per-move checks). No-op when minima are satisfied.
i64 kmp_lp_underload(
    kmp_lp_t *e,
    u32 k,
    const i64 *max_block_weights,
    const i64 *min_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    kmp_lp_stats_t *stats
) {
  if (kmp_lp_refine_begin(e, k, max_block_weights, partition, seed) != 0) {
    return -1;
  }
  if (e->d_minw) {
    HIP_CHECK(hipFree(e->d_minw));
    e->d_minw = nullptr;
  }
  HIP_CHECK(hipMalloc(&e->d_minw, sizeof(i64) * k));
  HIP_CHECK(hipMemcpy(e->d_minw, min_block_weights, sizeof(i64) * k, hipMemcpyHostToDevice));
  e->balance = 2;
  const u32 threads = 256;
  HIP_CHECK(hipMemcpyAsync(&e->h_moves[0], e->d_moves, sizeof(unsigned long long),
                           hipMemcpyDeviceToHost, e->stream));
  sync_spin(e);
  unsigned long long last = e->h_moves[0];
  for (int iter = 0; iter < iters; ++iter) {
    for (u32 chunk = 0; chunk < kmp::kNumChunks; ++chunk) {
      const u32 pos_lo = chunk * e->C;
      const u32 pos_hi = pos_lo + e->C > e->P ? e->P : pos_lo + e->C;
      if (pos_lo >= pos_hi) {
        continue;
      }
      const i64 cnt = kmp_lp_phase_a(e, iter, chunk, pos_lo, pos_hi, e->d_props, e->C);
      if (cnt < 0) {
        e->balance = 0;
        return -1;
      }
      const u64 iseed = iter_seed_of(e->seed, iter);
      hipLaunchKernelGGL(
          k_commit_underload, dim3(1), dim3(64), 0, e->stream, e->d_props, e->d_prop_count,
          e->d_maxw, e->d_minw, e->d_weights, e->d_labels, e->d_labels16,
          k <= 256 ? e->d_labels8 : nullptr, e->d_admitted_flags, e->d_moves
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_clear_active, dim3(ceil_div(pos_hi - pos_lo, threads)), dim3(threads), 0, e->stream,
          pos_lo, pos_hi, e->n, iseed, 0xFFFFFFFFu, e->d_xadj, e->d_active, e->d_unit_active,
          e->d_arcs
      );
      LAUNCH_CHECK();
      if (cnt > 0) {
        hipLaunchKernelGGL(
            k_activate, dim3(ceil_div(static_cast<u64>(cnt) * kWave, threads)), dim3(threads), 0,
            e->stream, static_cast<const u32 *>(nullptr), e->d_admitted_flags, e->d_props,
            static_cast<u32>(cnt), e->d_xadj, e->d_adjncy, e->d_active, e->d_unit_active
        );
        LAUNCH_CHECK();
      }
    }
    HIP_CHECK(hipMemcpyAsync(&e->h_moves[1], e->d_moves, sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, e->stream));
    sync_spin(e);
    const unsigned long long cur = e->h_moves[1];
    const u64 sweep_moves = cur - last;
    last = cur;
    if (sweep_moves == 0) {
      break;
    }
  }
  e->balance = 0;
  return kmp_lp_refine_end(e, partition, stats);
}

i64 kmp_lp_cluster(
    kmp_lp_t *e, i64 max_cluster_weight, u32 desired_clusters, u32 *clustering, u64 seed,
    int iters, kmp_lp_stats_t *stats
) {
  const u32 n = e->n;
  const u32 threads = 256;
  e->k = n;
  e->seed = seed;
  e->clusterer = true;
  e->maxw_uniform = max_cluster_weight;
  e->phase_a_ms = 0.0;
  e->commit_ms = 0.0;
  e->ev_used = 0;

  engine_free_k_buffers(e);
  engine_alloc_k_buffers(e, n);
  if (e->d_weights) {
    HIP_CHECK(hipFree(e->d_weights));
    e->d_weights = nullptr;
  }
  HIP_CHECK(hipMalloc(&e->d_weights, sizeof(i64) * n));
  if (!e->d_favored) {
    HIP_CHECK(hipMalloc(&e->d_favored, sizeof(u32) * n));
    HIP_CHECK(hipMalloc(&e->d_eflag, n));
    HIP_CHECK(hipMemset(e->d_eflag, 0, n));
    HIP_CHECK(hipMalloc(&e->d_emptied, sizeof(unsigned long long)));
    HIP_CHECK(hipMalloc(&e->d_l_hacc, sizeof(unsigned long long)));
    HIP_CHECK(hipMalloc(&e->d_l_hoff, sizeof(u64) * (e->C + 1)));
    HIP_CHECK(hipMalloc(&e->d_l_hbits, sizeof(u32) * e->C));
    // pooled hash for high-degree rows: ~8 slots per average chunk arc
    u64 slots = (e->m / kmp::kNumChunks) * 8;
    if (slots < (1ull << 22)) {
      slots = 1ull << 22;
    }
    if (slots > (1ull << 27)) {
      slots = 1ull << 27;
    }
    e->pool_slots = slots;
    HIP_CHECK(hipMalloc(&e->d_pool_keys, sizeof(u32) * slots));
    HIP_CHECK(hipMalloc(&e->d_pool_vals, sizeof(i32) * slots));
    HIP_CHECK(hipMemset(e->d_pool_keys, 0xFF, sizeof(u32) * slots));
    HIP_CHECK(hipMemset(e->d_pool_vals, 0, sizeof(i32) * slots));
    HIP_CHECK(hipMalloc(&e->d_l_clist, sizeof(u32) * (slots / 2)));
    HIP_CHECK(hipMalloc(&e->d_l_ccnt, sizeof(u32) * e->C));
    // two-hop buffers + temps
    HIP_CHECK(hipMalloc(&e->d_cand, sizeof(u64) * n));
    HIP_CHECK(hipMalloc(&e->d_cand2, sizeof(u64) * n));
    HIP_CHECK(hipMalloc(&e->d_cfav, sizeof(u32) * n));
    HIP_CHECK(hipMalloc(&e->d_crank, sizeof(u32) * n));
    HIP_CHECK(hipMalloc(&e->d_cones, sizeof(u32) * n));
    HIP_CHECK(rocprim::select(
        nullptr, e->cand_select_temp_bytes, e->d_cand, e->d_cand2, e->d_prop_count, n, CandValid()
    ));
    HIP_CHECK(hipMalloc(&e->d_cand_select_temp, e->cand_select_temp_bytes));
    rocprim::double_buffer<u64> ckeys(e->d_cand2, e->d_cand);
    HIP_CHECK(rocprim::radix_sort_keys(nullptr, e->cand_sort_temp_bytes, ckeys, n));
    HIP_CHECK(hipMalloc(&e->d_cand_sort_temp, e->cand_sort_temp_bytes));
    HIP_CHECK(rocprim::inclusive_scan_by_key(
        nullptr, e->cand_scan_temp_bytes, e->d_cfav, e->d_cones, e->d_crank, n,
        rocprim::plus<u32>(), rocprim::equal_to<u32>()
    ));
    HIP_CHECK(hipMalloc(&e->d_cand_scan_temp, e->cand_scan_temp_bytes));
  }
  // the L list cap is the full chunk span for clustering (a refine run may
  // have shrunk l_cap for its dense-gains buffer)
  e->l_cap = e->C;
  if (e->d_l_off) {
    HIP_CHECK(hipFree(e->d_l_off));
  }
  HIP_CHECK(hipMalloc(&e->d_l_off, sizeof(u32) * (e->l_cap + 1)));
  HIP_CHECK(hipMemsetAsync(e->d_emptied, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_arcs, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_moves, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_active, 1, n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_unit_active, 1, kmp::num_units(e->n), e->stream));

  const u32 ngrid = ceil_div(n, threads);
  hipLaunchKernelGGL(k_iota, dim3(ngrid), dim3(threads), 0, e->stream, n, e->d_labels);
  LAUNCH_CHECK();
  hipLaunchKernelGGL(k_iota, dim3(ngrid), dim3(threads), 0, e->stream, n, e->d_favored);
  LAUNCH_CHECK();
  hipLaunchKernelGGL(
      k_init_cluster_weights, dim3(ngrid), dim3(threads), 0, e->stream, n, e->d_vwgt, e->d_weights
  );
  LAUNCH_CHECK();
  HIP_CHECK(hipStreamSynchronize(e->stream));

  // ---- LP sweeps (driver: lp_clusterer.cc:89-109 + should_stop) ----
  u64 live = n;
  for (int iter = 0; iter < iters; ++iter) {
    u64 sweep_moves = 0;
    bool stopped = false;
    for (u32 chunk = 0; chunk < kmp::kNumChunks && !stopped; ++chunk) {
      const u32 pos_lo = chunk * e->C;
      const u32 pos_hi = pos_lo + e->C > e->P ? e->P : pos_lo + e->C;
      if (pos_lo >= pos_hi) {
        continue;
      }
      const i64 cnt = kmp_lp_phase_a(e, iter, chunk, pos_lo, pos_hi, e->d_props, e->C);
      if (cnt < 0) {
        return -1;
      }
      const i64 mv = kmp_lp_commit(e, iter, chunk, e->d_props, static_cast<u32>(cnt));
      if (mv < 0) {
        return -1;
      }
      sweep_moves += mv;
      live -= e->last_emptied;
      if (desired_clusters > 0 && live <= desired_clusters) {
        stopped = true;
      }
    }
    if (sweep_moves == 0) {
      break;
    }
  }

  // ---- isolated nodes + two-hop (default preset strategies;
  //      lp_clusterer.cc:112-162, oracle parity contract) ----
  const bool handle_two_hop = (1.0 - 1.0 * live / n) <= 0.5;
  if (handle_two_hop) {
    // isolated nodes, MATCH semantics: deterministic host chain (isolated
    // clusters stay singletons through LP, so their weights are the node
    // weights captured at engine creation)
    std::vector<u64> pairs;
    u32 pending = 0xFFFFFFFFu;
    i64 pending_w = 0;
    for (size_t i = 0; i < e->isolated.size(); ++i) {
      const u32 cu = e->isolated[i];
      const i64 w = e->iso_weights[i];
      // communities are a hard constraint: chains break at boundaries
      // (keep in sync with the oracle twin)
      if (pending != 0xFFFFFFFFu && !e->comm_host.empty() &&
          e->comm_host[pending] != e->comm_host[cu]) {
        pending = cu;
        pending_w = w;
        continue;
      }
      if (pending != 0xFFFFFFFFu && pending_w + w <= max_cluster_weight) {
        pairs.push_back((static_cast<u64>(pending) << 32) | cu);
        pending = 0xFFFFFFFFu;
      } else {
        pending = cu;
        pending_w = w;
      }
    }
    if (!pairs.empty()) {
      HIP_CHECK(hipMemcpyAsync(
          e->d_cand, pairs.data(), sizeof(u64) * pairs.size(), hipMemcpyHostToDevice, e->stream
      ));
      hipLaunchKernelGGL(
          k_apply_pairs, dim3(ceil_div(pairs.size(), threads)), dim3(threads), 0, e->stream,
          e->d_cand, static_cast<u32>(pairs.size()), e->d_labels, e->d_weights
      );
      LAUNCH_CHECK();
      live -= pairs.size();
    }

    // two-hop matching (MATCH_THREADWISE single-chain semantics): collect
    // singleton candidates, sort by (favored, u), pair consecutive equals
    hipLaunchKernelGGL(
        k_twohop_cand, dim3(ngrid), dim3(threads), 0, e->stream, n, max_cluster_weight, e->d_xadj,
        e->d_vwgt, e->d_labels, e->d_weights, e->d_favored, e->d_cand
    );
    LAUNCH_CHECK();
    size_t stb = e->cand_select_temp_bytes;
    HIP_CHECK(rocprim::select(
        e->d_cand_select_temp, stb, e->d_cand, e->d_cand2, e->d_prop_count, n, CandValid(),
        e->stream
    ));
    u32 cand_count = 0;
    HIP_CHECK(hipMemcpyAsync(
        &cand_count, e->d_prop_count, sizeof(u32), hipMemcpyDeviceToHost, e->stream
    ));
    HIP_CHECK(hipStreamSynchronize(e->stream));
    if (cand_count > 1) {
      rocprim::double_buffer<u64> ckeys(e->d_cand2, e->d_cand);
      size_t ktb = e->cand_sort_temp_bytes;
      HIP_CHECK(rocprim::radix_sort_keys(e->d_cand_sort_temp, ktb, ckeys, cand_count, 0, 64,
                                         e->stream));
      const u64 *sorted = ckeys.current();
      const u32 cgrid = ceil_div(cand_count, threads);
      hipLaunchKernelGGL(
          k_extract_fav, dim3(cgrid), dim3(threads), 0, e->stream, sorted, cand_count, e->d_cfav
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_ones, dim3(cgrid), dim3(threads), 0, e->stream, cand_count, e->d_cones
      );
      LAUNCH_CHECK();
      size_t sctb = e->cand_scan_temp_bytes;
      HIP_CHECK(rocprim::inclusive_scan_by_key(
          e->d_cand_scan_temp, sctb, e->d_cfav, e->d_cones, e->d_crank, cand_count,
          rocprim::plus<u32>(), rocprim::equal_to<u32>(), e->stream
      ));
      HIP_CHECK(hipMemsetAsync(e->d_emptied, 0, sizeof(unsigned long long), e->stream));
      hipLaunchKernelGGL(
          k_twohop_pair, dim3(cgrid), dim3(threads), 0, e->stream, sorted, e->d_crank, cand_count,
          e->d_labels, e->d_weights, e->d_emptied
      );
      LAUNCH_CHECK();
      unsigned long long merged = 0;
      HIP_CHECK(hipMemcpyAsync(&merged, e->d_emptied, sizeof(merged), hipMemcpyDeviceToHost,
                               e->stream));
      HIP_CHECK(hipStreamSynchronize(e->stream));
      live -= merged;
    }
  }

  // exact non-empty cluster count + download
  HIP_CHECK(hipMemsetAsync(e->d_cut, 0, sizeof(unsigned long long), e->stream));
  hipLaunchKernelGGL(
      k_count_nonempty, dim3(ngrid), dim3(threads), 0, e->stream, n, e->d_weights, e->d_cut
  );
  LAUNCH_CHECK();
  unsigned long long nonempty = 0, arcs = 0, moves = 0;
  HIP_CHECK(hipMemcpyAsync(&nonempty, e->d_cut, sizeof(nonempty), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpyAsync(&arcs, e->d_arcs, sizeof(arcs), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpyAsync(&moves, e->d_moves, sizeof(moves), hipMemcpyDeviceToHost, e->stream));
  HIP_CHECK(hipMemcpy(clustering, e->d_labels, sizeof(u32) * n, hipMemcpyDeviceToHost));
  HIP_CHECK(hipStreamSynchronize(e->stream));

  if (stats) {
    stats->arcs_scanned = arcs;
    stats->moves = moves;
    stats->phase_a_ns = static_cast<u64>(e->phase_a_ms * 1e6);
    stats->total_ns = static_cast<u64>(e->commit_ms * 1e6);
    stats->num_clusters = nonempty;
    stats->edge_cut = 0;
  }
  return static_cast<i64>(nonempty);
}

} // extern "C"

// ==================== cluster contraction (GPU) ====================
// Restates kaminpar-shm/coarsening/contraction/ semantics (see
// kmp_contract in include/kaminpar_lp.h). All steps are order-free integer
// sums / stable sorts, so the result is deterministic and bit-identical to
// the CPU oracle and (canonically sorted) to the reference implementation.

namespace {

__global__ void k_mark_clusters(
    u32 n, const u32 *__restrict__ clus, u32 *__restrict__ rank
) {
  const u32 u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u < n) {
    rank[clus[u]] = 1;
  }
}

__global__ void k_map_ranks(
    u32 n, const u32 *__restrict__ clus, const u32 *__restrict__ rank, u32 *__restrict__ map
) {
  const u32 u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u < n) {
    map[u] = rank[clus[u]] - 1;
  }
}

__global__ void k_coarse_vwgt(
    u32 n, const u32 *__restrict__ map, const i32 *__restrict__ vwgt, i32 *__restrict__ cvw
) {
  const u32 u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u < n) {
    atomicAdd(&cvw[map[u]], vwgt ? vwgt[u] : 1);
  }
}

// One 16-lane subgroup per vertex (grid-stride): emit (cu<<32|cv) keys per
// arc; intra-cluster arcs get the ~0 sentinel (sorted to the end, dropped).
__global__ void k_arc_keys(
    u32 n,
    const u64 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ map,
    u64 *__restrict__ keys,
    i32 *__restrict__ vals
) {
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 sub = lane >> 4;
  const u32 slot = lane & 15;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 num_waves = (gridDim.x * blockDim.x) >> 6;
  for (u32 base = wave_id * 4; base < n; base += num_waves * 4) {
    const u32 u = base + sub;
    if (u >= n) {
      continue;
    }
    const u64 row = xadj[u];
    const u32 deg = static_cast<u32>(xadj[u + 1] - row);
    const u32 cu = map[u];
    for (u32 e = slot; e < deg; e += 16) {
      const u32 cv = map[adjncy[row + e]];
      keys[row + e] = (cu == cv) ? ~0ull : ((static_cast<u64>(cu) << 32) | cv);
      vals[row + e] = adjwgt ? adjwgt[row + e] : 1;
    }
  }
}

__global__ void k_coarse_hist(
    u32 c_m, const u64 *__restrict__ ukeys, unsigned long long *__restrict__ cxadj
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < c_m) {
    atomicAdd(&cxadj[(ukeys[i] >> 32) + 1], 1ull);
  }
}

__global__ void k_coarse_adj(
    u32 c_m, const u64 *__restrict__ ukeys, u32 *__restrict__ cadj
) {
  const u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < c_m) {
    cadj[i] = static_cast<u32>(ukeys[i] & 0xFFFFFFFFu);
  }
}

} // namespace

extern "C" {

namespace {

// Output of contract_core: right-sized device arrays the caller either
// downloads (kmp_contract) or adopts into a new engine (kmp_contract_engine).
struct ContractOut {
  u32 c_n = 0;
  u32 c_m = 0;
  u32 *d_map = nullptr;   // n
  u64 *d_cxadj = nullptr; // c_n + 1 (prefix-summed)
  u32 *d_cadj = nullptr;  // max(c_m, 1)
  i32 *d_cvw = nullptr;   // c_n
  i32 *d_cwgt = nullptr;  // max(c_m, 1)
};

int contract_core(kmp_lp_t *e, const u32 *clustering, ContractOut &o) {
  const u32 n = e->n;
  const u64 m = e->m;
  if (m > 0xFFFFFFFFull) {
    fprintf(stderr, "kaminpar_amd: kmp_contract supports m < 2^32 arcs\n");
    return -1;
  }
  const u32 threads = 256;
  const u32 ngrid = ceil_div(n, threads);
  hipStream_t s = e->stream;

  u32 *d_clus = nullptr, *d_rank = nullptr;
  i32 *d_cvw_full = nullptr, *d_vals[2] = {nullptr, nullptr}, *d_usums = nullptr;
  u64 *d_keys[2] = {nullptr, nullptr}, *d_ukeys = nullptr;
  u32 *d_uniq = nullptr;
  HIP_CHECK(hipMalloc(&d_clus, sizeof(u32) * n));
  HIP_CHECK(hipMalloc(&d_rank, sizeof(u32) * n));
  HIP_CHECK(hipMalloc(&o.d_map, sizeof(u32) * n));
  HIP_CHECK(hipMalloc(&d_cvw_full, sizeof(i32) * n));
  HIP_CHECK(hipMalloc(&d_keys[0], sizeof(u64) * m));
  HIP_CHECK(hipMalloc(&d_keys[1], sizeof(u64) * m));
  HIP_CHECK(hipMalloc(&d_vals[0], sizeof(i32) * m));
  HIP_CHECK(hipMalloc(&d_vals[1], sizeof(i32) * m));
  HIP_CHECK(hipMalloc(&d_ukeys, sizeof(u64) * m));
  HIP_CHECK(hipMalloc(&d_usums, sizeof(i32) * m));
  HIP_CHECK(hipMalloc(&d_uniq, sizeof(u32)));

  HIP_CHECK(hipMemcpyAsync(d_clus, clustering, sizeof(u32) * n, hipMemcpyHostToDevice, s));
  HIP_CHECK(hipMemsetAsync(d_rank, 0, sizeof(u32) * n, s));
  hipLaunchKernelGGL(k_mark_clusters, dim3(ngrid), dim3(threads), 0, s, n, d_clus, d_rank);
  LAUNCH_CHECK();

  void *tmp = nullptr;
  size_t tmp_bytes = 0;
  HIP_CHECK(rocprim::inclusive_scan(nullptr, tmp_bytes, d_rank, d_rank, n, rocprim::plus<u32>()));
  HIP_CHECK(hipMalloc(&tmp, tmp_bytes));
  HIP_CHECK(rocprim::inclusive_scan(tmp, tmp_bytes, d_rank, d_rank, n, rocprim::plus<u32>(), s));
  u32 c_n = 0;
  HIP_CHECK(hipMemcpyAsync(&c_n, d_rank + n - 1, sizeof(u32), hipMemcpyDeviceToHost, s));
  hipLaunchKernelGGL(k_map_ranks, dim3(ngrid), dim3(threads), 0, s, n, d_clus, d_rank, o.d_map);
  LAUNCH_CHECK();
  HIP_CHECK(hipMemsetAsync(d_cvw_full, 0, sizeof(i32) * n, s));
  hipLaunchKernelGGL(
      k_coarse_vwgt, dim3(ngrid), dim3(threads), 0, s, n, o.d_map, e->d_vwgt, d_cvw_full
  );
  LAUNCH_CHECK();
  hipLaunchKernelGGL(
      k_arc_keys, dim3(16384), dim3(threads), 0, s, n, e->d_xadj, e->d_adjncy, e->d_adjwgt,
      o.d_map, d_keys[0], d_vals[0]
  );
  LAUNCH_CHECK();

  rocprim::double_buffer<u64> kb(d_keys[0], d_keys[1]);
  rocprim::double_buffer<i32> vb(d_vals[0], d_vals[1]);
  void *tmp2 = nullptr;
  size_t tmp2_bytes = 0;
  HIP_CHECK(rocprim::radix_sort_pairs(nullptr, tmp2_bytes, kb, vb, m, 0, 64));
  HIP_CHECK(hipMalloc(&tmp2, tmp2_bytes));
  HIP_CHECK(rocprim::radix_sort_pairs(tmp2, tmp2_bytes, kb, vb, m, 0, 64, s));

  void *tmp3 = nullptr;
  size_t tmp3_bytes = 0;
  HIP_CHECK(rocprim::reduce_by_key(
      nullptr, tmp3_bytes, kb.current(), vb.current(), m, d_ukeys, d_usums, d_uniq,
      rocprim::plus<i32>(), rocprim::equal_to<u64>()
  ));
  HIP_CHECK(hipMalloc(&tmp3, tmp3_bytes));
  HIP_CHECK(rocprim::reduce_by_key(
      tmp3, tmp3_bytes, kb.current(), vb.current(), m, d_ukeys, d_usums, d_uniq,
      rocprim::plus<i32>(), rocprim::equal_to<u64>(), s
  ));
  u32 uniq = 0;
  HIP_CHECK(hipMemcpyAsync(&uniq, d_uniq, sizeof(u32), hipMemcpyDeviceToHost, s));
  u64 last_key = 0;
  HIP_CHECK(hipStreamSynchronize(s));
  if (uniq > 0) {
    HIP_CHECK(hipMemcpy(&last_key, d_ukeys + uniq - 1, sizeof(u64), hipMemcpyDeviceToHost));
  }
  const u32 c_m = (uniq > 0 && last_key == ~0ull) ? uniq - 1 : uniq;

  HIP_CHECK(hipMalloc(&o.d_cxadj, sizeof(u64) * (c_n + 1)));
  HIP_CHECK(hipMalloc(&o.d_cadj, sizeof(u32) * (c_m > 0 ? c_m : 1)));
  HIP_CHECK(hipMemsetAsync(o.d_cxadj, 0, sizeof(u64) * (c_n + 1), s));
  if (c_m > 0) {
    hipLaunchKernelGGL(
        k_coarse_hist, dim3(ceil_div(c_m, threads)), dim3(threads), 0, s, c_m, d_ukeys,
        reinterpret_cast<unsigned long long *>(o.d_cxadj)
    );
    LAUNCH_CHECK();
    hipLaunchKernelGGL(
        k_coarse_adj, dim3(ceil_div(c_m, threads)), dim3(threads), 0, s, c_m, d_ukeys, o.d_cadj
    );
    LAUNCH_CHECK();
  }
  void *tmp4 = nullptr;
  size_t tmp4_bytes = 0;
  HIP_CHECK(rocprim::inclusive_scan(
      nullptr, tmp4_bytes, o.d_cxadj, o.d_cxadj, c_n + 1, rocprim::plus<u64>()
  ));
  HIP_CHECK(hipMalloc(&tmp4, tmp4_bytes));
  HIP_CHECK(rocprim::inclusive_scan(
      tmp4, tmp4_bytes, o.d_cxadj, o.d_cxadj, c_n + 1, rocprim::plus<u64>(), s
  ));

  // right-size the coarse node/edge weights
  HIP_CHECK(hipMalloc(&o.d_cvw, sizeof(i32) * (c_n > 0 ? c_n : 1)));
  HIP_CHECK(hipMalloc(&o.d_cwgt, sizeof(i32) * (c_m > 0 ? c_m : 1)));
  HIP_CHECK(
      hipMemcpyAsync(o.d_cvw, d_cvw_full, sizeof(i32) * c_n, hipMemcpyDeviceToDevice, s)
  );
  if (c_m > 0) {
    HIP_CHECK(hipMemcpyAsync(o.d_cwgt, d_usums, sizeof(i32) * c_m, hipMemcpyDeviceToDevice, s));
  }
  HIP_CHECK(hipStreamSynchronize(s));

  for (void *p : {(void *)d_clus, (void *)d_rank, (void *)d_cvw_full, (void *)d_keys[0],
                  (void *)d_keys[1], (void *)d_vals[0], (void *)d_vals[1], (void *)d_ukeys,
                  (void *)d_usums, (void *)d_uniq, tmp, tmp2, tmp3, tmp4}) {
    if (p) {
      (void)hipFree(p);
    }
  }
  o.c_n = c_n;
  o.c_m = c_m;
  return 0;
}

void contract_out_free(ContractOut &o) {
  for (void *p : {(void *)o.d_map, (void *)o.d_cxadj, (void *)o.d_cadj, (void *)o.d_cvw,
                  (void *)o.d_cwgt}) {
    if (p) {
      (void)hipFree(p);
    }
  }
}

} // namespace

i64 kmp_contract(
    kmp_lp_t *e, const u32 *clustering, u32 *mapping_out, kmp_graph_t **coarse_out
) {
  ContractOut o;
  if (contract_core(e, clustering, o) != 0) {
    return -1;
  }
  hipStream_t s = e->stream;
  const u32 c_n = o.c_n, c_m = o.c_m;

  std::vector<u64> h_cxadj64(c_n + 1);
  std::vector<u32> h_cxadj(c_n + 1), h_cadj(c_m);
  std::vector<i32> h_cvw(c_n), h_cwgt(c_m);
  HIP_CHECK(hipMemcpyAsync(mapping_out, o.d_map, sizeof(u32) * e->n, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(h_cxadj64.data(), o.d_cxadj, sizeof(u64) * (c_n + 1),
                           hipMemcpyDeviceToHost, s));
  if (c_m > 0) {
    HIP_CHECK(
        hipMemcpyAsync(h_cadj.data(), o.d_cadj, sizeof(u32) * c_m, hipMemcpyDeviceToHost, s)
    );
    HIP_CHECK(
        hipMemcpyAsync(h_cwgt.data(), o.d_cwgt, sizeof(i32) * c_m, hipMemcpyDeviceToHost, s)
    );
  }
  HIP_CHECK(hipMemcpyAsync(h_cvw.data(), o.d_cvw, sizeof(i32) * c_n, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  for (u32 i = 0; i <= c_n; ++i) {
    h_cxadj[i] = static_cast<u32>(h_cxadj64[i]); // c_m < 2^32 (entry guard)
  }

  *coarse_out = kmp_graph_from_csr(
      c_n, c_m, h_cxadj.data(), h_cadj.data(), h_cvw.data(), c_m ? h_cwgt.data() : nullptr
  );
  contract_out_free(o);
  return static_cast<i64>(c_n);
}

i64 kmp_contract_engine(
    kmp_lp_t *e, const u32 *clustering, u32 *mapping_out, kmp_lp_t **coarse_eng_out
) {
  ContractOut o;
  if (contract_core(e, clustering, o) != 0) {
    return -1;
  }
  hipStream_t s = e->stream;
  const u32 c_n = o.c_n, c_m = o.c_m;

  // host copies needed regardless: the mapping (projection happens on the
  // host) and xadj/vwgt for the isolated-vertex scan of the new engine
  std::vector<u64> h_cxadj(c_n + 1);
  std::vector<i32> h_cvw(c_n);
  HIP_CHECK(hipMemcpyAsync(mapping_out, o.d_map, sizeof(u32) * e->n, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(h_cxadj.data(), o.d_cxadj, sizeof(u64) * (c_n + 1),
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(h_cvw.data(), o.d_cvw, sizeof(i32) * c_n, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  HIP_CHECK(hipFree(o.d_map));
  o.d_map = nullptr;

  auto *e2 = new kmp_lp_t();
  e2->n = c_n;
  e2->m = c_m;
  e2->C = kmp::chunk_size_for(c_n);
  e2->P = kmp::pos_count(c_n);
  e2->has_vwgt = true;
  e2->has_adjwgt = true;
  HIP_CHECK(hipStreamCreate(&e2->stream));
  HIP_CHECK(hipEventCreateWithFlags(&e2->sync_ev, hipEventDisableTiming));
  {
    static int cached_mp = 0; // hipGetDeviceProperties costs ms; per-level calls add up
    if (cached_mp == 0) {
      int dev = 0;
      HIP_CHECK(hipGetDevice(&dev));
      hipDeviceProp_t props;
      HIP_CHECK(hipGetDeviceProperties(&props, dev));
      cached_mp = props.multiProcessorCount;
    }
    e2->mp_count = cached_mp;
  }
  e2->d_xadj = o.d_cxadj;
  e2->d_adjncy = o.d_cadj;
  e2->d_vwgt = o.d_cvw;
  e2->d_adjwgt = o.d_cwgt;
  engine_alloc_common(e2);
  engine_scan_isolated(e2, h_cxadj.data(), h_cvw.data());
  *coarse_eng_out = e2;
  return static_cast<i64>(c_n);
}

kmp_graph_t *kmp_lp_download_graph(const kmp_lp_t *e) {
  if (e->m > 0xFFFFFFFFull) {
    fprintf(stderr, "kaminpar_amd: kmp_lp_download_graph supports m < 2^32\n");
    return nullptr;
  }
  std::vector<u64> h_xadj64(e->n + 1);
  std::vector<u32> h_xadj(e->n + 1), h_adj(e->m);
  std::vector<i32> h_vw, h_wg;
  HIP_CHECK(
      hipMemcpy(h_xadj64.data(), e->d_xadj, sizeof(u64) * (e->n + 1), hipMemcpyDeviceToHost)
  );
  for (u32 i = 0; i <= e->n; ++i) {
    h_xadj[i] = static_cast<u32>(h_xadj64[i]);
  }
  HIP_CHECK(hipMemcpy(h_adj.data(), e->d_adjncy, sizeof(u32) * e->m, hipMemcpyDeviceToHost));
  if (e->has_vwgt) {
    h_vw.resize(e->n);
    HIP_CHECK(hipMemcpy(h_vw.data(), e->d_vwgt, sizeof(i32) * e->n, hipMemcpyDeviceToHost));
  }
  if (e->has_adjwgt) {
    h_wg.resize(e->m);
    HIP_CHECK(hipMemcpy(h_wg.data(), e->d_adjwgt, sizeof(i32) * e->m, hipMemcpyDeviceToHost));
  }
  return kmp_graph_from_csr(
      e->n, e->m, h_xadj.data(), h_adj.data(), e->has_vwgt ? h_vw.data() : nullptr,
      e->has_adjwgt ? h_wg.data() : nullptr
  );
}

// On-GPU degree-bucket rearrangement (the reference's default
// NodeOrdering::DEGREE_BUCKETS preprocessing, graphutils/permutator.cc:36-110;
// bit-identical to the host kmp_rearrange_degree_buckets): rebuilds the
// engine's device CSR in bucket order and writes perm_out[u_old] = u_new.
// Call before any refine/cluster; LP state is reset by the next *_begin.
int kmp_lp_rearrange_degree_buckets(kmp_lp_t *e, u32 *perm_out) {
  const u32 n = e->n;
  const u64 m = e->m;
  const u32 threads = 256;
  hipStream_t s = e->stream;
  const u32 rows = 1024;
  const u32 T = (n + 63) >> 6;
  const u32 tpw = (T + rows - 1) / rows;

  u32 *d_perm = nullptr, *d_inv = nullptr, *d_hist = nullptr, *d_off = nullptr;
  u64 *d_nx = nullptr;
  u32 *d_nadj = nullptr;
  i32 *d_nwgt = nullptr, *d_nvw = nullptr;
  HIP_CHECK(hipMalloc(&d_perm, sizeof(u32) * n));
  HIP_CHECK(hipMalloc(&d_inv, sizeof(u32) * n));
  HIP_CHECK(hipMalloc(&d_hist, sizeof(u32) * kDbBuckets * rows));
  HIP_CHECK(hipMalloc(&d_off, sizeof(u32) * kDbBuckets * rows));
  HIP_CHECK(hipMalloc(&d_nx, sizeof(u64) * (n + 1)));
  HIP_CHECK(hipMalloc(&d_nadj, sizeof(u32) * m));
  if (e->has_adjwgt) {
    HIP_CHECK(hipMalloc(&d_nwgt, sizeof(i32) * m));
  }
  if (e->has_vwgt) {
    HIP_CHECK(hipMalloc(&d_nvw, sizeof(i32) * n));
  }

  hipLaunchKernelGGL(
      k_db_hist, dim3(rows / 4), dim3(threads), 0, s, n, rows, tpw, e->d_xadj, d_hist
  );
  LAUNCH_CHECK();
  // exclusive scan of the 33 x rows matrix (single WG; one-time setup cost)
  {
    // reuse k_scan_small's machinery via a plain rocprim scan
    void *tmp = nullptr;
    size_t tb = 0;
    HIP_CHECK(rocprim::exclusive_scan(
        nullptr, tb, d_hist, d_off, 0u, kDbBuckets * rows, rocprim::plus<u32>()
    ));
    HIP_CHECK(hipMalloc(&tmp, tb));
    HIP_CHECK(rocprim::exclusive_scan(
        tmp, tb, d_hist, d_off, 0u, kDbBuckets * rows, rocprim::plus<u32>(), s
    ));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(hipFree(tmp));
  }
  hipLaunchKernelGGL(
      k_db_scatter, dim3(rows / 4), dim3(threads), 0, s, n, rows, tpw, e->d_xadj, d_off,
      d_perm, d_inv
  );
  LAUNCH_CHECK();
  HIP_CHECK(hipMemsetAsync(d_nx, 0, sizeof(u64), s));
  hipLaunchKernelGGL(
      k_db_degrees, dim3(ceil_div(n, threads)), dim3(threads), 0, s, n, e->d_xadj, d_inv, d_nx
  );
  LAUNCH_CHECK();
  {
    void *tmp = nullptr;
    size_t tb = 0;
    HIP_CHECK(rocprim::inclusive_scan(nullptr, tb, d_nx, d_nx, n + 1, rocprim::plus<u64>()));
    HIP_CHECK(hipMalloc(&tmp, tb));
    HIP_CHECK(rocprim::inclusive_scan(tmp, tb, d_nx, d_nx, n + 1, rocprim::plus<u64>(), s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(hipFree(tmp));
  }
  hipLaunchKernelGGL(
      k_db_gather, dim3(16384), dim3(threads), 0, s, n, e->d_xadj, e->d_adjncy, e->d_adjwgt,
      e->d_vwgt, d_perm, d_inv, d_nx, d_nadj, d_nwgt, d_nvw
  );
  LAUNCH_CHECK();
  HIP_CHECK(hipMemcpyAsync(perm_out, d_perm, sizeof(u32) * n, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));

  // adopt the rearranged CSR
  HIP_CHECK(hipFree(e->d_xadj));
  HIP_CHECK(hipFree(e->d_adjncy));
  e->d_xadj = d_nx;
  e->d_adjncy = d_nadj;
  if (e->has_adjwgt) {
    HIP_CHECK(hipFree(e->d_adjwgt));
    e->d_adjwgt = d_nwgt;
  }
  if (e->has_vwgt) {
    HIP_CHECK(hipFree(e->d_vwgt));
    e->d_vwgt = d_nvw;
  }
  // remap the isolated-vertex list (ids changed)
  for (u32 &u : e->isolated) {
    u = perm_out[u];
  }
  HIP_CHECK(hipFree(d_perm));
  HIP_CHECK(hipFree(d_inv));
  HIP_CHECK(hipFree(d_hist));
  HIP_CHECK(hipFree(d_off));
  return 0;
}

// ==================== C++ RCCL distributed driver ====================
// The per-chunk loop of the sharded multi-GPU refinement, driven from C++
// with RCCL called directly on the engine stream (the python/torch
// orchestration of kaminpar_amd.multi measured a ~0.65 ms/chunk host
// floor; this driver's floor is the per-round convergence readback only).
// Protocol identical to refine_dist_sharded / the 2-rank-sim-verified
// kernels (kaminpar-dist/refinement/lp/lp_refiner.cc:296-333).

#define NCCL_CHECK(cmd)                                                        \
  do {                                                                         \
    ncclResult_t r_ = (cmd);                                                   \
    if (r_ != ncclSuccess) {                                                   \
      fprintf(stderr, "kaminpar_amd: RCCL error %s at %s:%d\n",               \
              ncclGetErrorString(r_), __FILE__, __LINE__);                     \
      return -1;                                                               \
    }                                                                          \
  } while (0)

int kmp_nccl_unique_id(void *out128) {
  ncclUniqueId id;
  if (ncclGetUniqueId(&id) != ncclSuccess) {
    return -1;
  }
  memcpy(out128, &id, sizeof(id));
  return 0;
}

void *kmp_nccl_comm_init(int world, int rank, const void *id128) {
  ncclUniqueId id;
  memcpy(&id, id128, sizeof(id));
  ncclComm_t comm = nullptr;
  if (ncclCommInitRank(&comm, world, id, rank) != ncclSuccess) {
    return nullptr;
  }
  return comm;
}

void kmp_nccl_comm_destroy(void *comm) {
  if (comm) {
    (void)ncclCommDestroy(static_cast<ncclComm_t>(comm));
  }
}

i64 kmp_lp_refine_dist(
    kmp_lp_t *e,
    u32 k,
    const i64 *max_block_weights,
    u32 *partition,
    u64 seed,
    int iters,
    void *nccl_comm, // null at world 1
    int rank,
    int world,
    kmp_lp_stats_t *stats
) {
  if (kmp_lp_refine_begin(e, k, max_block_weights, partition, seed) != 0) {
    return -1;
  }
  if (e->coop_nblk < 8 || k > 256) {
    fprintf(stderr, "kaminpar_amd: refine_dist requires the v2 path (k <= 256)\n");
    return -1;
  }
  ncclComm_t comm = static_cast<ncclComm_t>(nccl_comm);
  const u32 threads = 256;
  const u32 cap = e->C / world + 2;      // per-rank proposal capacity
  const u32 seg = cap + 1;               // + trailing count row
  const u32 c_lo = (static_cast<u64>(rank) * k) / world;
  const u32 c_hi = (static_cast<u64>(rank) + 1) * k / world;

  Prop *d_send = nullptr, *d_recv = nullptr;
  long long *d_coll = nullptr; // dep[k+1] | delta[k+1] | meta[2k]
  HIP_CHECK(hipMalloc(&d_send, sizeof(Prop) * seg));
  HIP_CHECK(hipMalloc(&d_recv, sizeof(Prop) * seg * world));
  HIP_CHECK(hipMalloc(&d_coll, sizeof(long long) * (2 * (k + 1) + 2 * k)));
  long long *d_dep_g = d_coll;
  long long *d_delta = d_coll + (k + 1);
  long long *d_meta = d_coll + 2 * (k + 1); // [0..k) cutoffs, [k..2k) arr
  const u32 rows = 1024;
  const u32 cap_total = seg * world; // bound for tpw sizing
  const u32 tpw = (((cap_total + 63) >> 6) + rows - 1) / rows;
  const size_t lds_h = static_cast<size_t>(threads / kWave) * k * sizeof(u32);
  const size_t lds_sc = lds_h + static_cast<size_t>(2 * k) * sizeof(unsigned long long);

  unsigned long long moves_prev = 0;
  for (int iter = 0; iter < iters; ++iter) {
    const u64 iseed = iter_seed_of(e->seed, iter);
    e->ev_used = 0; // per-sweep phase-A event pairs (bench roofline input)
    for (u32 chunk = 0; chunk < kmp::kNumChunks; ++chunk) {
      const u32 lo = chunk * e->C;
      const u32 hi = lo + e->C > e->P ? e->P : lo + e->C;
      if (lo >= hi) {
        continue;
      }
      const u32 span = hi - lo;
      const u32 slo = lo + (static_cast<u64>(rank) * span) / world;
      const u32 shi = lo + ((static_cast<u64>(rank) + 1) * span) / world;

      // phase A on the rank's slice; compact (stable, position order =
      // rank order) into the send buffer; count rides the trailing row.
      // Slots MUST be pre-invalidated here: the memset-free phase A leaves
      // stale proposals in inactive units' slots, which the v2 commit
      // gates by unit_active but a flat compaction cannot.
      HIP_CHECK(hipMemsetAsync(e->d_slots, 0xFF, sizeof(Prop) * (shi - slo), e->stream));
      {
        hipEvent_t a0 = ev_one(e);
        HIP_CHECK(hipEventRecord(a0, e->stream));
      }
      phase_a_v2(e, iter, slo, shi, lo);
      {
        hipEvent_t a1 = ev_one(e);
        HIP_CHECK(hipEventRecord(a1, e->stream));
      }
      {
        size_t tb = e->select_temp_bytes;
        HIP_CHECK(rocprim::select(
            e->d_select_temp, tb, e->d_slots, d_send, e->d_prop_count, shi - slo, PropValid(),
            e->stream
        ));
      }
      hipLaunchKernelGGL(k_pack_count, dim3(1), dim3(1), 0, e->stream, e->d_prop_count,
                         d_send + cap);
      LAUNCH_CHECK();

      if (world > 1) {
        NCCL_CHECK(ncclAllGather(d_send, d_recv, static_cast<size_t>(seg) * 4, ncclUint32,
                                 comm, e->stream));
      } else {
        HIP_CHECK(hipMemcpyAsync(d_recv, d_send, sizeof(Prop) * seg, hipMemcpyDeviceToDevice,
                                 e->stream));
      }
      hipLaunchKernelGGL(
          k_compact_gathered, dim3(256), dim3(threads), 0, e->stream, world, cap, d_recv,
          e->d_props, e->d_prop_count
      );
      LAUNCH_CHECK();

      // sort own targets + full-admission dep/arr
      HIP_CHECK(hipMemsetAsync(e->d_dep, 0, sizeof(unsigned long long) * 2 * k, e->stream));
      HIP_CHECK(hipMemsetAsync(d_coll, 0, sizeof(long long) * (k + 1), e->stream));
      hipLaunchKernelGGL(
          k_hist_props, dim3(rows / 4), dim3(threads), lds_h, e->stream, e->d_prop_count, k,
          rows, tpw, c_lo, c_hi, e->d_props, e->d_histT
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_scan_small, dim3(1), dim3(threads), 0, e->stream, k, rows, e->d_histT, e->d_offT,
          e->d_seg_off, e->d_prefix_len, e->d_dep, e->d_changed
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_scatter_props, dim3(rows / 4), dim3(threads), lds_sc, e->stream, e->d_prop_count,
          k, rows, tpw, c_lo, c_hi, e->d_props, e->d_offT, e->d_labels, e->d_s_u, e->d_s_w,
          e->d_s_r, e->d_s_to, e->d_s_b, d_dep_g,
          reinterpret_cast<unsigned long long *>(e->d_dep) + k
      );
      LAUNCH_CHECK();
      if (e->has_vwgt) {
        hipLaunchKernelGGL(
            k_shard_pw, dim3(64), dim3(threads), 0, e->stream, k, c_lo, c_hi, e->d_seg_off,
            e->d_prefix_len, e->d_s_w, e->d_pw
        );
        LAUNCH_CHECK();
      }
      if (world > 1) {
        NCCL_CHECK(ncclAllReduce(d_dep_g, d_dep_g, k + 1, ncclInt64, ncclSum, comm,
                                 e->stream));
      }

      // fixpoint rounds: local cutoffs + allreduced de-admission deltas
      for (;;) {
        hipLaunchKernelGGL(
            k_shard_round, dim3(1), dim3(256), 0, e->stream, k, c_lo, c_hi,
            static_cast<u32>(e->has_vwgt ? 1 : 0), e->d_seg_off, e->d_prefix_len,
            reinterpret_cast<unsigned long long *>(e->d_dep) + k, d_dep_g, e->d_s_w, e->d_s_b,
            e->d_pw, e->d_weights, e->d_maxw, d_delta
        );
        LAUNCH_CHECK();
        if (world > 1) {
          NCCL_CHECK(ncclAllReduce(d_delta, d_delta, k + 1, ncclInt64, ncclSum, comm,
                                   e->stream));
        }
        HIP_CHECK(hipMemcpyAsync(e->h_moves, d_delta + k, sizeof(long long),
                                 hipMemcpyDeviceToHost, e->stream));
        sync_spin(e);
        if (*reinterpret_cast<long long *>(e->h_moves) == 0) {
          break;
        }
        hipLaunchKernelGGL(
            k_vec_sub_i64, dim3(ceil_div(k, threads)), dim3(threads), 0, e->stream, k, d_dep_g,
            d_delta
        );
        LAUNCH_CHECK();
      }

      // exchange cutoffs + admitted arrivals (one summed 2k vector), apply
      HIP_CHECK(hipMemsetAsync(d_meta, 0, sizeof(long long) * 2 * k, e->stream));
      hipLaunchKernelGGL(
          k_shard_finish_meta, dim3(ceil_div(k, 256u)), dim3(256), 0, e->stream, k, c_lo, c_hi,
          e->d_seg_off, e->d_prefix_len,
          reinterpret_cast<unsigned long long *>(e->d_dep) + k, e->d_s_r,
          reinterpret_cast<unsigned long long *>(d_meta), d_meta + k
      );
      LAUNCH_CHECK();
      if (world > 1) {
        NCCL_CHECK(ncclAllReduce(d_meta, d_meta, 2 * k, ncclInt64, ncclSum, comm, e->stream));
      }
      {
        const u32 span2 = cap_total > k ? cap_total : k;
        hipLaunchKernelGGL(
            k_shard_apply, dim3(ceil_div(span2, threads)), dim3(threads), 0, e->stream,
            e->d_prop_count, k, e->d_props,
            reinterpret_cast<const unsigned long long *>(d_meta), d_meta + k, d_dep_g,
            e->d_weights, e->d_labels, e->d_labels16, k <= 256 ? e->d_labels8 : nullptr,
            e->d_admitted_flags, e->d_moves
        );
        LAUNCH_CHECK();
      }
      hipLaunchKernelGGL(
          k_clear_active, dim3(ceil_div(span, threads)), dim3(threads), 0, e->stream, lo, hi,
          e->n, iseed, 0xFFFFFFFFu, e->d_xadj, e->d_active, e->d_unit_active, e->d_arcs
      );
      LAUNCH_CHECK();
      hipLaunchKernelGGL(
          k_activate_props_dev, dim3(2048), dim3(threads), 0, e->stream, e->d_prop_count,
          e->d_admitted_flags, e->d_props, e->d_xadj, e->d_adjncy, e->d_active,
          e->d_unit_active
      );
      LAUNCH_CHECK();
    }
    // early exit: moves counter is bit-identical on every rank
    HIP_CHECK(hipMemcpyAsync(&e->h_moves[1], e->d_moves, sizeof(unsigned long long),
                             hipMemcpyDeviceToHost, e->stream));
    sync_spin(e);
    for (size_t i = 0; i + 1 < e->ev_used; i += 2) {
      float ms = 0;
      HIP_CHECK(hipEventElapsedTime(&ms, e->ev_pool[i], e->ev_pool[i + 1]));
      e->phase_a_ms += ms;
    }
    if (e->h_moves[1] == moves_prev) {
      break;
    }
    moves_prev = e->h_moves[1];
  }
  // arcs are tallied per position slice: sum over ranks so stats report
  // the whole job (every other counter is already rank-identical)
  if (world > 1) {
    NCCL_CHECK(ncclAllReduce(
        reinterpret_cast<unsigned long long *>(e->d_arcs),
        reinterpret_cast<unsigned long long *>(e->d_arcs), 1, ncclUint64, ncclSum, comm,
        e->stream
    ));
    HIP_CHECK(hipStreamSynchronize(e->stream));
  }
  HIP_CHECK(hipFree(d_send));
  HIP_CHECK(hipFree(d_recv));
  HIP_CHECK(hipFree(d_coll));
  return kmp_lp_refine_end(e, partition, stats);
}

u32 kmp_lp_n(const kmp_lp_t *e) { return e->n; }
u64 kmp_lp_m(const kmp_lp_t *e) { return e->m; }

} // extern "C"

// ---------------------------------------------------------------------------
// Full multilevel pipeline (C-ABI twin of kaminpar_amd/partition.py --
// keep the schedule in sync; bit-identical by construction and pinned by
// tests/test_gpu_parity.py::test_c_abi_partition_matches_python).
// ---------------------------------------------------------------------------

namespace {

i64 level_cluster_weight(i64 total_w, u32 n, u32 k, double eps, u32 C) {
  u64 shrink = n / C;
  if (shrink < 2) {
    shrink = 2;
  }
  if (shrink > k) {
    shrink = k;
  }
  const i64 eps_rule = static_cast<i64>(eps * static_cast<double>(total_w) /
                                        static_cast<double>(shrink));
  const i64 block_rule = total_w / (12ll * k);
  i64 mcw = block_rule > 0 ? std::min(eps_rule, block_rule) : eps_rule;
  return mcw < 1 ? 1 : mcw;
}

} // namespace

extern "C" {

i64 kmp_partition(
    const kmp_graph_t *g, u32 k, double eps, u64 seed, int iters,
    u32 contraction_limit, u32 stop_n, int ip_reps, u32 *part_out
) {
  if (contraction_limit == 0) {
    contraction_limit = 2000;
  }
  if (stop_n == 0) {
    stop_n = 512;
  }
  if (ip_reps == 0) {
    ip_reps = 8;
  }
  const i64 total_w = kmp_graph_total_node_weight(g);
  const i64 mbw_val = kmp_max_block_weight(g, k, eps);
  std::vector<i64> mbw(k, mbw_val);

  std::vector<kmp_lp_t *> engines;
  std::vector<u32> sizes;
  std::vector<std::vector<u32>> mappings;
  engines.push_back(kmp_lp_create(g));
  if (!engines[0]) {
    return -1;
  }
  sizes.push_back(kmp_graph_n(g));

  i64 rc = -1;
  kmp_graph_t *coarsest = nullptr;
  std::vector<u32> clus, part;
  kmp_lp_stats_t st;

  const u32 stop = std::max(stop_n, 2 * k);
  while (sizes.back() > stop) {
    const u32 cur_n = sizes.back();
    const i64 mcw = level_cluster_weight(total_w, cur_n, k, eps, contraction_limit);
    clus.resize(cur_n);
    if (kmp_lp_cluster(engines.back(), mcw, 0, clus.data(),
                       seed + mappings.size(), iters, &st) < 0) {
      goto done;
    }
    {
      std::vector<u32> mapping(cur_n);
      kmp_lp_t *coarse_eng = nullptr;
      if (kmp_contract_engine(engines.back(), clus.data(), mapping.data(),
                              &coarse_eng) < 0) {
        goto done;
      }
      const u32 c_n = kmp_lp_n(coarse_eng);
      if (static_cast<double>(c_n) > 0.95 * static_cast<double>(cur_n)) {
        kmp_lp_free(coarse_eng);
        break;
      }
      engines.push_back(coarse_eng);
      mappings.push_back(std::move(mapping));
      sizes.push_back(c_n);
    }
  }

  // initial partition on the coarsest graph (CPU)
  coarsest = engines.size() > 1 ? kmp_lp_download_graph(engines.back()) : nullptr;
  part.resize(sizes.back());
  if (kmp_initial_partition(coarsest ? coarsest : g, k, mbw_val, ip_reps,
                            part.data()) != 0) {
    goto done;
  }

  // uncoarsen: refine at every level, projecting through the mappings;
  // per-level k-way boundary FM on small graphs (keep in sync with
  // partition() / the oracle mirror)
  for (size_t level = engines.size(); level-- > 0;) {
    rc = kmp_lp_refine(engines[level], k, mbw.data(), part.data(), seed,
                       iters, &st);
    if (rc < 0) {
      goto done;
    }
    if (kmp_graph_n(g) <= (1u << 21)) {
      kmp_graph_t *hg_owned =
          level == 0 ? nullptr : kmp_lp_download_graph(engines[level]);
      const kmp_graph_t *hg = level == 0 ? g : hg_owned;
      kmp_kway_fm(hg, k, mbw.data(), part.data(), 0, 0);
      if (level == 0) {
        rc = kmp_edge_cut_host(g, part.data());
      }
      if (hg_owned) {
        kmp_graph_free(hg_owned);
      }
    }
    if (level > 0) {
      const std::vector<u32> &map = mappings[level - 1];
      std::vector<u32> fine(map.size());
      for (size_t u = 0; u < map.size(); ++u) {
        fine[u] = part[map[u]];
      }
      part = std::move(fine);
    }
  }
  for (u32 u = 0; u < kmp_graph_n(g); ++u) {
    part_out[u] = part[u];
  }

done:
  if (coarsest) {
    kmp_graph_free(coarsest);
  }
  for (kmp_lp_t *e : engines) {
    kmp_lp_free(e);
  }
  return rc;
}

// --------------------------- ckaminpar-shaped shim (see kaminpar_lp.h) ----

struct kaminpar_amd_t {
  kmp_graph_t *g = nullptr;
  u32 k = 2;
  double eps = 0.03;
  u64 seed = 1;
  std::vector<i64> abs_maxw; // per-block max weights (kaminpar.h:961)
  std::vector<i64> minw;     // per-block min weights (kaminpar.h:965-968)
};

kaminpar_amd_t *kaminpar_amd_create(int /*num_threads*/) {
  return new kaminpar_amd_t();
}

void kaminpar_amd_free(kaminpar_amd_t *shm) {
  if (shm) {
    if (shm->g) {
      kmp_graph_free(shm->g);
    }
    delete shm;
  }
}

void kaminpar_amd_reseed(kaminpar_amd_t *shm, int seed) {
  shm->seed = static_cast<u64>(seed);
}

void kaminpar_amd_copy_graph(
    kaminpar_amd_t *shm, u32 n, const u32 *xadj, const u32 *adjncy,
    const i32 *vwgt, const i32 *adjwgt
) {
  if (shm->g) {
    kmp_graph_free(shm->g);
  }
  shm->g = kmp_graph_from_csr(n, xadj[n], xadj, adjncy, vwgt, adjwgt);
}

void kaminpar_amd_set_k(kaminpar_amd_t *shm, u32 k) { shm->k = k; }

void kaminpar_amd_set_uniform_max_block_weights(kaminpar_amd_t *shm, double epsilon) {
  shm->eps = epsilon;
  shm->abs_maxw.clear();
}

// kaminpar.h:961 set_absolute_max_block_weights: explicit per-block caps.
void kaminpar_amd_set_absolute_max_block_weights(
    kaminpar_amd_t *shm, const i64 *weights, u32 count
) {
  shm->abs_maxw.assign(weights, weights + count);
}

// kaminpar.h:965 set_uniform_min_block_weights: minimum block weights as a
// fraction of the perfectly balanced weight (context.cc:72-80); triggers the
// underload balancer after partitioning (the reference's refiner chain runs
// it only when minima are configured, presets.cc:332-338).
void kaminpar_amd_set_uniform_min_block_weights(kaminpar_amd_t *shm, double min_epsilon) {
  // marker encoding {-1, min_epsilon in ppb}; materialized at compute time
  // (the actual minima need k and the total node weight)
  shm->minw.assign(2, -1);
  shm->minw[1] = static_cast<i64>(min_epsilon * 1e9);
}

// kaminpar.h:966-967 absolute variant.
void kaminpar_amd_set_absolute_min_block_weights(
    kaminpar_amd_t *shm, const i64 *weights, u32 count
) {
  shm->minw.assign(weights, weights + count);
}

void kaminpar_amd_clear_min_block_weights(kaminpar_amd_t *shm) {
  shm->minw.clear();
}

i64 kaminpar_amd_compute_partition(kaminpar_amd_t *shm, u32 *partition) {
  if (!shm->g) {
    fprintf(stderr, "kaminpar_amd: no graph set (call kaminpar_amd_copy_graph)\n");
    return -1;
  }
  const u32 k = shm->k;
  const i64 W = kmp_graph_total_node_weight(shm->g);
  double eps = shm->eps;
  std::vector<i64> caps;
  if (!shm->abs_maxw.empty()) {
    // pipeline under a uniform proxy derived from the average cap; exact
    // per-block enforcement happens below via balance + refine under the
    // true caps (both engine paths take per-block arrays)
    caps = shm->abs_maxw;
    caps.resize(k, caps.empty() ? 0 : caps.back());
    i64 cap_sum = 0;
    for (i64 c : caps) {
      cap_sum += c;
    }
    const double avg_cap = static_cast<double>(cap_sum) / k;
    eps = avg_cap * k / static_cast<double>(W) - 1.0;
    if (eps < 0.001) {
      eps = 0.001;
    }
  }
  // progressive-k (deep) pipeline: best measured cuts
  i64 cut = kmp_partition_deep(shm->g, k, eps, shm->seed, 5, 0, 0, 0, 0, partition);
  if (cut < 0) {
    return cut;
  }
  const bool want_min = !shm->minw.empty();
  if (!caps.empty() || want_min) {
    kmp_lp_t *e = kmp_lp_create(shm->g);
    if (!e) {
      return -1;
    }
    if (caps.empty()) {
      const i64 uni = kmp_max_block_weight(shm->g, k, shm->eps);
      caps.assign(k, uni);
    }
    if (!caps.empty() && !shm->abs_maxw.empty()) {
      // exact per-block caps: repair + refine under them
      cut = kmp_lp_balance(e, k, caps.data(), partition, shm->seed, 5, nullptr);
      if (cut >= 0) {
        cut = kmp_lp_refine(e, k, caps.data(), partition, shm->seed, 5, nullptr);
      }
    }
    if (cut >= 0 && want_min) {
      std::vector<i64> minw = shm->minw;
      if (minw.size() == 2 && minw[0] == -1) {
        // uniform min epsilon (ppb-encoded): ceil((1-eps_min) * W/k)
        const double me = static_cast<double>(minw[1]) / 1e9;
        const i64 v = static_cast<i64>(
            std::ceil((1.0 - me) * static_cast<double>(W) / k));
        minw.assign(k, v);
      } else {
        minw.resize(k, 0);
      }
      cut = kmp_lp_underload(e, k, caps.data(), minw.data(), partition, shm->seed, 5, nullptr);
    }
    kmp_lp_free(e);
  }
  return cut;
}

} // extern "C"

extern "C" {

i64 kmp_partition_deep(
    const kmp_graph_t *g, u32 k, double eps, u64 seed, int iters,
    u32 contraction_limit, u32 stop_n, u32 split_c, int ip_reps, u32 *part_out
) {
  if (contraction_limit == 0) {
    contraction_limit = 2000;
  }
  if (stop_n == 0) {
    stop_n = 512;
  }
  if (split_c == 0) {
    // auto: fine-level splits up to ~2M vertices, reference-like block
    // sizes beyond (see kaminpar_amd/partition.py partition_deep)
    split_c = kmp_graph_n(g) <= (1u << 21) ? 262144 : 2000;
  }
  if (ip_reps == 0) {
    ip_reps = 8;
  }
  // split-schedule dispatch by degree variance of the fine graph (keep in
  // sync with partition_deep): heavy-tailed graphs defer all splits to the
  // finest level (no eager coarsest split); CV^2 >= 1 as for the bisector
  // dispatch.
  bool late_splits = false;
  {
    const u32 fn = kmp_graph_n(g);
    const u32 *fx = kmp_graph_xadj(g);
    unsigned __int128 sum = 0, sq = 0;
    for (u32 u = 0; u < fn; ++u) {
      const u64 d = fx[u + 1] - fx[u];
      sum += d;
      sq += d * d;
    }
    const bool heavy =
        static_cast<unsigned __int128>(fn) * sq >= 2 * sum * sum;
    late_splits = heavy && (fn <= (1u << 21) || split_c >= fn);
  }
  const i64 total_w = kmp_graph_total_node_weight(g);
  const i64 mbw_val = kmp_max_block_weight(g, k, eps);

  std::vector<kmp_lp_t *> engines;
  std::vector<u32> sizes;
  std::vector<std::vector<u32>> mappings;
  engines.push_back(kmp_lp_create(g));
  if (!engines[0]) {
    return -1;
  }
  sizes.push_back(kmp_graph_n(g));

  i64 rc = -1;
  std::vector<u32> clus, part;
  std::vector<u32> group_lo(k, 0), group_w(k, 0);
  u32 num_groups = 1;
  group_w[0] = k;
  std::vector<i64> caps(k, 0);
  kmp_lp_stats_t st;
  size_t coarsest = 0;

  const u32 stop = std::max(stop_n, 2 * k);
  while (sizes.back() > stop) {
    const u32 cur_n = sizes.back();
    const i64 mcw = level_cluster_weight(total_w, cur_n, k, eps, contraction_limit);
    clus.resize(cur_n);
    if (kmp_lp_cluster(engines.back(), mcw, 0, clus.data(),
                       seed + mappings.size(), iters, &st) < 0) {
      goto done;
    }
    {
      std::vector<u32> mapping(cur_n);
      kmp_lp_t *coarse_eng = nullptr;
      if (kmp_contract_engine(engines.back(), clus.data(), mapping.data(),
                              &coarse_eng) < 0) {
        goto done;
      }
      const u32 c_n = kmp_lp_n(coarse_eng);
      if (static_cast<double>(c_n) > 0.95 * static_cast<double>(cur_n)) {
        kmp_lp_free(coarse_eng);
        break;
      }
      engines.push_back(coarse_eng);
      mappings.push_back(std::move(mapping));
      sizes.push_back(c_n);
    }
  }

  part.assign(sizes.back(), 0);
  coarsest = engines.size() - 1;
  for (size_t level = engines.size(); level-- > 0;) {
    const u32 sc = (level == coarsest && !late_splits)
                       ? std::min(split_c, 48u) : split_c;
    kmp_graph_t *hg_owned = nullptr;
    const kmp_graph_t *hg = nullptr;
    if (num_groups < k &&
        (static_cast<u64>(sizes[level]) >= 2ull * sc * num_groups ||
         level == 0)) {
      hg_owned = level == 0 ? nullptr : kmp_lp_download_graph(engines[level]);
      hg = level == 0 ? g : hg_owned;
      kmp_extend_partition(hg, part.data(), k, mbw_val, sc, ip_reps,
                           level == 0 ? 1 : 0, group_lo.data(),
                           group_w.data(), &num_groups);
      if (num_groups == k) {
        kmp_balance_partition(hg, k, mbw_val, part.data());
      }
    }
    std::fill(caps.begin(), caps.end(), 0);
    for (u32 i = 0; i < num_groups; ++i) {
      caps[group_lo[i]] = static_cast<i64>(group_w[i]) * mbw_val;
    }
    rc = kmp_lp_refine(engines[level], k, caps.data(), part.data(), seed,
                       iters, &st);
    if (rc < 0) {
      if (hg_owned) {
        kmp_graph_free(hg_owned);
      }
      goto done;
    }
    // per-level k-way boundary FM on small graphs (keep in sync with
    // partition_deep / the oracle mirror)
    if (kmp_graph_n(g) <= (1u << 21)) {
      if (!hg) {
        hg_owned =
            level == 0 ? nullptr : kmp_lp_download_graph(engines[level]);
        hg = level == 0 ? g : hg_owned;
      }
      kmp_kway_fm(hg, k, caps.data(), part.data(), 0, 0);
      if (level == 0) {
        rc = kmp_edge_cut_host(g, part.data());
      }
    }
    if (hg_owned) {
      kmp_graph_free(hg_owned);
    }
    if (level > 0) {
      const std::vector<u32> &map = mappings[level - 1];
      std::vector<u32> fine(map.size());
      for (size_t u = 0; u < map.size(); ++u) {
        fine[u] = part[map[u]];
      }
      part = std::move(fine);
    }
  }
  for (u32 u = 0; u < kmp_graph_n(g); ++u) {
    part_out[u] = part[u];
  }

done:
  for (kmp_lp_t *e : engines) {
    kmp_lp_free(e);
  }
  return rc;
}

} // extern "C"
