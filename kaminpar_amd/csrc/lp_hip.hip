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

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string.h>
#include <vector>

#include <hip/hip_runtime.h>

#include <rocprim/rocprim.hpp>

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
    return 16;
  }
  if (k <= 256) {
    return 4;
  }
  return 1;
}

// Weight-acceptance predicate (refiner variant, lp_refiner.cc:185-230).
__device__ inline bool accept_refine(
    u32 c, u32 cur, i32 u_w, i64 cw, i64 maxw, i64 cur_w, i64 cur_maxw
) {
  return (cw + u_w <= maxw) || ((cw - maxw) < (cur_w - cur_maxw)) || (c == cur);
}

// ------------------------------------------------------------ S path
// Covers EVERY position of the slice: 4 positions per wave, 16 lanes each.
// Owns the slot for: tail positions (u >= n), inactive or degree-filtered
// vertices (invalid slot), and active deg <= 16 vertices (computed result).
// Leaves deg in (16, inf) active slots for the M/L kernels.
__global__ void k_phase_s(
    u32 pos_lo,
    u32 pos_hi,
    u32 chunk_base,
    u32 n,
    u64 iter_seed,
    u32 max_degree,
    const u32 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const uint8_t *__restrict__ active,
    Prop *__restrict__ slots,
    u64 *__restrict__ l_list,
    u32 *__restrict__ l_count
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
  const u32 u = perm(p);
  const u32 sidx = p - pos_lo;

  bool emit_invalid = false;
  u32 row = 0, deg = 0;
  if (u >= n) {
    emit_invalid = true;
  } else {
    row = xadj[u];
    deg = xadj[u + 1] - row;
    if (!active[u] || deg > max_degree) {
      emit_invalid = true;
    }
  }

  // append active high-degree positions to the L work list (rare;
  // wave-aggregated: one atomic per wave that holds an L candidate)
  {
    const bool is_l = !emit_invalid && u < n && deg > kMidDeg && slot == 0;
    const unsigned long long ll = __ballot(is_l);
    if (ll) {
      const u32 leader = __ffsll(static_cast<unsigned long long>(ll)) - 1;
      u32 bbase = 0;
      if (lane == leader) {
        bbase = atomicAdd(l_count, static_cast<u32>(__popcll(ll)));
      }
      bbase = __shfl(bbase, leader, kWave);
      if (is_l) {
        l_list[bbase + __popcll(ll & ((1ull << lane) - 1))] =
            (static_cast<u64>(p) << 32) | u;
      }
    }
  }
  if (!emit_invalid && deg > kSmallDeg) {
    return; // M/L owns this slot
  }
  if (emit_invalid) {
    if (slot == 0) {
      slots[sidx] = Prop{0, kInvalid, 0, 0};
    }
    return;
  }

  // candidate load: lane handles one edge
  u32 c = kInvalid;
  i32 w = 0;
  if (slot < deg) {
    const u32 v = adjncy[row + slot];
    c = labels[v];
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

  BestState best{0, 0, 0, false};
  if (owner && c != kInvalid) {
    const i64 cw = weights[c];
    const i64 mw = maxw[c];
    if (accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw)) {
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
    if (best.have && best.c != cur) {
      slots[sidx] = Prop{u, best.c, p - chunk_base, static_cast<u32>(u_w)};
    } else {
      slots[sidx] = Prop{0, kInvalid, 0, 0};
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
template <bool kUnitWeights>
__global__ void k_phase_m(
    u32 pos_lo,
    u32 pos_hi,
    u32 chunk_base,
    u32 n,
    u64 iter_seed,
    u32 max_degree,
    u32 k,
    const u32 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
    const uint8_t *__restrict__ active,
    Prop *__restrict__ slots
) {
  extern __shared__ i32 lds[];
  const u32 lane = threadIdx.x & (kWave - 1);
  const u32 wave_in_wg = threadIdx.x >> 6;
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 p = pos_lo + wave_id;
  if (p >= pos_hi) {
    return;
  }

  const BlockPerm perm(n, iter_seed);
  const u32 u = perm(p);
  if (u >= n) {
    return;
  }
  const u32 row = xadj[u];
  const u32 deg = xadj[u + 1] - row;
  if (deg <= kSmallDeg || deg > kMidDeg || deg > max_degree || !active[u]) {
    return; // S or L owns this slot
  }

  const u32 R = gain_replicas(k);
  i32 *gains = lds + wave_in_wg * k * R; // R replicas of k counters
  for (u32 c = lane; c < k * R; c += kWave) {
    gains[c] = 0;
  }
  __threadfence_block(); // LDS ordering; gains slice is private to this wave

  const u32 rep_off = (lane % R) * k;
  for (u32 e = lane; e < deg; e += kWave) {
    const u32 v = adjncy[row + e];
    const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
    atomicAdd(&gains[rep_off + labels[v]], w);
  }
  __threadfence_block();

  const u32 cur = labels[u];
  const i32 u_w = vwgt ? vwgt[u] : 1;
  const i64 cur_w = weights[cur];
  const i64 cur_maxw = maxw[cur];

  BestState best{0, 0, 0, false};
  for (u32 c = lane; c < k; c += kWave) {
    i32 g = gains[c];
    for (u32 r = 1; r < R; ++r) {
      g += gains[r * k + c];
    }
    if (g <= 0) {
      continue;
    }
    const i64 cw = weights[c];
    const i64 mw = maxw[c];
    if (!accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw)) {
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
    const u32 sidx = p - pos_lo;
    if (best.have && best.c != cur) {
      slots[sidx] = Prop{u, best.c, p - chunk_base, static_cast<u32>(u_w)};
    } else {
      slots[sidx] = Prop{0, kInvalid, 0, 0};
    }
  }
}

// ------------------------------------------------------------ L path
// High-degree vertices are processed slice-parallel: rows are cut into
// kLSlice-edge slices, each handled by one workgroup accumulating into a
// per-vertex global gains row (via a per-WG replicated LDS histogram), then
// a selection kernel reduces each row. This keeps mega-hubs (100K+ edges)
// from serializing on a single workgroup.
constexpr u32 kLSlice = 8192;

// Single tiny kernel: per-vertex slice counts -> exclusive prefix (l_off),
// total in l_off[count]. One wave; l_count is small (hubs are rare).
__global__ void k_l_prep(
    const u64 *__restrict__ l_list,
    const u32 *__restrict__ l_count,
    const u32 *__restrict__ xadj,
    u32 l_cap,
    u32 *__restrict__ l_off
) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    const u32 count = *l_count < l_cap ? *l_count : l_cap;
    u32 acc = 0;
    for (u32 i = 0; i < count; ++i) {
      l_off[i] = acc;
      const u32 u = static_cast<u32>(l_list[i]);
      const u32 deg = xadj[u + 1] - xadj[u];
      acc += (deg + kLSlice - 1) / kLSlice;
    }
    l_off[count] = acc;
  }
}

// Accumulate one slice per workgroup into the vertex's global gains row.
template <bool kUnitWeights>
__global__ void k_phase_l_acc(
    u32 k,
    const u32 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
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
    const u32 row = xadj[u];
    const u32 deg = xadj[u + 1] - row;
    const u32 e_lo = (s - l_off[vid]) * kLSlice;
    const u32 e_hi = e_lo + kLSlice < deg ? e_lo + kLSlice : deg;

    for (u32 c = threadIdx.x; c < k * R; c += blockDim.x) {
      hist[c] = 0;
    }
    __syncthreads();
    const u32 rep_off = (threadIdx.x % R) * k;
    for (u32 e = e_lo + threadIdx.x; e < e_hi; e += blockDim.x) {
      const u32 v = adjncy[row + e];
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      atomicAdd(&hist[rep_off + labels[v]], w);
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
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    u32 k,
    const u32 *__restrict__ xadj,
    const i32 *__restrict__ vwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
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
    i32 *grow = l_gains + static_cast<size_t>(vid) * k;

    BestState best{0, 0, 0, false};
    for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
      const i32 g = grow[c];
      grow[c] = 0; // reset for the next chunk
      if (g <= 0) {
        continue;
      }
      const i64 cw = weights[c];
      const i64 mw = maxw[c];
      if (!accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw)) {
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
      if (total.have && total.c != cur) {
        slots[p - pos_lo] = Prop{u, total.c, p - chunk_base, static_cast<u32>(u_w)};
      } else {
        slots[p - pos_lo] = Prop{0, kInvalid, 0, 0};
      }
    }
    __syncthreads();
  }
}

// Fallback for L entries beyond l_cap (pathological): one workgroup per
// vertex, whole row, replicated LDS histogram.
template <bool kUnitWeights>
__global__ void k_phase_l_direct(
    u32 pos_lo,
    u32 chunk_base,
    u64 iter_seed,
    u32 k,
    const u32 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    const i32 *__restrict__ vwgt,
    const i32 *__restrict__ adjwgt,
    const u32 *__restrict__ labels,
    const i64 *__restrict__ weights,
    const i64 *__restrict__ maxw,
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
    const u32 row = xadj[u];
    const u32 deg = xadj[u + 1] - row;

    const u32 rep_off = (threadIdx.x % R) * k;
    for (u32 e = threadIdx.x; e < deg; e += blockDim.x) {
      const u32 v = adjncy[row + e];
      const i32 w = kUnitWeights ? 1 : adjwgt[row + e];
      atomicAdd(&gains[rep_off + labels[v]], w);
    }
    __syncthreads();

    const u32 cur = labels[u];
    const i32 u_w = vwgt ? vwgt[u] : 1;
    const i64 cur_w = weights[cur];
    const i64 cur_maxw = maxw[cur];

    BestState best{0, 0, 0, false};
    for (u32 c = threadIdx.x; c < k; c += blockDim.x) {
      i32 g = gains[c];
      for (u32 r = 1; r < R; ++r) {
        g += gains[r * k + c];
      }
      if (g <= 0) {
        continue;
      }
      const i64 cw = weights[c];
      const i64 mw = maxw[c];
      if (!accept_refine(c, cur, u_w, cw, mw, cur_w, cur_maxw)) {
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
      if (total.have && total.c != cur) {
        slots[p - pos_lo] = Prop{u, total.c, p - chunk_base, static_cast<u32>(u_w)};
      } else {
        slots[p - pos_lo] = Prop{0, kInvalid, 0, 0};
      }
    }
    __syncthreads();
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

__global__ void k_seg_len(
    u32 k_or_n,
    const u32 *__restrict__ seg_begin,
    const u32 *__restrict__ seg_end,
    u32 *__restrict__ prefix_len
) {
  const u32 c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c < k_or_n && seg_end[c] > seg_begin[c]) {
    prefix_len[c] = seg_end[c] - seg_begin[c];
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
      labels[props[order[i]].u] = to;
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
    const u32 *__restrict__ xadj,
    uint8_t *__restrict__ active,
    unsigned long long *__restrict__ arcs
) {
  __shared__ unsigned long long wg_sum[4];
  const u32 tid = blockIdx.x * blockDim.x + threadIdx.x;
  const u32 p = chunk_lo + tid;
  const BlockPerm perm(n, iter_seed);

  u64 my_deg = 0;
  if (p < chunk_hi) {
    const u32 u = perm(p);
    if (u < n) {
      const u32 deg = xadj[u + 1] - xadj[u];
      if (deg <= max_degree && active[u]) {
        my_deg = deg;
        active[u] = 0;
      }
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
    const u32 *__restrict__ xadj,
    const u32 *__restrict__ adjncy,
    uint8_t *__restrict__ active
) {
  const u32 wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const u32 lane = threadIdx.x & (kWave - 1);
  if (wave_id >= count || !admitted_flags[wave_id]) {
    return;
  }
  const u32 u = props[order[wave_id]].u;
  const u32 row = xadj[u];
  const u32 deg = xadj[u + 1] - row;
  for (u32 e = lane; e < deg; e += kWave) {
    active[adjncy[row + e]] = 1;
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
    const u32 *__restrict__ xadj,
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
    const u32 row = xadj[u];
    const u32 deg = xadj[u + 1] - row;
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

  // device graph
  u32 *d_xadj = nullptr;
  u32 *d_adjncy = nullptr;
  i32 *d_vwgt = nullptr;
  i32 *d_adjwgt = nullptr;

  // device LP state
  u32 *d_labels = nullptr;
  u32 *d_labels0 = nullptr; // initial labels (for kmp_lp_reset)
  i64 *d_weights = nullptr;
  i64 *d_maxw = nullptr;
  uint8_t *d_active = nullptr;

  // phase buffers
  Prop *d_slots = nullptr; // C
  Prop *d_props = nullptr; // C (compacted; single-GPU commit input)
  u64 *d_l_list = nullptr; // C
  u32 *d_l_count = nullptr;
  u32 *d_l_off = nullptr;  // l_cap + 1 (slice prefix)
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

  // pinned host mirrors
  u32 *h_count = nullptr;
  int *h_changed = nullptr;
  unsigned long long *h_moves = nullptr; // [0]=before [1]=after

  hipStream_t stream = nullptr;

  // run bookkeeping
  double phase_a_ms = 0.0;
  double commit_ms = 0.0;
  std::vector<hipEvent_t> ev_pool;
  size_t ev_used = 0;

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

void engine_alloc_k_buffers(kmp_lp_t *e, u32 k_or_n) {
  HIP_CHECK(hipMalloc(&e->d_seg_begin, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_seg_end, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_prefix_len, sizeof(u32) * k_or_n));
  HIP_CHECK(hipMalloc(&e->d_dep, sizeof(unsigned long long) * k_or_n));
  HIP_CHECK(hipMemsetAsync(e->d_seg_begin, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_seg_end, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_prefix_len, 0, sizeof(u32) * k_or_n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_dep, 0, sizeof(unsigned long long) * k_or_n, e->stream));
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

} // namespace

extern "C" {

kmp_lp_t *kmp_lp_create(const kmp_graph_t *g) {
  int ndev = 0;
  if (hipGetDeviceCount(&ndev) != hipSuccess || ndev == 0) {
    fprintf(stderr, "kaminpar_amd: no HIP device available -- the LP engine requires a GPU\n");
    return nullptr;
  }

  auto *e = new kmp_lp_t();
  e->n = kmp_graph_n(g);
  e->m = kmp_graph_m(g);
  e->C = kmp::chunk_size_for(e->n);
  e->P = kmp::pos_count(e->n);
  e->has_vwgt = kmp_graph_vwgt(g) != nullptr;
  e->has_adjwgt = kmp_graph_adjwgt(g) != nullptr;
  HIP_CHECK(hipStreamCreate(&e->stream));

  HIP_CHECK(hipMalloc(&e->d_xadj, sizeof(u32) * (e->n + 1)));
  HIP_CHECK(hipMalloc(&e->d_adjncy, sizeof(u32) * e->m));
  HIP_CHECK(hipMemcpy(e->d_xadj, kmp_graph_xadj(g), sizeof(u32) * (e->n + 1), hipMemcpyHostToDevice));
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

  HIP_CHECK(hipMalloc(&e->d_labels, sizeof(u32) * e->n));
  HIP_CHECK(hipMalloc(&e->d_labels0, sizeof(u32) * e->n));
  HIP_CHECK(hipMalloc(&e->d_active, e->n));

  const u32 C = e->C;
  HIP_CHECK(hipMalloc(&e->d_slots, sizeof(Prop) * C));
  HIP_CHECK(hipMalloc(&e->d_props, sizeof(Prop) * C));
  HIP_CHECK(hipMalloc(&e->d_l_list, sizeof(u64) * C));
  HIP_CHECK(hipMalloc(&e->d_l_count, sizeof(u32)));
  HIP_CHECK(hipMalloc(&e->d_prop_count, sizeof(u32)));
  HIP_CHECK(hipMalloc(&e->d_arcs, sizeof(unsigned long long)));
  HIP_CHECK(hipMalloc(&e->d_moves, sizeof(unsigned long long)));

  HIP_CHECK(hipMalloc(&e->d_sort_keys[0], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_keys[1], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_vals[0], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sort_vals[1], sizeof(u32) * C));
  HIP_CHECK(hipMalloc(&e->d_sw, sizeof(i64) * C));
  HIP_CHECK(hipMalloc(&e->d_pw, sizeof(i64) * C));
  HIP_CHECK(hipMalloc(&e->d_changed, sizeof(int)));
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
  HIP_CHECK(hipHostMalloc(&e->h_moves, sizeof(unsigned long long) * 2));
  return e;
}

void kmp_lp_free(kmp_lp_t *e) {
  if (!e) {
    return;
  }
  HIP_CHECK(hipDeviceSynchronize());
  for (hipEvent_t ev : e->ev_pool) {
    (void)hipEventDestroy(ev);
  }
  engine_free_k_buffers(e);
  for (void *p : {(void *)e->d_xadj, (void *)e->d_adjncy, (void *)e->d_vwgt, (void *)e->d_adjwgt,
                  (void *)e->d_labels, (void *)e->d_labels0, (void *)e->d_weights, (void *)e->d_maxw, (void *)e->d_active,
                  (void *)e->d_slots, (void *)e->d_props, (void *)e->d_l_list,
                  (void *)e->d_l_count, (void *)e->d_l_off, (void *)e->d_l_gains, (void *)e->d_prop_count, (void *)e->d_arcs,
                  (void *)e->d_moves, (void *)e->d_sort_keys[0], (void *)e->d_sort_keys[1],
                  (void *)e->d_sort_vals[0], (void *)e->d_sort_vals[1], (void *)e->d_sort_temp,
                  (void *)e->d_select_temp, (void *)e->d_sw, (void *)e->d_pw,
                  (void *)e->d_scan_temp, (void *)e->d_changed, (void *)e->d_admitted_flags,
                  (void *)e->d_cut}) {
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
  (void)hipStreamDestroy(e->stream);
  delete e;
}

u32 kmp_lp_num_chunks(const kmp_lp_t *) { return kmp::kNumChunks; }

int kmp_lp_refine_begin(
    kmp_lp_t *e, u32 k, const i64 *max_block_weights, const u32 *partition, u64 seed
) {
  if (k > kMaxDenseK) {
    fprintf(stderr, "kaminpar_amd: refine currently supports k <= %u (got %u)\n", kMaxDenseK, k);
    return -1;
  }
  e->k = k;
  e->seed = seed;
  e->phase_a_ms = 0.0;
  e->commit_ms = 0.0;
  e->ev_used = 0;

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
  HIP_CHECK(hipMalloc(&e->d_l_gains, sizeof(i32) * e->l_cap * k));
  HIP_CHECK(hipMemsetAsync(e->d_l_gains, 0, sizeof(i32) * e->l_cap * k, e->stream));
  HIP_CHECK(hipMemcpy(e->d_maxw, max_block_weights, sizeof(i64) * k, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(e->d_labels, partition, sizeof(u32) * e->n, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpyAsync(
      e->d_labels0, e->d_labels, sizeof(u32) * e->n, hipMemcpyDeviceToDevice, e->stream
  ));
  HIP_CHECK(hipMemsetAsync(e->d_active, 1, e->n, e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_arcs, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_moves, 0, sizeof(unsigned long long), e->stream));

  HIP_CHECK(hipMemsetAsync(e->d_weights, 0, sizeof(i64) * k, e->stream));
  {
    const u32 threads = 256;
    const size_t lds = static_cast<size_t>(k) * sizeof(unsigned long long);
    hipLaunchKernelGGL(
        k_init_weights, dim3(2048), dim3(threads), lds, e->stream, e->n, k, e->d_labels, e->d_vwgt,
        reinterpret_cast<unsigned long long *>(e->d_weights)
    );
    LAUNCH_CHECK();
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
  const u32 threads = 256;
  const u32 max_degree = 0xFFFFFFFFu;

  HIP_CHECK(hipMemsetAsync(e->d_l_count, 0, sizeof(u32), e->stream));

  hipEvent_t ev0, ev1;
  e->ev_pair(ev0, ev1);
  HIP_CHECK(hipEventRecord(ev0, e->stream));

  // S: 4 positions/wave (default owner of every position's slot)
  hipLaunchKernelGGL(
      k_phase_s, dim3(ceil_div(static_cast<u64>(ceil_div(span, 4)) * kWave, threads)),
      dim3(threads), 0, e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed, max_degree, e->d_xadj,
      e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_active,
      e->d_slots, e->d_l_list, e->d_l_count
  );
  LAUNCH_CHECK();
  // M: one wave per position
  {
    const size_t lds =
        static_cast<size_t>(threads / kWave) * e->k * gain_replicas(e->k) * sizeof(i32);
    auto *kern = e->has_adjwgt ? k_phase_m<false> : k_phase_m<true>;
    hipLaunchKernelGGL(
        kern, dim3(ceil_div(static_cast<u64>(span) * kWave, threads)), dim3(threads), lds,
        e->stream, pos_lo, pos_hi, chunk_base, e->n, iseed, max_degree, e->k, e->d_xadj,
        e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_active,
        e->d_slots
    );
    LAUNCH_CHECK();
  }
  // L: slice-parallel accumulation over the (rare) high-degree list
  {
    hipLaunchKernelGGL(
        k_l_prep, dim3(1), dim3(64), 0, e->stream, e->d_l_list, e->d_l_count, e->d_xadj, e->l_cap,
        e->d_l_off
    );
    LAUNCH_CHECK();
    const size_t hist_lds = static_cast<size_t>(e->k) * gain_replicas(e->k) * sizeof(i32);
    {
      auto *kern = e->has_adjwgt ? k_phase_l_acc<false> : k_phase_l_acc<true>;
      hipLaunchKernelGGL(
          kern, dim3(2048), dim3(256), hist_lds, e->stream, e->k, e->d_xadj, e->d_adjncy,
          e->d_adjwgt, e->d_labels, e->d_l_list, e->d_l_count, e->l_cap, e->d_l_off, e->d_l_gains
      );
      LAUNCH_CHECK();
    }
    hipLaunchKernelGGL(
        k_phase_l_sel, dim3(2048), dim3(256), 0, e->stream, pos_lo, chunk_base, iseed, e->k,
        e->d_xadj, e->d_vwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_l_list, e->d_l_count,
        e->l_cap, e->d_l_gains, e->d_slots
    );
    LAUNCH_CHECK();
    // pathological overflow beyond l_cap: direct per-vertex workgroups
    {
      const size_t lds =
          ((static_cast<size_t>(e->k) * gain_replicas(e->k) + 1) & ~1ull) * sizeof(i32) +
          16 * sizeof(i64);
      auto *kern = e->has_adjwgt ? k_phase_l_direct<false> : k_phase_l_direct<true>;
      hipLaunchKernelGGL(
          kern, dim3(512), dim3(256), lds, e->stream, pos_lo, chunk_base, iseed, e->k, e->d_xadj,
          e->d_adjncy, e->d_vwgt, e->d_adjwgt, e->d_labels, e->d_weights, e->d_maxw, e->d_l_list,
          e->d_l_count, e->l_cap, e->d_slots
      );
      LAUNCH_CHECK();
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
  HIP_CHECK(hipStreamSynchronize(e->stream));
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
  hipEvent_t cev0, cev1;
  e->ev_pair(cev0, cev1);
  HIP_CHECK(hipEventRecord(cev0, e->stream));
  const u32 chunk_lo = chunk * e->C;
  const u32 chunk_hi = chunk_lo + e->C > e->P ? e->P : chunk_lo + e->C;

  HIP_CHECK(hipMemcpyAsync(&e->h_moves[0], e->d_moves, sizeof(unsigned long long),
                           hipMemcpyDeviceToHost, e->stream));

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
    sto = keys.current(); // sorted target clusters

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
    const u32 kgrid = ceil_div(e->k, threads);
    hipLaunchKernelGGL(
        k_seg_len, dim3(kgrid), dim3(threads), 0, e->stream, e->k, e->d_seg_begin, e->d_seg_end,
        e->d_prefix_len
    );
  LAUNCH_CHECK();

    // greatest-fixpoint rollback (kaminpar-dist lp_refiner.cc:296-333).
    // Two rounds are enqueued per host sync (the fixpoint almost always
    // converges within two); the changed flag of the SECOND round decides.
    while (true) {
      for (int half = 0; half < 2; ++half) {
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
            k_cutoff, dim3(kgrid), dim3(threads), 0, e->stream, e->k, e->d_seg_begin, e->d_seg_end,
            e->d_prefix_len, e->d_pw, e->d_weights, e->d_maxw, e->d_dep, e->d_changed
        );
        LAUNCH_CHECK();
      }
      HIP_CHECK(
          hipMemcpyAsync(e->h_changed, e->d_changed, sizeof(int), hipMemcpyDeviceToHost, e->stream)
      );
      HIP_CHECK(hipStreamSynchronize(e->stream));
      if (!*e->h_changed) {
        break;
      }
    }

    hipLaunchKernelGGL(
        k_weights_update, dim3(kgrid), dim3(threads), 0, e->stream, e->k, e->d_seg_begin,
        e->d_seg_end, e->d_prefix_len, e->d_pw, e->d_dep, e->d_weights
    );
  LAUNCH_CHECK();
    hipLaunchKernelGGL(
        k_apply, dim3(grid), dim3(threads), 0, e->stream, order, props, sto, count, e->d_seg_begin,
        e->d_prefix_len, e->d_labels, e->d_admitted_flags, e->d_moves
    );
  LAUNCH_CHECK();
  }

  // clear active for the WHOLE chunk's processed set (identical on all
  // ranks) and tally scanned arcs
  hipLaunchKernelGGL(
      k_clear_active, dim3(ceil_div(chunk_hi - chunk_lo, threads)), dim3(threads), 0, e->stream,
      chunk_lo, chunk_hi, e->n, iseed, 0xFFFFFFFFu, e->d_xadj, e->d_active, e->d_arcs
  );
  LAUNCH_CHECK();
  if (count > 0) {
    hipLaunchKernelGGL(
        k_activate, dim3(ceil_div(static_cast<u64>(count) * kWave, threads)), dim3(threads), 0,
        e->stream, order, e->d_admitted_flags, props, count, e->d_xadj, e->d_adjncy, e->d_active
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
  HIP_CHECK(hipStreamSynchronize(e->stream));
  float cms = 0;
  HIP_CHECK(hipEventElapsedTime(&cms, cev0, cev1));
  e->commit_ms += cms;
  return static_cast<i64>(e->h_moves[1] - e->h_moves[0]);
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
  HIP_CHECK(hipMemsetAsync(e->d_arcs, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_moves, 0, sizeof(unsigned long long), e->stream));
  HIP_CHECK(hipMemsetAsync(e->d_weights, 0, sizeof(i64) * e->k, e->stream));
  e->phase_a_ms = 0.0;
  e->commit_ms = 0.0;
  e->ev_used = 0;
  {
    const u32 threads = 256;
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
// resident in HBM, no host transfers besides the per-chunk control syncs).
// Returns total committed moves.
i64 kmp_lp_run_sweeps(kmp_lp_t *e, int iters) {
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
    if (sweep_moves == 0) {
      break;
    }
  }
  return kmp_lp_refine_end(e, partition, stats);
}

i64 kmp_lp_cluster(
    kmp_lp_t *e, i64 max_cluster_weight, u32 desired_clusters, u32 *clustering, u64 seed,
    int iters, kmp_lp_stats_t *stats
) {
  (void)e;
  (void)max_cluster_weight;
  (void)desired_clusters;
  (void)clustering;
  (void)seed;
  (void)iters;
  (void)stats;
  fprintf(stderr, "kaminpar_amd: kmp_lp_cluster GPU path not implemented yet\n");
  return -1;
}

} // extern "C"
