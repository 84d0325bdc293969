// Shared host/device primitives for the deterministic LP schedule.
// The PRNG/permutation/tie-hash definitions here are the GPU-side half of the
// parity contract; oracle/lp_oracle.cpp restates them independently and the
// bit-parity suites in tests/test_gpu_parity.py assert both sides agree.
#pragma once

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define KMP_HD __host__ __device__
#else
#define KMP_HD
#endif

namespace kmp {

using u32 = uint32_t;
using u64 = uint64_t;
using i32 = int32_t;
using i64 = int64_t;

// Chunks per sweep (deterministic schedule; see oracle/lp_oracle.cpp header).
constexpr u32 kNumChunks = 64;

KMP_HD inline u64 splitmix64(u64 x) {
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

KMP_HD inline u64 mix_seed(u64 seed, u64 salt) {
  return splitmix64(seed ^ (salt * 0xD1B54A32D192ED03ULL));
}

KMP_HD inline u64 iter_seed_of(u64 seed, int iter) {
  return mix_seed(seed, 0x17E5ULL + static_cast<u64>(iter));
}

// Tie-breaking hash: h(u, c) under the per-iteration seed.
KMP_HD inline u64 tie_hash(u64 iter_seed, u32 u, u32 c) {
  return splitmix64(iter_seed ^ (static_cast<u64>(u) * 0x9E3779B97F4A7C15ULL) ^ c);
}

// 4-round Feistel permutation of [0, n) with cycle-walking; stateless O(1)
// bijection so host and device agree without materializing the permutation.
struct FeistelPerm {
  u32 n;
  u32 half_bits;
  u32 half_mask;
  u64 keys[4];

  KMP_HD FeistelPerm(u32 n_, u64 seed) : n(n_) {
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

  KMP_HD inline u32 apply_once(u32 x) const {
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

  KMP_HD inline u32 operator()(u32 p) const {
    u32 x = apply_once(p);
    while (x >= n) {
      x = apply_once(x);
    }
    return x;
  }
};

// Block permutation: vertices are grouped into units of 64 consecutive ids
// (the reference's own randomization granularity, kPermutationSize = 64,
// label_propagation.h:52) and the UNITS are permuted by the Feistel network.
// Consecutive ids inside a unit keep CSR reads coalesced on the GPU.
// Position space: [0, num_blocks(n) * 64); a position may map to u >= n
// (tail of the last unit) and is then skipped.
constexpr u32 kUnit = 64;

KMP_HD inline u32 num_units(u32 n) { return (n + kUnit - 1) / kUnit; }

// positions per chunk (multiple of kUnit; kNumChunks chunks cover all units)
KMP_HD inline u32 chunk_size_for(u32 n) {
  const u32 nu = num_units(n);
  return ((nu + kNumChunks - 1) / kNumChunks) * kUnit;
}

KMP_HD inline u32 pos_count(u32 n) { return num_units(n) * kUnit; }

struct BlockPerm {
  FeistelPerm fp;
  u32 n;

  KMP_HD BlockPerm(u32 n_, u64 seed) : fp(num_units(n_), seed), n(n_) {}

  // maps position -> vertex id; result >= n means "no vertex" (skip)
  KMP_HD inline u32 operator()(u32 p) const {
    return fp(p / kUnit) * kUnit + (p % kUnit);
  }
};

} // namespace kmp
