// Microbenchmark: resident-grid barrier variants on gfx950.
// Build: hipcc --offload-arch=gfx950 -O3 tools/bar_bench.hip -o /tmp/bar_bench
// Times N back-to-back grid barriers inside one kernel, for several
// orderings of the arrival atomics and spin loads, and block counts.
#include <cstdio>
#include <hip/hip_runtime.h>

#define CHK(c)                                                                 \
  do {                                                                         \
    hipError_t e_ = (c);                                                       \
    if (e_ != hipSuccess) {                                                    \
      printf("err %s @%d\n", hipGetErrorString(e_), __LINE__);                 \
      return 1;                                                                \
    }                                                                          \
  } while (0)

using u32 = unsigned;

// variant 0: ACQ_REL arrivals, ACQUIRE spin (round-1 implementation)
// variant 1: RELAXED arrivals + threadfence before; RELAXED spin + fence after
// variant 2: like 1 but single flat counter
// variant 3: like 1, no s_sleep
template <int V>
__device__ void bar(u32 *b, u32 nblk) {
  __syncthreads();
  if (threadIdx.x == 0) {
    u32 *sub = b;
    u32 *root = b + 8;
    u32 *gen = b + 9;
    if (V == 1 || V == 3) {
      __threadfence();
    }
    const u32 g = __hip_atomic_load(gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    bool last = false;
    if (V == 2) {
      if (__hip_atomic_fetch_add(root, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) ==
          nblk - 1) {
        __hip_atomic_store(root, 0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        __hip_atomic_fetch_add(gen, 1u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        last = true;
      }
    } else {
      const int mo = (V == 0) ? __ATOMIC_ACQ_REL : __ATOMIC_RELAXED;
      const u32 grp = blockIdx.x & 7u;
      const u32 gsz = nblk >> 3;
      if (__hip_atomic_fetch_add(&sub[grp], 1u, mo, __HIP_MEMORY_SCOPE_AGENT) == gsz - 1) {
        if (__hip_atomic_fetch_add(root, 1u, mo, __HIP_MEMORY_SCOPE_AGENT) == 7u) {
          for (u32 i = 0; i < 8; ++i) {
            __hip_atomic_store(&sub[i], 0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
          }
          __hip_atomic_store(root, 0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
          __hip_atomic_fetch_add(gen, 1u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
          last = true;
        }
      }
    }
    if (!last) {
      const int mo = (V == 0) ? __ATOMIC_ACQUIRE : __ATOMIC_RELAXED;
      u32 spins = 0;
      while (__hip_atomic_load(gen, mo, __HIP_MEMORY_SCOPE_AGENT) == g) {
        if (V != 3) {
          __builtin_amdgcn_s_sleep(1);
        }
        if (++spins > 400000000u) {
          __builtin_trap();
        }
      }
    }
    if (V == 1 || V == 3) {
      __threadfence();
    }
  }
  __syncthreads();
}

template <int V>
__global__ void k_bar(u32 *b, u32 nblk, u32 iters) {
  for (u32 i = 0; i < iters; ++i) {
    bar<V>(b, nblk);
  }
}

__global__ void k_empty() {}

int main() {
  u32 *d_b;
  CHK(hipMalloc(&d_b, 64));
  CHK(hipMemset(d_b, 0, 64));
  const u32 iters = 2000;
  hipEvent_t e0, e1;
  CHK(hipEventCreate(&e0));
  CHK(hipEventCreate(&e1));

  for (u32 nblk : {64u, 128u, 256u}) {
    // warm
    hipLaunchKernelGGL(k_bar<1>, dim3(nblk), dim3(256), 0, 0, d_b, nblk, 10);
    CHK(hipDeviceSynchronize());
#define RUN(V)                                                                 \
  {                                                                            \
    CHK(hipEventRecord(e0));                                                   \
    hipLaunchKernelGGL(k_bar<V>, dim3(nblk), dim3(256), 0, 0, d_b, nblk,      \
                       iters);                                                 \
    CHK(hipEventRecord(e1));                                                   \
    CHK(hipEventSynchronize(e1));                                              \
    float ms;                                                                  \
    CHK(hipEventElapsedTime(&ms, e0, e1));                                     \
    printf("nblk=%3u variant=%d  %7.2f ns/barrier\n", nblk, V,                \
           ms * 1e6 / iters);                                                  \
  }
    RUN(0) RUN(1) RUN(2) RUN(3)
#undef RUN
  }

  // empty-kernel launch train for reference
  CHK(hipEventRecord(e0));
  for (int i = 0; i < 2000; ++i) {
    hipLaunchKernelGGL(k_empty, dim3(256), dim3(256), 0, 0);
  }
  CHK(hipEventRecord(e1));
  CHK(hipEventSynchronize(e1));
  float ms;
  CHK(hipEventElapsedTime(&ms, e0, e1));
  printf("empty 256-block kernel back-to-back: %7.2f ns each\n", ms * 1e6 / 2000);
  return 0;
}
