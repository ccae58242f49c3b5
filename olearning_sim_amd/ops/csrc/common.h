// Shared helpers for the olearning_sim_amd gfx950 kernels.
//
// Design notes (see /root/repo/ARCHITECTURE.md):
// - every kernel here is bandwidth-bound elementwise/reduction work, so
//   the rules that matter are: 16-byte-per-lane vector IO, grids well
//   past 256 workgroups (8 XCDs x 32 CUs), fp32 accumulation for
//   reductions, wave64 shuffles for row reductions.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE 64
#define OLS_THREADS 256

static inline int ols_grid(int64_t work_items, int per_block) {
  int64_t g = (work_items + per_block - 1) / per_block;
  if (g < 1) g = 1;
  if (g > 1073741824L) g = 1073741824L;
  return (int)g;
}

// float <-> storage conversions ------------------------------------------
__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ float to_f32(__half x) { return __half2float(x); }

template <typename T> __device__ __forceinline__ T from_f32(float x);
template <> __device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <> __device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float x) {
  return __float2half(x);
}

// 16-byte packs -----------------------------------------------------------
template <typename T, int V>
struct __align__(16) Pack {
  T v[V];
};

// binary search: largest b with offs[b] <= x < offs[b+1]; offs has n+1
// entries. The table is tiny (#params + 1) and L2-resident.
__device__ __forceinline__ int find_block(const int64_t* offs, int nblocks,
                                          int64_t x) {
  int lo = 0, hi = nblocks - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (offs[mid] <= x) lo = mid; else hi = mid - 1;
  }
  return lo;
}

// wave64 butterfly reductions --------------------------------------------
__device__ __forceinline__ float wave_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

__device__ __forceinline__ float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_xor(x, off, WAVE);
  return x;
}
