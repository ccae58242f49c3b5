// Weighted client-delta reduction: the FedAvg aggregation the reference
// hands to an external service via its Pulsar outbound
// (ols_core/deviceflow/non_grpc/dispatcher.py:47-63) runs here as one
// on-GPU reduction over the co-resident clients:
//
//   delta[j] += sum_c w[c] * (buf[c,j] - master[j])
//            =  sum_c w[c] * buf[blk_base + c*n + q]  -  W * master[j]
//
// (W = sum_c w[c], hoisted).  Parallel over master elements j; for a
// fixed c, consecutive threads read consecutive addresses, so each step
// of the c-loop is a fully coalesced stream; client weights are staged
// in LDS once per workgroup.  fp32 accumulation regardless of the
// replica dtype.

#include "common.h"

#define MAX_LDS_W 4096

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_delta_accum(
    float* __restrict__ delta, const T* __restrict__ buf,
    const T* __restrict__ master, const float* __restrict__ weights,
    const int64_t* __restrict__ offs, int nblocks, int64_t clients,
    int64_t pglobal, float wsum) {
  __shared__ float w_lds[MAX_LDS_W];
  const bool w_in_lds = clients <= MAX_LDS_W;
  if (w_in_lds) {
    for (int c = threadIdx.x; c < clients; c += blockDim.x)
      w_lds[c] = weights[c];
    __syncthreads();
  }
  const float* w = w_in_lds ? w_lds : weights;

  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       j < pglobal; j += stride) {
    int b = find_block(offs, nblocks, j);
    int64_t n = offs[b + 1] - offs[b];
    int64_t q = j - offs[b];
    const T* col = buf + clients * offs[b] + q;
    float acc = 0.f;
    for (int64_t c = 0; c < clients; ++c)
      acc += w[c] * to_f32(col[c * n]);
    delta[j] += acc - wsum * to_f32(master[j]);
  }
}

// Vector variant for the common single-param call (nblocks == 1,
// n % 8 == 0): 8 consecutive j per thread, so each c-step of a wave
// streams 1 KiB contiguous (16 B/lane) instead of 128 B of scalar bf16
// loads — the scalar form measured ~8x off the HBM roofline at
// C=1250 (profiles/resnet_v6_r02.md, k_delta_accum 41 ms/round).
template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_delta_accum_v8(
    float* __restrict__ delta, const T* __restrict__ buf,
    const T* __restrict__ master, const float* __restrict__ weights,
    int64_t clients, int64_t n, float wsum) {
  __shared__ float w_lds[MAX_LDS_W];
  const bool w_in_lds = clients <= MAX_LDS_W;
  if (w_in_lds) {
    for (int c = threadIdx.x; c < clients; c += blockDim.x)
      w_lds[c] = weights[c];
    __syncthreads();
  }
  const float* w = w_in_lds ? w_lds : weights;

  const int64_t nv = n / 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       v < nv; v += stride) {
    const int64_t j = v * 8;
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    const T* col = buf + j;
    for (int64_t c = 0; c < clients; ++c) {
      Pack<T, 8> p = *reinterpret_cast<const Pack<T, 8>*>(&col[c * n]);
      const float wc = w[c];
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += wc * to_f32(p.v[e]);
    }
    Pack<T, 8> pm = *reinterpret_cast<const Pack<T, 8>*>(&master[j]);
#pragma unroll
    for (int e = 0; e < 8; ++e)
      delta[j + e] += acc[e] - wsum * to_f32(pm.v[e]);
  }
}

// Client-parallel variant for SMALL params: the j-parallel kernels
// above give a parameter with n=64 only 8 active lanes, each walking a
// C-deep serial latency chain (a 64-element GroupNorm gain at C=1250
// measured ~250 us — pure latency).  Here the grid also spans client
// tiles; each block reduces its tile's partial and atomicAdd's fp32
// into delta; the -wsum*master term is applied by the c-tile-0 blocks.
template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_delta_accum_cpar(
    float* __restrict__ delta, const T* __restrict__ buf,
    const T* __restrict__ master, const float* __restrict__ weights,
    int64_t clients, int64_t n, float wsum, int ctile) {
  const int64_t nv = n / 8;
  const int64_t j8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x);
  if (j8 >= nv) return;
  const int64_t j = j8 * 8;
  const int64_t c0 = (int64_t)blockIdx.y * ctile;
  const int64_t c1 = min(clients, c0 + ctile);
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  const T* col = buf + j;
  for (int64_t c = c0; c < c1; ++c) {
    Pack<T, 8> p = *reinterpret_cast<const Pack<T, 8>*>(&col[c * n]);
    const float wc = weights[c];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] += wc * to_f32(p.v[e]);
  }
  if (blockIdx.y == 0) {
    Pack<T, 8> pm = *reinterpret_cast<const Pack<T, 8>*>(&master[j]);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] -= wsum * to_f32(pm.v[e]);
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(&delta[j + e], acc[e]);
}

// Scalar flavour of the client-parallel kernel for small params whose
// n is not a multiple of 8 (e.g. a 450-element LeNet conv1.w at
// C=6250 measured ~1.5 ms on the j-parallel kernel: 450 lanes walking
// a 6250-deep serial latency chain).
template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_delta_accum_cpar_s(
    float* __restrict__ delta, const T* __restrict__ buf,
    const T* __restrict__ master, const float* __restrict__ weights,
    int64_t clients, int64_t n, float wsum, int ctile) {
  const int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= n) return;
  const int64_t c0 = (int64_t)blockIdx.y * ctile;
  const int64_t c1 = min(clients, c0 + ctile);
  float acc = 0.f;
  for (int64_t c = c0; c < c1; ++c)
    acc += weights[c] * to_f32(buf[c * n + j]);
  if (blockIdx.y == 0) acc -= wsum * to_f32(master[j]);
  atomicAdd(&delta[j], acc);
}

extern "C" void ols_weighted_delta_accum_flat(
    float* delta, const void* buf, const void* master, const float* weights,
    const int64_t* offs, int nblocks, int64_t clients, int64_t pglobal,
    float wsum, int dtype, hipStream_t stream) {
  dim3 block(OLS_THREADS);
  if (nblocks == 1 && pglobal % 8 != 0 && dtype == 1 &&
      pglobal < 131072 && clients >= 64) {
    const int ctile = 64;
    dim3 grid((unsigned)((pglobal + OLS_THREADS - 1) / OLS_THREADS),
              (unsigned)((clients + ctile - 1) / ctile));
    hipLaunchKernelGGL((k_delta_accum_cpar_s<__hip_bfloat16>), grid, block,
                       0, stream, delta, (const __hip_bfloat16*)buf,
                       (const __hip_bfloat16*)master, weights, clients,
                       pglobal, wsum, ctile);
    return;
  }
  if (nblocks == 1 && pglobal % 8 == 0 && dtype == 1 &&
      pglobal / 8 < 16384 && clients >= 64) {
    // small param: client-parallel tiles (fills the chip; the j-only
    // kernel would be a serial latency chain on a handful of lanes)
    const int ctile = 64;
    dim3 grid((unsigned)((pglobal / 8 + OLS_THREADS - 1) / OLS_THREADS),
              (unsigned)((clients + ctile - 1) / ctile));
    hipLaunchKernelGGL((k_delta_accum_cpar<__hip_bfloat16>), grid, block, 0,
                       stream, delta, (const __hip_bfloat16*)buf,
                       (const __hip_bfloat16*)master, weights, clients,
                       pglobal, wsum, ctile);
    return;
  }
  if (nblocks == 1 && pglobal % 8 == 0 && dtype != 0) {
    dim3 grid(ols_grid(pglobal / 8, OLS_THREADS));
    if (dtype == 1)
      hipLaunchKernelGGL((k_delta_accum_v8<__hip_bfloat16>), grid, block, 0,
                         stream, delta, (const __hip_bfloat16*)buf,
                         (const __hip_bfloat16*)master, weights, clients,
                         pglobal, wsum);
    else
      hipLaunchKernelGGL((k_delta_accum_v8<__half>), grid, block, 0, stream,
                         delta, (const __half*)buf, (const __half*)master,
                         weights, clients, pglobal, wsum);
    return;
  }
  dim3 grid(ols_grid(pglobal, OLS_THREADS));
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((k_delta_accum<float>), grid, block, 0, stream,
                         delta, (const float*)buf, (const float*)master,
                         weights, offs, nblocks, clients, pglobal, wsum);
      break;
    case 1:
      hipLaunchKernelGGL((k_delta_accum<__hip_bfloat16>), grid, block, 0,
                         stream, delta, (const __hip_bfloat16*)buf,
                         (const __hip_bfloat16*)master, weights, offs,
                         nblocks, clients, pglobal, wsum);
      break;
    default:
      hipLaunchKernelGGL((k_delta_accum<__half>), grid, block, 0, stream,
                         delta, (const __half*)buf, (const __half*)master,
                         weights, offs, nblocks, clients, pglobal, wsum);
  }
}
