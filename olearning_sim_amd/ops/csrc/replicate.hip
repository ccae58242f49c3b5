// Broadcast kernels for the per-round client-replica setup.
//
// 1. k_replicate: dst[C, n] = src[n].  torch's expand().clone() runs the
//    stride-0 broadcast through TensorIterator's unvectorised copy path —
//    measured 0.39 TB/s on the flagship (aten::copy_, ~65 ms/round at
//    C=1250; profiles/resnet_optable_r02).  Here every lane issues 16 B
//    stores and the [n] source stays L2-resident, so the write side runs
//    at HBM roofline.
// 2. k_synth_batch: out_bf16[C*B, n] = x_f32[b, n] + s*y[row] + t — the
//    synthetic-data class-conditional shift (engine/data.py) fused with
//    the dtype cast, replacing a stride-0 fp32 broadcast-add plus a
//    second full-size cast copy.
//
// Both require n % 8 == 0 (the callers' parameter/activation rows are);
// the Python wrappers fall back to composed torch ops otherwise.

#include "common.h"

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_replicate(
    const T* __restrict__ src, T* __restrict__ dst, int64_t clients,
    int64_t nv) {
  const int64_t total = clients * nv;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t u = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       u < total; u += stride) {
    const int64_t off = u % nv;
    const Pack<T, 8> p =
        *reinterpret_cast<const Pack<T, 8>*>(&src[off * 8]);
    *reinterpret_cast<Pack<T, 8>*>(&dst[u * 8]) = p;
  }
}

__global__ __launch_bounds__(OLS_THREADS) void k_synth_batch(
    const float* __restrict__ x, const int64_t* __restrict__ y,
    __hip_bfloat16* __restrict__ out, int64_t rows, int64_t batch,
    int64_t nv, float s, float t) {
  const int64_t total = rows * nv;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t u = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       u < total; u += stride) {
    const int64_t row = u / nv;
    const int64_t off = u - row * nv;
    const float add = s * (float)y[row] + t;
    const Pack<float, 4> a = *reinterpret_cast<const Pack<float, 4>*>(
        &x[((row % batch) * nv + off) * 8]);
    const Pack<float, 4> b = *reinterpret_cast<const Pack<float, 4>*>(
        &x[((row % batch) * nv + off) * 8 + 4]);
    Pack<__hip_bfloat16, 8> po;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      po.v[e] = from_f32<__hip_bfloat16>(a.v[e] + add);
      po.v[4 + e] = from_f32<__hip_bfloat16>(b.v[e] + add);
    }
    *reinterpret_cast<Pack<__hip_bfloat16, 8>*>(&out[u * 8]) = po;
  }
}

extern "C" void ols_replicate(const void* src, void* dst, int64_t clients,
                              int64_t n, int dtype, hipStream_t stream) {
  const int64_t nv = n / 8;
  dim3 grid(ols_grid(clients * nv, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_replicate<__hip_bfloat16>), grid, block, 0, stream,
                       (const __hip_bfloat16*)src, (__hip_bfloat16*)dst,
                       clients, nv);
  else
    hipLaunchKernelGGL((k_replicate<float>), grid, block, 0, stream,
                       (const float*)src, (float*)dst, clients, nv);
}

extern "C" void ols_synth_batch(const float* x, const int64_t* y, void* out,
                                int64_t rows, int64_t batch, int64_t n,
                                float s, float t, hipStream_t stream) {
  const int64_t nv = n / 8;
  dim3 grid(ols_grid(rows * nv, OLS_THREADS)), block(OLS_THREADS);
  hipLaunchKernelGGL(k_synth_batch, grid, block, 0, stream, x, y,
                     (__hip_bfloat16*)out, rows, batch, nv, s, t);
}


// dy * (y > 0) in one pass (the composed torch chain is a bool compare
// tensor + a cast + a multiply — three extra full-size passes on the
// conv5 ReLU backward path).
template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_relu_mask(
    const T* __restrict__ dy, const T* __restrict__ y, T* __restrict__ out,
    int64_t nv) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t u = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       u < nv; u += stride) {
    Pack<T, 8> pd = *reinterpret_cast<const Pack<T, 8>*>(&dy[u * 8]);
    Pack<T, 8> py = *reinterpret_cast<const Pack<T, 8>*>(&y[u * 8]);
    Pack<T, 8> po;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      po.v[e] = to_f32(py.v[e]) > 0.f ? pd.v[e] : from_f32<T>(0.f);
    *reinterpret_cast<Pack<T, 8>*>(&out[u * 8]) = po;
  }
}

extern "C" void ols_relu_mask(const void* dy, const void* y, void* out,
                              int64_t n, int dtype, hipStream_t stream) {
  const int64_t nv = n / 8;
  dim3 grid(ols_grid(nv, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_relu_mask<__hip_bfloat16>), grid, block, 0, stream,
                       (const __hip_bfloat16*)dy, (const __hip_bfloat16*)y,
                       (__hip_bfloat16*)out, nv);
  else
    hipLaunchKernelGGL((k_relu_mask<float>), grid, block, 0, stream,
                       (const float*)dy, (const float*)y, (float*)out, nv);
}
