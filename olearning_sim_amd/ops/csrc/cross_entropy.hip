// Fused cross-entropy: one pass computes the per-row loss AND the
// gradient dlogits = (softmax(x) - onehot(y)) / N, so the backward pass
// costs only a scalar scale.  Rows are the flattened clients x batch of
// a client-batched model (models/base.py loss); K ranges from 10
// (LeNet) to 30k+ (BERT vocab).
//
// One workgroup per row; two passes over the row held to registers via
// grid-stride chunks: (1) max+sumexp with wave reductions + LDS
// cross-wave combine, (2) write dlogits.  bf16/f32 in, fp32 math.

#include "common.h"

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_ce_fwd_bwd(
    const T* __restrict__ logits, const int64_t* __restrict__ labels,
    float* __restrict__ loss, T* __restrict__ dlogits, int64_t nrows,
    int64_t K, float inv_n) {
  __shared__ float red[OLS_THREADS / WAVE];
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int nw = blockDim.x / WAVE;

  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const T* x = logits + row * K;
    T* dx = dlogits + row * K;
    const int64_t y = labels[row];

    float m = -3.4e38f;
    for (int64_t k = tid; k < K; k += blockDim.x)
      m = fmaxf(m, to_f32(x[k]));
    m = wave_max(m);
    if (nw > 1) {
      if ((tid & (WAVE - 1)) == 0) red[wid] = m;
      __syncthreads();
      m = red[0];
      for (int w = 1; w < nw; ++w) m = fmaxf(m, red[w]);
      __syncthreads();
    }

    float s = 0.f;
    for (int64_t k = tid; k < K; k += blockDim.x)
      s += __expf(to_f32(x[k]) - m);
    s = wave_sum(s);
    if (nw > 1) {
      if ((tid & (WAVE - 1)) == 0) red[wid] = s;
      __syncthreads();
      s = red[0];
      for (int w = 1; w < nw; ++w) s += red[w];
      __syncthreads();
    }
    const float inv_s = 1.f / s;

    for (int64_t k = tid; k < K; k += blockDim.x) {
      float p = __expf(to_f32(x[k]) - m) * inv_s;
      dx[k] = from_f32<T>((p - (k == y ? 1.f : 0.f)) * inv_n);
    }
    if (tid == 0)
      loss[row] = (m - to_f32(x[y])) + __logf(s);
  }
}

extern "C" void ols_cross_entropy_fwd_bwd(
    const void* logits, const int64_t* labels, float* loss, void* dlogits,
    int64_t nrows, int64_t K, float inv_n, int dtype, hipStream_t stream) {
  int threads = K >= 512 ? OLS_THREADS : WAVE;
  dim3 grid(ols_grid(nrows, 1) < 16384 ? (int)nrows : 16384);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((k_ce_fwd_bwd<float>), grid, dim3(threads), 0,
                         stream, (const float*)logits, labels, loss,
                         (float*)dlogits, nrows, K, inv_n);
      break;
    case 1:
      hipLaunchKernelGGL((k_ce_fwd_bwd<__hip_bfloat16>), grid, dim3(threads),
                         0, stream, (const __hip_bfloat16*)logits, labels,
                         loss, (__hip_bfloat16*)dlogits, nrows, K, inv_n);
      break;
    default:
      hipLaunchKernelGGL((k_ce_fwd_bwd<__half>), grid, dim3(threads), 0,
                         stream, (const __half*)logits, labels, loss,
                         (__half*)dlogits, nrows, K, inv_n);
  }
}
