// Fused 2x2/2 max-pool forward + backward for the client-batched LeNet
// path.  torch's generic nchw max_pool kernels measured ~8x (fwd) and
// ~15x (bwd) off the memory roofline on these shapes and were ~8% of a
// FedProx round (profiles/fedprox_conv5_r02.md); these are plain
// streaming kernels: fwd saves a 2-bit argmax per output element, bwd
// writes all four input positions per output (one carries the grad),
// so the backward needs no atomics and no pre-zeroing pass.
//
// Layout: [N, H, W] planes (N = C*ch*B for the [C, ch, B, H, W]
// activations — pooling is per-plane, the caller flattens).  H and W
// even.

#include "common.h"

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_pool2x2_fwd(
    const T* __restrict__ x, T* __restrict__ y,
    unsigned char* __restrict__ arg, int64_t n_out, int OH, int OW) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int W = OW * 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += stride) {
    const int64_t plane = i / (OH * OW);
    const int rem = (int)(i - plane * (OH * OW));
    const int oi = rem / OW, oj = rem - (rem / OW) * OW;
    const T* p = x + (plane * (2 * OH) + 2 * oi) * W + 2 * oj;
    float v00 = to_f32(p[0]), v01 = to_f32(p[1]);
    float v10 = to_f32(p[W]), v11 = to_f32(p[W + 1]);
    float m0 = fmaxf(v00, v01), m1 = fmaxf(v10, v11);
    float m = fmaxf(m0, m1);
    unsigned char a;
    if (m == v00) a = 0;
    else if (m == v01) a = 1;
    else if (m == v10) a = 2;
    else a = 3;
    y[i] = from_f32<T>(m);
    arg[i] = a;
  }
}

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_pool2x2_bwd(
    const T* __restrict__ dy, const unsigned char* __restrict__ arg,
    T* __restrict__ dx, int64_t n_out, int OH, int OW) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int W = OW * 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += stride) {
    const int64_t plane = i / (OH * OW);
    const int rem = (int)(i - plane * (OH * OW));
    const int oi = rem / OW, oj = rem - (rem / OW) * OW;
    T* p = dx + (plane * (2 * OH) + 2 * oi) * W + 2 * oj;
    const T g = dy[i];
    const unsigned char a = arg[i];
    const T z = from_f32<T>(0.f);
    p[0] = a == 0 ? g : z;
    p[1] = a == 1 ? g : z;
    p[W] = a == 2 ? g : z;
    p[W + 1] = a == 3 ? g : z;
  }
}

extern "C" void ols_pool2x2_fwd(const void* x, void* y, unsigned char* arg,
                                int64_t planes, int OH, int OW, int dtype,
                                hipStream_t stream) {
  const int64_t n_out = planes * OH * OW;
  dim3 grid(ols_grid(n_out, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_pool2x2_fwd<__hip_bfloat16>), grid, block, 0,
                       stream, (const __hip_bfloat16*)x, (__hip_bfloat16*)y,
                       arg, n_out, OH, OW);
  else
    hipLaunchKernelGGL((k_pool2x2_fwd<float>), grid, block, 0, stream,
                       (const float*)x, (float*)y, arg, n_out, OH, OW);
}

extern "C" void ols_pool2x2_bwd(const void* dy, const unsigned char* arg,
                                void* dx, int64_t planes, int OH, int OW,
                                int dtype, hipStream_t stream) {
  const int64_t n_out = planes * OH * OW;
  dim3 grid(ols_grid(n_out, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_pool2x2_bwd<__hip_bfloat16>), grid, block, 0,
                       stream, (const __hip_bfloat16*)dy, arg,
                       (__hip_bfloat16*)dx, n_out, OH, OW);
  else
    hipLaunchKernelGGL((k_pool2x2_bwd<float>), grid, block, 0, stream,
                       (const float*)dy, arg, (float*)dx, n_out, OH, OW);
}

// ---------------------------------------------------------------------------
// 2x stride subsample (the 1x1-conv downsample shortcut): fwd picks the
// top-left of every 2x2; bwd writes the whole input once (grad at the
// picked positions, zero elsewhere) — torch's slice backward is a full
// zero-fill plus a strided scatter through the generic 5-D kernels.
// One thread handles an aligned pair of output columns (fwd) or input
// columns (bwd); W even.

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_subsample2_fwd(
    const T* __restrict__ x, T* __restrict__ y, int64_t n_out2, int OH,
    int OW) {
  // n_out2 = planes * OH * (OW/2); each thread emits 2 output cols
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int W = OW * 2, OW2 = OW / 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out2; i += stride) {
    const int64_t plane = i / (OH * OW2);
    const int rem = (int)(i - plane * (OH * OW2));
    const int oi = rem / OW2, oj2 = (rem - (rem / OW2) * OW2) * 2;
    const T* p = x + (plane * (2 * OH) + 2 * oi) * (int64_t)W + 2 * oj2;
    Pack<T, 2> o;
    o.v[0] = p[0];
    o.v[1] = p[2];
    *reinterpret_cast<Pack<T, 2>*>(
        &y[(plane * OH + oi) * (int64_t)OW + oj2]) = o;
  }
}

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_subsample2_bwd(
    const T* __restrict__ dy, T* __restrict__ dx, int64_t n_in2, int H,
    int W) {
  // n_in2 = planes * H * (W/2); each thread writes 2 input cols
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int W2 = W / 2, OW = W / 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_in2; i += stride) {
    const int64_t plane = i / (H * W2);
    const int rem = (int)(i - plane * (H * W2));
    const int r = rem / W2, c = (rem - (rem / W2) * W2) * 2;
    Pack<T, 2> o;
    o.v[1] = from_f32<T>(0.f);
    o.v[0] = ((r & 1) == 0)
                 ? dy[(plane * (H / 2) + (r >> 1)) * (int64_t)OW + (c >> 1)]
                 : from_f32<T>(0.f);
    *reinterpret_cast<Pack<T, 2>*>(&dx[(plane * H + r) * (int64_t)W + c]) = o;
  }
}

extern "C" void ols_subsample2_fwd(const void* x, void* y, int64_t planes,
                                   int OH, int OW, int dtype,
                                   hipStream_t stream) {
  const int64_t n2 = planes * OH * (OW / 2);
  dim3 grid(ols_grid(n2, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_subsample2_fwd<__hip_bfloat16>), grid, block, 0,
                       stream, (const __hip_bfloat16*)x, (__hip_bfloat16*)y,
                       n2, OH, OW);
  else
    hipLaunchKernelGGL((k_subsample2_fwd<float>), grid, block, 0, stream,
                       (const float*)x, (float*)y, n2, OH, OW);
}

extern "C" void ols_subsample2_bwd(const void* dy, void* dx, int64_t planes,
                                   int H, int W, int dtype,
                                   hipStream_t stream) {
  const int64_t n2 = planes * H * (W / 2);
  dim3 grid(ols_grid(n2, OLS_THREADS)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_subsample2_bwd<__hip_bfloat16>), grid, block, 0,
                       stream, (const __hip_bfloat16*)dy, (__hip_bfloat16*)dx,
                       n2, H, W);
  else
    hipLaunchKernelGGL((k_subsample2_bwd<float>), grid, block, 0, stream,
                       (const float*)dy, (float*)dx, n2, H, W);
}
