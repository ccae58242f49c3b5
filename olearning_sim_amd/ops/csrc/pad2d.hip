// Single-pass constant zero-pad of the trailing two dims:
// out[p][r][c] = in[p][r-PAD][c-PAD] inside, 0 on the halo.
//
// torch's constant_pad_nd runs TWO passes (a fill of the whole output
// then the strided interior copy) through the generic 5-D indexing
// kernels.  The conv padded-gather path (client_conv2.hip) pads every
// activation and every dgrad dy, ~175 GB/round of the flagship, so this
// kernel has to run at the write roofline: one thread emits one aligned
// u32 (two bf16), each block owns a GROUP of consecutive planes (so
// blocks do thousands of stores, not one, and the index math is small
// u32 div/mod per plane-local offset instead of a 64-bit div/mod on the
// global linear index).
//
// Requires W % 2 == 0 (the wrapper falls back to F.pad otherwise).

#include "common.h"

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_pad2d(
    const T* __restrict__ in, T* __restrict__ out, int64_t planes,
    int H, int W, int pad, int group) {
  const int Hp = H + 2 * pad, Wp = W + 2 * pad;
  const uint32_t planeu = (uint32_t)Hp * (uint32_t)(Wp / 2);
  const int64_t p0 = (int64_t)blockIdx.x * group;
  const uint32_t nu = (uint32_t)(min((int64_t)group, planes - p0) * planeu);
  const T* __restrict__ ing = in + p0 * H * W;
  T* __restrict__ outg = out + p0 * Hp * Wp;
  for (uint32_t idx = threadIdx.x; idx < nu; idx += blockDim.x) {
    const uint32_t pl = idx / planeu;             // small u32 divs: the
    const uint32_t rem = idx - pl * planeu;       // divisors are plane-
    const uint32_t r = rem / (uint32_t)(Wp / 2);  // local (< 2^16)
    const int c0 = (int)(rem - r * (uint32_t)(Wp / 2)) * 2;
    const int rr = (int)r - pad;
    T v0 = from_f32<T>(0.f), v1 = v0;
    if (rr >= 0 && rr < H) {
      const T* row = ing + ((int64_t)pl * H + rr) * W - pad;
      const int ca = c0, cb = c0 + 1;        // padded cols
      if (ca >= pad && ca < W + pad) v0 = row[ca];
      if (cb >= pad && cb < W + pad) v1 = row[cb];
    }
    ushort2 pk;
    pk.x = *reinterpret_cast<ushort*>(&v0);
    pk.y = *reinterpret_cast<ushort*>(&v1);
    *reinterpret_cast<ushort2*>(
        &outg[((int64_t)pl * Hp + r) * Wp + c0]) = pk;
  }
}

extern "C" void ols_pad2d(const void* in, void* out, int64_t planes, int H,
                          int W, int pad, int dtype, hipStream_t stream) {
  const int Wp = W + 2 * pad;
  const uint32_t planeu = (uint32_t)(H + 2 * pad) * (uint32_t)(Wp / 2);
  // ~4K u32 stores per block: enough work to amortise the launch,
  // grids in the 100K range for the big activations (vs one-store
  // blocks from a flat grid over 10^8+ elements)
  int group = (int)(4096 / planeu);
  if (group < 1) group = 1;
  dim3 grid((unsigned)((planes + group - 1) / group)), block(OLS_THREADS);
  if (dtype == 1)
    hipLaunchKernelGGL((k_pad2d<__hip_bfloat16>), grid, block, 0, stream,
                       (const __hip_bfloat16*)in, (__hip_bfloat16*)out,
                       planes, H, W, pad, group);
  else
    hipLaunchKernelGGL((k_pad2d<__half>), grid, block, 0, stream,
                       (const __half*)in, (__half*)out, planes, H, W,
                       pad, group);   // 2-byte elements only (bf16/fp16)
}
