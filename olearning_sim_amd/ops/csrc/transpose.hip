// Batched 2-D transpose: out[b][n][m] = in[b][m][n] (bf16/fp32).
//
// torch's transpose+contiguous runs the generic strided copy at
// 0.6-1.7 TB/s on the BERT shapes (profiles/bert_tiedhead_r02.md:
// ~87 ms/round of aten::copy_ is transposes — the tied-head tok^T is
// 19 ms alone).  Classic LDS-tiled transpose: 64x64 tiles, 16-byte
// reads AND writes, padded LDS rows; edge tiles take the scalar path.

#include "common.h"

#define TR_TILE 64
#define TR_PAD 8           // shorts; keeps write-side b128 reads off one bank

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_transpose2d(
    const T* __restrict__ in, T* __restrict__ out, int M, int N) {
  __shared__ T tile[TR_TILE][TR_TILE + TR_PAD];
  const int64_t b = blockIdx.z;
  const int m0 = blockIdx.y * TR_TILE;
  const int n0 = blockIdx.x * TR_TILE;
  const T* src = in + b * (int64_t)M * N;
  T* dst = out + b * (int64_t)M * N;

  const bool full = (m0 + TR_TILE <= M) && (n0 + TR_TILE <= N);
  if (full) {
    // 256 threads x 8 elems = 2048 per pass; 2 passes read the tile
    const int c0 = (threadIdx.x % 8) * 8;
    const int r = threadIdx.x / 8;             // 0..31
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      Pack<T, 8> v = *reinterpret_cast<const Pack<T, 8>*>(
          &src[(int64_t)(m0 + r + 32 * p) * N + n0 + c0]);
      // contiguous LDS row span: one vector store (the per-element
      // loop measured the kernel instruction-bound at ~1.25 TB/s)
      *reinterpret_cast<Pack<T, 8>*>(&tile[r + 32 * p][c0]) = v;
    }
    __syncthreads();
    // write rows of OUT (= columns of the tile), 16 B per store
    const int rn = threadIdx.x / 8;            // out row within tile
    const int cm0 = (threadIdx.x % 8) * 8;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      Pack<T, 8> v;
#pragma unroll
      for (int e = 0; e < 8; ++e) v.v[e] = tile[cm0 + e][rn + 32 * p];
      *reinterpret_cast<Pack<T, 8>*>(
          &dst[(int64_t)(n0 + rn + 32 * p) * M + m0 + cm0]) = v;
    }
    return;
  }
  // edge tile: scalar
  for (int i = threadIdx.x; i < TR_TILE * TR_TILE; i += blockDim.x) {
    int r = i / TR_TILE, c = i % TR_TILE;
    int m = m0 + r, n = n0 + c;
    if (m < M && n < N)
      dst[(int64_t)n * M + m] = src[(int64_t)m * N + n];
  }
}


// Thin-matrix path (M <= 32, M % 8 == 0): the 64x64 tile kernel would
// route every block through its scalar edge path (measured 0.39 TB/s on
// the [C, 16, features] FC activations of the LeNet family).  One block
// owns a [M x 128] column chunk: coalesced Pack8 row loads into LDS,
// coalesced Pack8 stores of the transposed rows.
#define TRT_NC 128
#define TRT_PAD 8

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_transpose_thin(
    const T* __restrict__ in, T* __restrict__ out, int M, int N) {
  __shared__ T tile[TRT_NC][32 + TRT_PAD];
  const int64_t b = blockIdx.z;
  const int n0 = blockIdx.x * TRT_NC;
  const int nn = min(TRT_NC, N - n0);
  const T* src = in + b * (int64_t)M * N;
  T* dst = out + b * (int64_t)M * N;
  const int mv = M / 8;

  if ((nn % 8) == 0) {
    // load: pack p covers in[m][n0+c0 .. +8)
    for (int i = threadIdx.x; i < M * (nn / 8); i += blockDim.x) {
      const int m = i / (nn / 8);
      const int c0 = (i - m * (nn / 8)) * 8;
      Pack<T, 8> v = *reinterpret_cast<const Pack<T, 8>*>(
          &src[(int64_t)m * N + n0 + c0]);
#pragma unroll
      for (int e = 0; e < 8; ++e) tile[c0 + e][m] = v.v[e];
    }
  } else {
    for (int i = threadIdx.x; i < M * nn; i += blockDim.x) {
      const int m = i / nn, c = i - m * nn;
      tile[c][m] = src[(int64_t)m * N + n0 + c];
    }
  }
  __syncthreads();
  // store: out[b][n0+nl][mp*8 ..) — consecutive threads, consecutive
  // 16-byte stores
  for (int i = threadIdx.x; i < nn * mv; i += blockDim.x) {
    const int nl = i / mv;
    const int mp = (i - nl * mv) * 8;
    Pack<T, 8> v;
#pragma unroll
    for (int e = 0; e < 8; ++e) v.v[e] = tile[nl][mp + e];
    *reinterpret_cast<Pack<T, 8>*>(&dst[(int64_t)(n0 + nl) * M + mp]) = v;
  }
}

extern "C" void ols_transpose2d(const void* in, void* out, int64_t B, int M,
                                int N, int dtype, hipStream_t stream) {
  dim3 block(OLS_THREADS);
  if (M <= 32 && M % 8 == 0 && B <= 65535) {
    dim3 grid((N + TRT_NC - 1) / TRT_NC, 1, (unsigned)B);
    if (dtype == 1)
      hipLaunchKernelGGL((k_transpose_thin<__hip_bfloat16>), grid, block, 0,
                         stream, (const __hip_bfloat16*)in,
                         (__hip_bfloat16*)out, M, N);
    else
      hipLaunchKernelGGL((k_transpose_thin<float>), grid, block, 0, stream,
                         (const float*)in, (float*)out, M, N);
    return;
  }
  dim3 grid((N + TR_TILE - 1) / TR_TILE, (M + TR_TILE - 1) / TR_TILE,
            (unsigned)B);
  if (dtype == 1)
    hipLaunchKernelGGL((k_transpose2d<__hip_bfloat16>), grid, block, 0,
                       stream, (const __hip_bfloat16*)in,
                       (__hip_bfloat16*)out, M, N);
  else
    hipLaunchKernelGGL((k_transpose2d<float>), grid, block, 0, stream,
                       (const float*)in, (float*)out, M, N);
}
