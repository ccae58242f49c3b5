// Client-batched 5x5 VALID convolution (LeNet-5 family) as implicit
// GEMM on MFMA — BASELINE configs 2 and 4 (LeNet CIFAR-10 / FedProx
// churn) spend ~50% of GPU time in MIOpen naive_conv_* fallbacks at
// thousands of groups (profiles/fedprox_churn_r01.md); these kernels
// replace them.
//
// Channel counts are tiny (3->6, 6->16), so the GEMM M dimension is a
// single 16-row MFMA fragment (BM=16) and the 4 waves of a workgroup
// split the N (=B*OH*OW) dimension.  VALID convolution means every
// gather address is in bounds — no masks at all; the n -> (b,oh,ow)
// plane offset is a precomputed int32 table (ntab, built host-side
// once per geometry) because the output planes are not powers of two
// (28x28, 10x10).
//
// Views (per client c):
//   fwd   : y[oc][n] = relu?(bias[oc] + sum_k W[oc][k] P[k][n]),
//           k=(ic,dh,dw) over IC*25, P[k][n] = x[ic][b][oh+dh][ow+dw]
//   dgrad : dX[ic][n=(b,ih,iw)] = sum_{k=(oc,dh,dw)} W[oc][ic][24-(5dh+dw)]
//           * dy_pad4[oc][b][ih+dh][iw+dw]      (full correlation)
//   wgrad : dW[oc][k] = sum_q dY[oc][q] P[k][q]  (fp32 out)
//
// K is padded to a BK=32 multiple with zero A-columns (B-side clamps
// the channel index; the zero A rows annihilate the products).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CV5_BM 16
#define CV5_BN 128
#define CV5_BK 32
#define CV5_PAD 8
#define CV5_THREADS 256

struct ConvGeom5 {
  int B, H, W;          // input plane (or padded dy plane for dgrad)
  int OH, OW;           // output plane
  int IC, OC;           // contraction channels / output channels
  int C, tiles_n;
  int K, KP;            // true K and padded-to-32 K
};

__device__ __forceinline__ bool xcd_remap5(int lid, int C, int T,
                                           int& c, int& tile) {
  const int xcd = lid & 7;
  const int s = lid >> 3;
  const int grp = s / T;
  tile = s - grp * T;
  c = xcd + 8 * grp;
  return c < C;
}

// ---------------------------------------------------------------------------
// fwd (RELU templated): grid = xcd_blocks(C, tiles_n)
template <bool RELU>
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_fwd(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    const __hip_bfloat16* __restrict__ bias, __hip_bfloat16* __restrict__ y,
    const int* __restrict__ ntab, ConvGeom5 g) {
  int c, tile;
  if (!xcd_remap5(blockIdx.x, g.C, g.tiles_n, c, tile)) return;
  const int n0 = tile * CV5_BN;
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* a_lds = smem;                        // [16][KP]
  short* bT_lds = smem + 16 * g.KP;           // [2][BN][BK+PAD]
  const int N = g.B * g.OH * g.OW;
  const int HW = g.H * g.W;
  const int64_t planeB = (int64_t)g.B * HW;
  const ushort* xc =
      reinterpret_cast<const ushort*>(x) + (int64_t)c * g.IC * planeB;
  const ushort* wc =
      reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * g.K;
  const __hip_bfloat16* bc = bias + (int64_t)c * g.OC;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  // stage A = W[0:16][0:KP] once (zero padding rows/cols)
  for (int i = threadIdx.x; i < 16 * g.KP; i += CV5_THREADS) {
    int m = i / g.KP, k = i - m * g.KP;
    a_lds[i] = (m < g.OC && k < g.K) ? (short)wc[(int64_t)m * g.K + k]
                                     : (short)0;
  }

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;          // wave w: n-subtiles {2w,2w+1}
  f32x4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int kk = threadIdx.x % CV5_BK;
  const int nn0 = (threadIdx.x / CV5_BK) * (CV5_BN / 8);
  ushort breg[CV5_BN / 8];
  const int LDB = CV5_BK + CV5_PAD;

  auto gather = [&](int k0) {
    const int k = k0 + kk;
    const int kc = min(k, g.K - 1);           // K-tail: A is zero there
    const int ic = kc / 25, r = kc - ic * 25;
    const int dh = r / 5, dw = r - dh * 5;
    const ushort* plane = xc + (int64_t)ic * planeB + dh * g.W + dw;
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j) {
      int n = min(n0 + nn0 + j, N - 1);
      breg[j] = plane[ntab[n]];
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j)
      bT_lds[buf * CV5_BN * LDB + (nn0 + j) * LDB + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < g.KP; k0 += CV5_BK) {
    __syncthreads();
    if (k0 + CV5_BK < g.KP) gather(k0 + CV5_BK);
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(lane & 15) * g.KP + k0 + 8 * (lane >> 4)]);
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      int sub = wave * 2 + t;
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur * CV5_BN * LDB + (sub * 16 + (lane & 15)) * LDB
                  + 8 * (lane >> 4)]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
    if (k0 + CV5_BK < g.KP) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int t = 0; t < 2; ++t) {
    int n = n0 + (wave * 2 + t) * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = (lane >> 4) * 4 + r;
      if (m < g.OC) {
        float v = acc[t][r] + to_f32(bc[m]);
        if (RELU) v = fmaxf(v, 0.f);
        yc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dgrad: dy_pad4 [C][OC][B][OH+8][OW+8]; output plane = (H,W) of x.
// g: B; H,W := padded dy plane dims; OH,OW := dX plane; IC := dX
// channels (M dim), OC := contracted channels; K = OC*25.
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_dgrad(
    const __hip_bfloat16* __restrict__ dyp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, const int* __restrict__ ntab,
    ConvGeom5 g) {
  int c, tile;
  if (!xcd_remap5(blockIdx.x, g.C, g.tiles_n, c, tile)) return;
  const int n0 = tile * CV5_BN;
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* a_lds = smem;                        // [16][KP]
  short* bT_lds = smem + 16 * g.KP;
  const int N = g.B * g.OH * g.OW;            // dX positions
  const int HW = g.H * g.W;                   // padded dy plane
  const int64_t planeB = (int64_t)g.B * HW;
  const ushort* dyc =
      reinterpret_cast<const ushort*>(dyp) + (int64_t)c * g.OC * planeB;
  const ushort* wc =
      reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * g.IC * 25;
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * N;

  // stage A once: A[ic][k=(oc,dh,dw)] = W[oc][ic][24 - (5dh+dw)]
  for (int i = threadIdx.x; i < 16 * g.KP; i += CV5_THREADS) {
    int m = i / g.KP, k = i - m * g.KP;
    short v = 0;
    if (m < g.IC && k < g.K) {
      int oc = k / 25, r = k - oc * 25;
      v = (short)wc[((int64_t)oc * g.IC + m) * 25 + (24 - r)];
    }
    a_lds[i] = v;
  }

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  const int kk = threadIdx.x % CV5_BK;
  const int nn0 = (threadIdx.x / CV5_BK) * (CV5_BN / 8);
  ushort breg[CV5_BN / 8];
  const int LDB = CV5_BK + CV5_PAD;

  auto gather = [&](int k0) {
    const int k = k0 + kk;
    const int kc = min(k, g.K - 1);
    const int oc = kc / 25, r = kc - oc * 25;
    const int dh = r / 5, dw = r - dh * 5;
    const ushort* plane = dyc + (int64_t)oc * planeB + dh * g.W + dw;
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j) {
      int n = min(n0 + nn0 + j, N - 1);
      breg[j] = plane[ntab[n]];
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j)
      bT_lds[buf * CV5_BN * LDB + (nn0 + j) * LDB + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < g.KP; k0 += CV5_BK) {
    __syncthreads();
    if (k0 + CV5_BK < g.KP) gather(k0 + CV5_BK);
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(lane & 15) * g.KP + k0 + 8 * (lane >> 4)]);
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      int sub = wave * 2 + t;
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur * CV5_BN * LDB + (sub * 16 + (lane & 15)) * LDB
                  + 8 * (lane >> 4)]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
    if (k0 + CV5_BK < g.KP) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int t = 0; t < 2; ++t) {
    int n = n0 + (wave * 2 + t) * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = (lane >> 4) * 4 + r;
      if (m < g.IC)
        dxc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[t][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad: dW[oc][k=(ic,dh,dw)] (fp32) = sum_q dY[oc][q] x[ic][..q..];
// A = dY rows direct from global (contiguous, NN%32==0), B = x gather
// with per-column offsets hoisted; per q-step only ntab[q] changes.
// g: H,W := x plane; OH,OW := y plane; K := NN (reduction), KP unused.
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_wgrad(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, const int* __restrict__ ntab, ConvGeom5 g) {
  int c, tile;
  if (!xcd_remap5(blockIdx.x, g.C, g.tiles_n, c, tile)) return;
  const int n0 = tile * CV5_BN;               // over IC*25
  __shared__ short bT_lds[2][CV5_BN * (CV5_BK + CV5_PAD)];
  const int K25 = g.IC * 25;
  const int NN = g.B * g.OH * g.OW;
  const int HW = g.H * g.W;
  const int64_t planeB = (int64_t)g.B * HW;
  const ushort* xc =
      reinterpret_cast<const ushort*>(x) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K25;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = lane & 15;                 // dY row (oc), BM=16
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  const int qq = threadIdx.x % CV5_BK;
  const int nn0 = (threadIdx.x / CV5_BK) * (CV5_BN / 8);
  ushort breg[CV5_BN / 8];
  const int LDB = CV5_BK + CV5_PAD;

  int64_t off[CV5_BN / 8];
#pragma unroll
  for (int j = 0; j < CV5_BN / 8; ++j) {
    int k = min(n0 + nn0 + j, K25 - 1);
    int ic = k / 25, r = k - ic * 25;
    int dh = r / 5, dw2 = r - dh * 5;
    off[j] = (int64_t)ic * planeB + dh * g.W + dw2;
  }

  auto gather = [&](int q0) {
    int q = min(q0 + qq, NN - 1);
    const ushort* base = xc + ntab[q];
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j) breg[j] = base[off[j]];
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV5_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * LDB + qq] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += CV5_BK) {
    __syncthreads();
    if (q0 + CV5_BK < NN) gather(q0 + CV5_BK);
    bf16x8 a;
    {
      uint4 av = *reinterpret_cast<const uint4*>(dyrow + q0 + 8 * (lane >> 4));
      a = *reinterpret_cast<const bf16x8*>(&av);
      if (!arow_ok) {
#pragma unroll
        for (int e = 0; e < 8; ++e) a[e] = 0;
      }
    }
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      int sub = wave * 2 + t;
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(sub * 16 + (lane & 15)) * LDB + 8 * (lane >> 4)]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
    if (q0 + CV5_BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int t = 0; t < 2; ++t) {
    int k = n0 + (wave * 2 + t) * 16 + (lane & 15);
    if (k >= K25) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = (lane >> 4) * 4 + r;
      if (m < g.OC)
        dwc[(int64_t)m * K25 + k] = __float2bfloat16(acc[t][r]);
    }
  }
}

// ---------------------------------------------------------------------------
static inline int cdiv5(int a, int b) { return (a + b - 1) / b; }
static inline int xcd_blocks5(int C, int T) { return ((C + 7) / 8 * 8) * T; }


// ---------------------------------------------------------------------------
// Direct VALU forward.  PMC on the implicit-GEMM fwd: MFMA busy 1.4%,
// 59% SQ_WAIT_ANY + 26% SQ_WAIT_INST — with M = OC <= 16 and
// K = IC*25 <= 400 the MFMA tile shape wastes the matrix cores and the
// kernel is a pure gather machine.  Here one block owns one (client, b)
// image: x plane and the client weights stage in LDS once, and each
// thread accumulates whole output pixels with an unrolled 25-tap loop
// — ~300 TFLOP/s of f32 FMA against ~0.15 TFLOP of work.
template <bool RELU>
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_fwd_direct(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    const __hip_bfloat16* __restrict__ bias, __hip_bfloat16* __restrict__ y,
    ConvGeom5 g) {
  const int c = blockIdx.x / g.B;
  const int b = blockIdx.x - c * g.B;
  const int HW = g.H * g.W, OHW = g.OH * g.OW;
  const int K = g.IC * 25;
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* x_lds = smem;                         // [IC*HW]
  short* w_lds = smem + g.IC * HW;             // [OC*K]
  const ushort* xc = reinterpret_cast<const ushort*>(x)
                     + ((int64_t)c * g.IC * g.B + b) * HW;
  const ushort* wc = reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * K;
  for (int ic = 0; ic < g.IC; ++ic) {
    const ushort* plane = xc + (int64_t)ic * g.B * HW;
    for (int i = threadIdx.x; i < HW; i += CV5_THREADS)
      x_lds[ic * HW + i] = (short)plane[i];
  }
  for (int i = threadIdx.x; i < g.OC * K; i += CV5_THREADS)
    w_lds[i] = (short)wc[i];
  __syncthreads();

  // register blocking: one item = (oc, oh, 4 consecutive ow).  The
  // tail block re-computes overlapping columns (ow0 = OW-4) so every
  // x row read [ow0 .. ow0+7] stays inside the W = OW+4 input row and
  // starts at an even column (u32 LDS reads).  4 independent
  // accumulator chains per thread hide the ds_read latency that made
  // the one-output-per-thread version 3-4x slower than MFMA.
  const int wblk = (g.OW + 3) / 4;
  const int nitems = g.OC * g.OH * wblk;
  __hip_bfloat16* yc = y + ((int64_t)c * g.OC * g.B + b) * OHW;
  for (int it = threadIdx.x; it < nitems; it += CV5_THREADS) {
    const int oc = it / (g.OH * wblk);
    const int p = it - oc * (g.OH * wblk);
    const int oh = p / wblk;
    const int ow0 = min((p - oh * wblk) * 4, g.OW - 4);
    const float bv = to_f32(bias[(int64_t)c * g.OC + oc]);
    float acc0 = bv, acc1 = bv, acc2 = bv, acc3 = bv;
    const short* wr = w_lds + oc * K;
    for (int ic = 0; ic < g.IC; ++ic) {
      const short* xrow = x_lds + ic * HW + oh * g.W + ow0;
      const short* wrow = wr + ic * 25;
#pragma unroll
      for (int dh = 0; dh < 5; ++dh) {
        const uint32_t* xr32 = reinterpret_cast<const uint32_t*>(
            xrow + dh * g.W);
        float f[8];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const uint32_t v = xr32[u];
          ushort lo = (ushort)v, hi = (ushort)(v >> 16);
          f[2 * u] = to_f32(*reinterpret_cast<__hip_bfloat16*>(&lo));
          f[2 * u + 1] = to_f32(*reinterpret_cast<__hip_bfloat16*>(&hi));
        }
#pragma unroll
        for (int dw = 0; dw < 5; ++dw) {
          ushort wv = (ushort)wrow[dh * 5 + dw];
          const float wf = to_f32(*reinterpret_cast<__hip_bfloat16*>(&wv));
          acc0 += f[dw] * wf;
          acc1 += f[dw + 1] * wf;
          acc2 += f[dw + 2] * wf;
          acc3 += f[dw + 3] * wf;
        }
      }
    }
    if (RELU) {
      acc0 = fmaxf(acc0, 0.f); acc1 = fmaxf(acc1, 0.f);
      acc2 = fmaxf(acc2, 0.f); acc3 = fmaxf(acc3, 0.f);
    }
    __hip_bfloat16* yr = yc + (int64_t)oc * g.B * OHW + oh * g.OW + ow0;
    yr[0] = __float2bfloat16(acc0);
    yr[1] = __float2bfloat16(acc1);
    yr[2] = __float2bfloat16(acc2);
    yr[3] = __float2bfloat16(acc3);
  }
}

extern "C" void ols_conv5x5_fwd(const void* x, const void* w, const void* b,
                                void* y, const int* ntab, int C, int IC,
                                int OC, int B, int H, int W, int relu,
                                hipStream_t stream) {
  ConvGeom5 g;
  g.B = B; g.H = H; g.W = W; g.OH = H - 4; g.OW = W - 4;
  g.IC = IC; g.OC = OC; g.C = C;
  g.K = IC * 25; g.KP = cdiv5(g.K, CV5_BK) * CV5_BK;
  g.tiles_n = cdiv5(B * g.OH * g.OW, CV5_BN);
  // direct register-blocked VALU kernel: the MFMA implicit GEMM has
  // M = OC <= 16 and K <= 400 (PMC: MFMA busy 1.4%) and is a pure
  // gather machine; the 4-output-per-thread direct kernel measured
  // fedprox 52.0 vs 60.8 ms/round and lenet 8.4 vs 9.2.  (The naive
  // one-output-per-thread version was 3-4x SLOWER — single dependent
  // ds_read chain; the register blocking is what wins.)
  // OLSIM_CONV5=mfma restores the MFMA path.
  const size_t direct_lds = ((size_t)IC * H * W + (size_t)OC * g.K)
                            * sizeof(short);
  const char* c5 = getenv("OLSIM_CONV5");
  if (direct_lds <= 32768 && g.OW >= 4 && W % 2 == 0
      && (c5 == nullptr || c5[0] != 'm')) {
    dim3 gridd((unsigned)((int64_t)C * B));
    if (relu)
      hipLaunchKernelGGL((k_conv5x5_fwd_direct<true>), gridd,
                         dim3(CV5_THREADS), direct_lds, stream,
                         (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                         (const __hip_bfloat16*)b, (__hip_bfloat16*)y, g);
    else
      hipLaunchKernelGGL((k_conv5x5_fwd_direct<false>), gridd,
                         dim3(CV5_THREADS), direct_lds, stream,
                         (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                         (const __hip_bfloat16*)b, (__hip_bfloat16*)y, g);
    return;
  }
  size_t lds = (16 * g.KP + 2 * CV5_BN * (CV5_BK + CV5_PAD)) * sizeof(short);
  dim3 grid(xcd_blocks5(C, g.tiles_n));
  if (relu)
    hipLaunchKernelGGL((k_conv5x5_fwd<true>), grid, dim3(CV5_THREADS), lds,
                       stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                       (__hip_bfloat16*)y, ntab, g);
  else
    hipLaunchKernelGGL((k_conv5x5_fwd<false>), grid, dim3(CV5_THREADS), lds,
                       stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                       (__hip_bfloat16*)y, ntab, g);
}

// H, W: the dX plane; dyp is [C][OC][B][OH+8][OW+8] with OH=H-4

// Direct register-blocked dgrad: same structure as the direct fwd
// (the dgrad is a full correlation of dy_pad4 with the flipped
// weights, contraction over OC) — the flip happens while staging the
// weights into LDS.  Geometry as in ols_conv5x5_dgrad: g.H/g.W = the
// padded dy plane, g.OH/g.OW = the dX plane.
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_dgrad_direct(
    const __hip_bfloat16* __restrict__ dyp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom5 g) {
  const int c = blockIdx.x / g.B;
  const int b = blockIdx.x - c * g.B;
  const int HW = g.H * g.W;                    // padded dy plane
  const int OHW = g.OH * g.OW;                 // dX plane
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* d_lds = smem;                         // [OC*HW]
  short* w_lds = smem + g.OC * HW;             // [OC*IC*25], pre-flipped
  const ushort* dc = reinterpret_cast<const ushort*>(dyp)
                     + ((int64_t)c * g.OC * g.B + b) * HW;
  const ushort* wc = reinterpret_cast<const ushort*>(w)
                     + (int64_t)c * g.OC * g.IC * 25;
  for (int oc = 0; oc < g.OC; ++oc) {
    const ushort* plane = dc + (int64_t)oc * g.B * HW;
    for (int i = threadIdx.x; i < HW; i += CV5_THREADS)
      d_lds[oc * HW + i] = (short)plane[i];
  }
  for (int i = threadIdx.x; i < g.OC * g.IC * 25; i += CV5_THREADS) {
    const int oi = i / 25, r = i - oi * 25;
    w_lds[i] = (short)wc[oi * 25 + (24 - r)];
  }
  __syncthreads();

  const int wblk = (g.OW + 3) / 4;
  const int nitems = g.IC * g.OH * wblk;
  __hip_bfloat16* xc = dx + ((int64_t)c * g.IC * g.B + b) * OHW;
  for (int it = threadIdx.x; it < nitems; it += CV5_THREADS) {
    const int ic = it / (g.OH * wblk);
    const int p = it - ic * (g.OH * wblk);
    const int ih = p / wblk;
    const int iw0 = min((p - ih * wblk) * 4, g.OW - 4);
    float acc0 = 0.f, acc1 = 0.f, acc2 = 0.f, acc3 = 0.f;
    for (int oc = 0; oc < g.OC; ++oc) {
      const short* drow = d_lds + oc * HW + ih * g.W + iw0;
      const short* wrow = w_lds + (oc * g.IC + ic) * 25;
#pragma unroll
      for (int dh = 0; dh < 5; ++dh) {
        const uint32_t* dr32 = reinterpret_cast<const uint32_t*>(
            drow + dh * g.W);
        float f[8];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const uint32_t v = dr32[u];
          ushort lo = (ushort)v, hi = (ushort)(v >> 16);
          f[2 * u] = to_f32(*reinterpret_cast<__hip_bfloat16*>(&lo));
          f[2 * u + 1] = to_f32(*reinterpret_cast<__hip_bfloat16*>(&hi));
        }
#pragma unroll
        for (int dw = 0; dw < 5; ++dw) {
          ushort wv = (ushort)wrow[dh * 5 + dw];
          const float wf = to_f32(*reinterpret_cast<__hip_bfloat16*>(&wv));
          acc0 += f[dw] * wf;
          acc1 += f[dw + 1] * wf;
          acc2 += f[dw + 2] * wf;
          acc3 += f[dw + 3] * wf;
        }
      }
    }
    __hip_bfloat16* xr = xc + (int64_t)ic * g.B * OHW + ih * g.OW + iw0;
    xr[0] = __float2bfloat16(acc0);
    xr[1] = __float2bfloat16(acc1);
    xr[2] = __float2bfloat16(acc2);
    xr[3] = __float2bfloat16(acc3);
  }
}

extern "C" void ols_conv5x5_dgrad(const void* dyp, const void* w, void* dx,
                                  const int* ntab, int C, int IC, int OC,
                                  int B, int H, int W, hipStream_t stream) {
  ConvGeom5 g;
  g.B = B; g.H = (H - 4) + 8; g.W = (W - 4) + 8;   // padded dy plane
  g.OH = H; g.OW = W;                              // dX plane
  g.IC = IC; g.OC = OC; g.C = C;
  g.K = OC * 25; g.KP = cdiv5(g.K, CV5_BK) * CV5_BK;
  g.tiles_n = cdiv5(B * H * W, CV5_BN);
  // direct register-blocked kernel (see fwd): OLSIM_CONV5=mfma falls
  // back to the implicit GEMM
  const size_t direct_lds = ((size_t)OC * g.H * g.W
                             + (size_t)OC * IC * 25) * sizeof(short);
  const char* c5 = getenv("OLSIM_CONV5");
  if (direct_lds <= 32768 && g.OW >= 4 && g.OW % 2 == 0 && g.W % 2 == 0
      && (c5 == nullptr || c5[0] != 'm')) {
    dim3 gridd((unsigned)((int64_t)C * B));
    hipLaunchKernelGGL(k_conv5x5_dgrad_direct, gridd, dim3(CV5_THREADS),
                       direct_lds, stream, (const __hip_bfloat16*)dyp,
                       (const __hip_bfloat16*)w, (__hip_bfloat16*)dx, g);
    return;
  }
  size_t lds = (16 * g.KP + 2 * CV5_BN * (CV5_BK + CV5_PAD)) * sizeof(short);
  dim3 grid(xcd_blocks5(C, g.tiles_n));
  hipLaunchKernelGGL(k_conv5x5_dgrad, grid, dim3(CV5_THREADS), lds, stream,
                     (const __hip_bfloat16*)dyp, (const __hip_bfloat16*)w,
                     (__hip_bfloat16*)dx, ntab, g);
}



// Two-stage direct wgrad (default): stage 1 computes per-(client, b)
// fp32 tap partials — grid C*B keeps the chip full and each tap's
// q-loop is only OH*OW deep (the single-stage per-client version
// measured ~2x slower: C-only grid + 1600-deep serial chains).
// Stage 2 reduces the B partials and stores bf16.
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_wgrad_part(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    float* __restrict__ part, ConvGeom5 g) {
  const int c = blockIdx.x / g.B;
  const int b = blockIdx.x - c * g.B;
  const int HW = g.H * g.W, OHW = g.OH * g.OW;
  const int K = g.IC * 25;
  const int ntaps = g.OC * K;
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* x_lds = smem;                          // [IC*HW]
  short* d_lds = smem + g.IC * HW;              // [OC*OHW]
  const ushort* xc = reinterpret_cast<const ushort*>(x)
                     + ((int64_t)c * g.IC * g.B + b) * HW;
  const ushort* dc = reinterpret_cast<const ushort*>(dy)
                     + ((int64_t)c * g.OC * g.B + b) * OHW;
  for (int ic = 0; ic < g.IC; ++ic) {
    const ushort* plane = xc + (int64_t)ic * g.B * HW;
    for (int i = threadIdx.x; i < HW; i += CV5_THREADS)
      x_lds[ic * HW + i] = (short)plane[i];
  }
  for (int oc = 0; oc < g.OC; ++oc) {
    const ushort* plane = dc + (int64_t)oc * g.B * OHW;
    for (int i = threadIdx.x; i < OHW; i += CV5_THREADS)
      d_lds[oc * OHW + i] = (short)plane[i];
  }
  __syncthreads();
  float* pout = part + ((int64_t)c * g.B + b) * ntaps;
  for (int tap = threadIdx.x; tap < ntaps; tap += CV5_THREADS) {
    const int oc = tap / K;
    const int r = tap - oc * K;
    const int ic = r / 25, rr = r - ic * 25;
    const int dh = rr / 5, dw2 = rr - dh * 5;
    const short* dr = d_lds + oc * OHW;
    const short* xr = x_lds + ic * HW + dh * g.W + dw2;
    float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
    for (int oh = 0; oh < g.OH; ++oh) {
      const short* drow = dr + oh * g.OW;
      const short* xrow = xr + oh * g.W;
      int ow = 0;
      for (; ow + 4 <= g.OW; ow += 4) {
        ushort d0 = (ushort)drow[ow], x0 = (ushort)xrow[ow];
        ushort d1 = (ushort)drow[ow + 1], x1 = (ushort)xrow[ow + 1];
        ushort d2 = (ushort)drow[ow + 2], x2 = (ushort)xrow[ow + 2];
        ushort d3 = (ushort)drow[ow + 3], x3 = (ushort)xrow[ow + 3];
        p0 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d0))
              * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x0));
        p1 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d1))
              * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x1));
        p2 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d2))
              * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x2));
        p3 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d3))
              * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x3));
      }
      for (; ow < g.OW; ++ow) {
        ushort d0 = (ushort)drow[ow], x0 = (ushort)xrow[ow];
        p0 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d0))
              * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x0));
      }
    }
    pout[tap] = (p0 + p1) + (p2 + p3);
  }
}

__global__ __launch_bounds__(OLS_THREADS) void k_conv5x5_wgrad_reduce(
    const float* __restrict__ part, __hip_bfloat16* __restrict__ dw,
    int64_t C, int B, int ntaps) {
  const int64_t total = C * ntaps;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t u = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       u < total; u += stride) {
    const int64_t c = u / ntaps;
    const int tap = (int)(u - c * ntaps);
    const float* p = part + c * B * (int64_t)ntaps + tap;
    float acc = 0.f;
    for (int b = 0; b < B; ++b) acc += p[(int64_t)b * ntaps];
    dw[u] = __float2bfloat16(acc);
  }
}

// Direct wgrad: one block per CLIENT; x and dy stage per-b into LDS
// and each thread owns whole dW taps, accumulating over all B images
// with 4 q-strided partial sums (independent FMA chains).  Output is
// bf16 like the MFMA kernel (fp32 accumulation in registers).
__global__ __launch_bounds__(CV5_THREADS) void k_conv5x5_wgrad_direct(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom5 g) {
  const int c = blockIdx.x;
  const int HW = g.H * g.W, OHW = g.OH * g.OW;
  const int K = g.IC * 25;
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* x_lds = smem;                          // [IC*HW]
  short* d_lds = smem + g.IC * HW;              // [OC*OHW]
  const ushort* xc = reinterpret_cast<const ushort*>(x)
                     + (int64_t)c * g.IC * g.B * HW;
  const ushort* dc = reinterpret_cast<const ushort*>(dy)
                     + (int64_t)c * g.OC * g.B * OHW;
  const int ntaps = g.OC * K;
  // per-thread tap accumulators: ntaps <= 2400 -> <= 10 taps/thread
  float acc[10];
  const int mytaps = (ntaps - (int)threadIdx.x + CV5_THREADS - 1)
                     / CV5_THREADS;
#pragma unroll
  for (int i = 0; i < 10; ++i) acc[i] = 0.f;

  for (int b = 0; b < g.B; ++b) {
    __syncthreads();
    for (int ic = 0; ic < g.IC; ++ic) {
      const ushort* plane = xc + ((int64_t)ic * g.B + b) * HW;
      for (int i = threadIdx.x; i < HW; i += CV5_THREADS)
        x_lds[ic * HW + i] = (short)plane[i];
    }
    for (int oc = 0; oc < g.OC; ++oc) {
      const ushort* plane = dc + ((int64_t)oc * g.B + b) * OHW;
      for (int i = threadIdx.x; i < OHW; i += CV5_THREADS)
        d_lds[oc * OHW + i] = (short)plane[i];
    }
    __syncthreads();
    for (int t = 0; t < mytaps; ++t) {
      const int tap = (int)threadIdx.x + t * CV5_THREADS;
      const int oc = tap / K;
      const int r = tap - oc * K;
      const int ic = r / 25, rr = r - ic * 25;
      const int dh = rr / 5, dw2 = rr - dh * 5;
      const short* dr = d_lds + oc * OHW;
      const short* xr = x_lds + ic * HW + dh * g.W + dw2;
      float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
      for (int oh = 0; oh < g.OH; ++oh) {
        const short* drow = dr + oh * g.OW;
        const short* xrow = xr + oh * g.W;
        int ow = 0;
        for (; ow + 4 <= g.OW; ow += 4) {
          ushort d0 = (ushort)drow[ow], x0 = (ushort)xrow[ow];
          ushort d1 = (ushort)drow[ow + 1], x1 = (ushort)xrow[ow + 1];
          ushort d2 = (ushort)drow[ow + 2], x2 = (ushort)xrow[ow + 2];
          ushort d3 = (ushort)drow[ow + 3], x3 = (ushort)xrow[ow + 3];
          p0 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d0))
                * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x0));
          p1 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d1))
                * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x1));
          p2 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d2))
                * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x2));
          p3 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d3))
                * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x3));
        }
        for (; ow < g.OW; ++ow) {
          ushort d0 = (ushort)drow[ow], x0 = (ushort)xrow[ow];
          p0 += to_f32(*reinterpret_cast<__hip_bfloat16*>(&d0))
                * to_f32(*reinterpret_cast<__hip_bfloat16*>(&x0));
        }
      }
      acc[t] += (p0 + p1) + (p2 + p3);
    }
  }
  __hip_bfloat16* dwc = dw + (int64_t)c * ntaps;
  for (int t = 0; t < mytaps; ++t)
    dwc[(int)threadIdx.x + t * CV5_THREADS] = __float2bfloat16(acc[t]);
}

extern "C" void ols_conv5x5_wgrad(const void* x, const void* dy, void* dw,
                                  float* part, const int* ntab, int C,
                                  int IC, int OC, int B, int H, int W,
                                  hipStream_t stream) {
  ConvGeom5 g;
  g.B = B; g.H = H; g.W = W; g.OH = H - 4; g.OW = W - 4;
  g.IC = IC; g.OC = OC; g.C = C;
  g.K = 0; g.KP = 0;
  g.tiles_n = cdiv5(IC * 25, CV5_BN);
  if (part != nullptr) {
    // two-stage direct path: binding allocated the [C, B, ntaps] fp32
    // partial buffer after checking the same LDS/shape conditions
    const size_t lds1 = ((size_t)IC * H * W
                         + (size_t)OC * g.OH * g.OW) * sizeof(short);
    hipLaunchKernelGGL(k_conv5x5_wgrad_part, dim3((unsigned)((int64_t)C * B)),
                       dim3(CV5_THREADS), lds1, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       part, g);
    const int ntaps = OC * IC * 25;
    hipLaunchKernelGGL(k_conv5x5_wgrad_reduce,
                       dim3(ols_grid((int64_t)C * ntaps, OLS_THREADS)),
                       dim3(OLS_THREADS), 0, stream, part,
                       (__hip_bfloat16*)dw, (int64_t)C, B, ntaps);
    return;
  }
  // direct per-client kernel: MEASURED NEGATIVE (fedprox 97.5 vs 49.2
  // ms/round) — one block per client is only C blocks (grid too small
  // at C<=6250 for 256 CUs x many waves) and each tap's 1600-deep
  // q-loop is FMA-latency serial.  The MFMA implicit GEMM keeps wgrad
  // (K = B*OH*OW = 1600 is the one conv5 direction with a real GEMM
  // shape).  Kept behind OLSIM_CONV5=direct for the record.
  const size_t direct_lds = ((size_t)IC * H * W
                             + (size_t)OC * g.OH * g.OW) * sizeof(short);
  const char* c5 = getenv("OLSIM_CONV5");
  if (direct_lds <= 32768 && OC * IC * 25 <= 10 * CV5_THREADS
      && c5 != nullptr && c5[0] == 'd') {
    hipLaunchKernelGGL(k_conv5x5_wgrad_direct, dim3((unsigned)C),
                       dim3(CV5_THREADS), direct_lds, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       (__hip_bfloat16*)dw, g);
    return;
  }
  dim3 grid(xcd_blocks5(C, g.tiles_n));
  hipLaunchKernelGGL(k_conv5x5_wgrad, grid, dim3(CV5_THREADS), 0, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                     (__hip_bfloat16*)dw, ntab, g);
}
