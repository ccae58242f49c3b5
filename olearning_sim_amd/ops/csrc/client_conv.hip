// Client-batched 3x3 convolution as implicit GEMM on MFMA — the core
// compute kernel of the simulator: every co-resident virtual client
// convolves with ITS OWN filters, so the client dimension is a grid
// dimension and each workgroup computes one (client, M-tile, N-tile)
// output block on `v_mfma_f32_16x16x32_bf16` matrix cores with
// LDS-staged tiles.  Replaces MIOpen grouped conv (which falls back to
// per-group GEMM loops / naive kernels at thousands of groups).
//
// Activation layout: [C, ch, B, H, W] ("client-channel-first"): for a
// fixed (client, channel) the [B, H, W] block is contiguous, so the
// implicit-GEMM N dimension n=(b, oh, ow) walks nearly contiguous
// memory and output writes are coalesced.
//
// GEMM views (per client, pad=1, stride s, kernel 3x3):
//   fwd   : Y[oc][n]    = sum_k  W[oc][k]      * P[k][n]
//           k=(ic,dh,dw), P[k][n] = x[ic][b][oh*s+dh-1][ow*s+dw-1]
//   dgrad : dX[ic][n]   = sum_k  W'[ic][k]     * Q[k][n]
//           k=(oc,dh,dw), W'[ic][(oc,dh,dw)] = W[oc][ic][2-dh][2-dw],
//           Q[k][n] = dY[oc][b][oh][ow] where oh=(h+1-dh')/s if exact
//   wgrad : dW[oc][k]   = sum_n dY[oc][n] * P[k][n]   (K = n = B*OH*OW)
//
// Tiles: BM=64 (4 waves x 16 rows), BN=64, BK=32 (one MFMA K per step),
// 256 threads.  Gathered operands are staged through LDS; gathers are
// the price of implicit im2col — they read each input element 9 times
// through L2 instead of materialising a 9x patch matrix in HBM.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ short bf16_bits(float v) {
  __hip_bfloat16 h = from_f32<__hip_bfloat16>(v);
  return *reinterpret_cast<short*>(&h);
}

#define CONV_BM 64
#define CONV_BN 64
#define CONV_BK 32
#define CONV_THREADS 256

// fragment maps for v_mfma_f32_16x16x32_bf16 (cdna_hip_programming.md §3):
//   A[row][k]: lane l holds rows row=l&15, k = 8*(l>>4) + e   (e=0..7)
//   B[k][col]: lane l holds col=l&15,      k = 8*(l>>4) + e
//   C/D:       lane l, reg r -> row = (l>>4)*4 + r, col = l&15

// ---------------------------------------------------------------------------
// Tiny layout self-test: D[16][16] = A[16][32] x B[32][16] for one
// workgroup of 64 threads — verifies the fragment maps on hardware.
__global__ __launch_bounds__(64) void k_mfma_selftest(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ B,
    float* __restrict__ D) {
  int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    a[e] = *reinterpret_cast<const short*>(&A[(l & 15) * 32 + 8 * (l >> 4) + e]);
    b[e] = *reinterpret_cast<const short*>(&B[(8 * (l >> 4) + e) * 16 + (l & 15)]);
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

// ---------------------------------------------------------------------------
// shared gather helpers

struct ConvGeom {
  int B, H, W, OH, OW, IC, OC, stride;
  // log2 of the spatial sizes when all are powers of two (CIFAR-style
  // 32/16/8/4 planes) — lets the implicit-GEMM gather decompose n with
  // shifts instead of ~40-instruction integer divisions per element.
  int lg_ow, lg_ohw, pow2;
};

__device__ __forceinline__ void n_decomp(const ConvGeom& g, int n, int& b,
                                         int& oh, int& ow) {
  if (g.pow2) {
    b = n >> g.lg_ohw;
    int q = n & ((1 << g.lg_ohw) - 1);
    oh = q >> g.lg_ow;
    ow = q & ((1 << g.lg_ow) - 1);
  } else {
    int OHW = g.OH * g.OW;
    b = n / OHW;
    int q = n % OHW;
    oh = q / g.OW;
    ow = q % g.OW;
  }
}

// stage the fwd/wgrad patch tile P[k][n] (k in [k0,k0+BK), n in
// [n0,n0+BN)) into lds[BK][BN]; k=(ic,dh,dw) w/ dw fastest.
template <int TILE_K, int TILE_N>
__device__ void stage_patch(const __hip_bfloat16* __restrict__ x,
                            short* lds, const ConvGeom g,
                            int k0, int n0, int kmax, int nmax) {
  const int HW = g.H * g.W;
  const int OHW = g.OH * g.OW;
  // each thread fills (TILE_K*TILE_N)/CONV_THREADS elements, n fastest
  for (int i = threadIdx.x; i < TILE_K * TILE_N; i += CONV_THREADS) {
    int kk = i / TILE_N, nn = i % TILE_N;
    int k = k0 + kk, n = n0 + nn;
    float v = 0.f;
    if (k < kmax && n < nmax) {
      int ic = k / 9, r = k % 9;
      int dh = r / 3, dw = r % 3;
      int b, oh, ow;
      n_decomp(g, n, b, oh, ow);
      int ih = oh * g.stride + dh - 1, iw = ow * g.stride + dw - 1;
      if (ih >= 0 && ih < g.H && iw >= 0 && iw < g.W)
        v = to_f32(x[((int64_t)ic * g.B + b) * HW + ih * g.W + iw]);
    }
    lds[kk * TILE_N + nn] = bf16_bits(v);
  }
}

// ---------------------------------------------------------------------------
// forward: grid (ntiles_n, ntiles_m, C); x[C,IC,B,H,W] w[C,OC,IC,3,3]
// y[C,OC,B,OH,OW]
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom g) {
  __shared__ short a_lds[CONV_BM * CONV_BK];
  __shared__ short b_lds[CONV_BK * CONV_BN];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CONV_BM;
  const int n0 = blockIdx.x * CONV_BN;
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const __hip_bfloat16* xc = x + (int64_t)c * g.IC * g.B * g.H * g.W;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;   // wave owns rows [m0+wave*16, +16)
  f32x4 acc[CONV_BN / 16];
#pragma unroll
  for (int i = 0; i < CONV_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += CONV_BK) {
    // stage A = W[m0.., k0..]
    for (int i = threadIdx.x; i < CONV_BM * CONV_BK; i += CONV_THREADS) {
      int mm = i / CONV_BK, kk = i % CONV_BK;
      int m = m0 + mm, k = k0 + kk;
      a_lds[i] = (m < g.OC && k < K)
                     ? bf16_bits(to_f32(wc[(int64_t)m * K + k])) : (short)0;
    }
    stage_patch<CONV_BK, CONV_BN>(xc, b_lds, g, k0, n0, K, N);
    __syncthreads();

    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(wave * 16 + (lane & 15)) * CONV_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CONV_BN / 16; ++nt) {
      bf16x8 b;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        b[e] = b_lds[(8 * (lane >> 4) + e) * CONV_BN + nt * 16 + (lane & 15)];
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

  // write D: row = m0 + wave*16 + (lane>>4)*4 + r, col = n0 + nt*16 + (lane&15)
#pragma unroll
  for (int nt = 0; nt < CONV_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.OC)
        yc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// dgrad: dX[C,IC,B,H,W] from dY[C,OC,B,OH,OW]; k=(oc,dh,dw),
// A[ic][k] = W[oc][ic][2-dh][2-dw]
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_dgrad(
    const __hip_bfloat16* __restrict__ dy, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom g) {
  __shared__ short a_lds[CONV_BM * CONV_BK];
  __shared__ short b_lds[CONV_BK * CONV_BN];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CONV_BM;           // over IC
  const int n0 = blockIdx.x * CONV_BN;           // over B*H*W
  const int K = g.OC * 9;
  const int N = g.B * g.H * g.W;
  const int OHW = g.OH * g.OW;
  const int HW = g.H * g.W;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * g.B * OHW;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * g.IC * 9;
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CONV_BN / 16];
#pragma unroll
  for (int i = 0; i < CONV_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += CONV_BK) {
    // A[ic][k=(oc,dh,dw)] = W[oc][ic][2-dh][2-dw]
    for (int i = threadIdx.x; i < CONV_BM * CONV_BK; i += CONV_THREADS) {
      int mm = i / CONV_BK, kk = i % CONV_BK;
      int ic = m0 + mm, k = k0 + kk;
      float v = 0.f;
      if (ic < g.IC && k < K) {
        int oc = k / 9, r = k % 9;
        int dh = r / 3, dw = r % 3;
        v = to_f32(wc[((int64_t)oc * g.IC + ic) * 9 + (2 - dh) * 3 + (2 - dw)]);
      }
      a_lds[i] = bf16_bits(v);
    }
    // B[k][n] = dY[oc][b][oh][ow] with h+dh-1 = oh*s (flipped offsets)
    for (int i = threadIdx.x; i < CONV_BK * CONV_BN; i += CONV_THREADS) {
      int kk = i / CONV_BN, nn = i % CONV_BN;
      int k = k0 + kk, n = n0 + nn;
      float v = 0.f;
      if (k < K && n < N) {
        int oc = k / 9, r = k % 9;
        int dh = r / 3, dw = r % 3;
        int b = n / HW, q = n % HW;
        int h = q / g.W, wd = q % g.W;   // dgrad walks input planes (pow2 too but cheap enough here)
        int num_h = h + dh - 1, num_w = wd + dw - 1;
        if (num_h >= 0 && num_w >= 0 && num_h % g.stride == 0
            && num_w % g.stride == 0) {
          int oh = num_h / g.stride, ow = num_w / g.stride;
          if (oh < g.OH && ow < g.OW)
            v = to_f32(dyc[((int64_t)oc * g.B + b) * OHW + oh * g.OW + ow]);
        }
      }
      b_lds[kk * CONV_BN + nn] = bf16_bits(v);
    }
    __syncthreads();

    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(wave * 16 + (lane & 15)) * CONV_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CONV_BN / 16; ++nt) {
      bf16x8 b;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        b[e] = b_lds[(8 * (lane >> 4) + e) * CONV_BN + nt * 16 + (lane & 15)];
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int nt = 0; nt < CONV_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.IC)
        dxc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad: dW[C,OC,IC*9] (fp32 out) = dY[oc][n] x P[k][n] over n.
// grid (ntiles_k9, ntiles_oc, C); K-loop over n.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom g) {
  __shared__ short a_lds[CONV_BM * CONV_BK];   // dY tile [oc][n]
  __shared__ short b_lds[CONV_BK * CONV_BN];   // P^T tile [n][k9]
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CONV_BM;          // over OC
  const int n0 = blockIdx.x * CONV_BN;          // over IC*9
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;             // reduction dim
  const int OHW = g.OH * g.OW;
  const int HW = g.H * g.W;
  const __hip_bfloat16* xc = x + (int64_t)c * g.IC * g.B * HW;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CONV_BN / 16];
#pragma unroll
  for (int i = 0; i < CONV_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int q0 = 0; q0 < NN; q0 += CONV_BK) {
    // A[oc][q] = dY[oc][q0+q]
    for (int i = threadIdx.x; i < CONV_BM * CONV_BK; i += CONV_THREADS) {
      int mm = i / CONV_BK, qq = i % CONV_BK;
      int m = m0 + mm, q = q0 + qq;
      a_lds[i] = (m < g.OC && q < NN)
                     ? bf16_bits(to_f32(dyc[(int64_t)m * NN + q])) : (short)0;
    }
    // B[q][k9] = P[k9][q0+q]
    for (int i = threadIdx.x; i < CONV_BK * CONV_BN; i += CONV_THREADS) {
      int qq = i / CONV_BN, kk = i % CONV_BN;
      int q = q0 + qq, k = n0 + kk;
      float v = 0.f;
      if (q < NN && k < K9) {
        int ic = k / 9, r = k % 9;
        int dh = r / 3, dw2 = r % 3;
        int b, oh, ow;
        n_decomp(g, q, b, oh, ow);
        int ih = oh * g.stride + dh - 1, iw = ow * g.stride + dw2 - 1;
        if (ih >= 0 && ih < g.H && iw >= 0 && iw < g.W)
          v = to_f32(xc[((int64_t)ic * g.B + b) * HW + ih * g.W + iw]);
      }
      b_lds[qq * CONV_BN + kk] = bf16_bits(v);
    }
    __syncthreads();

    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(wave * 16 + (lane & 15)) * CONV_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CONV_BN / 16; ++nt) {
      bf16x8 b;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        b[e] = b_lds[(8 * (lane >> 4) + e) * CONV_BN + nt * 16 + (lane & 15)];
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int nt = 0; nt < CONV_BN / 16; ++nt) {
    int k = n0 + nt * 16 + (lane & 15);
    if (k >= K9) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.OC)
        dwc[(int64_t)m * K9 + k] = __float2bfloat16(acc[nt][r]);
    }
  }
}


// ---------------------------------------------------------------------------
// forward v2: BN=128, B tile stored transposed [BN][BK+4] so a B
// fragment is two aligned ds_read_b64, gathers hoisted per (k-row).
#define CV2_BM 64
#define CV2_BN 128
#define CV2_BK 32
#define CV2_PAD 8
typedef __attribute__((ext_vector_type(4))) short bf16x4;

__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd_v2(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom g) {
  __shared__ short a_lds[CV2_BM * CV2_BK];
  __shared__ short bT_lds[CV2_BN * (CV2_BK + CV2_PAD)];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CV2_BM;
  const int n0 = blockIdx.x * CV2_BN;
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HW = g.H * g.W;
  const int OHW = g.OH * g.OW;
  const __hip_bfloat16* xc = x + (int64_t)c * g.IC * g.B * HW;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CV2_BN / 16];
#pragma unroll
  for (int i = 0; i < CV2_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  // B staging: thread stages row kk, 16 consecutive n; A staging: row amm,
  // 8 consecutive k.  All loads are UNCONDITIONAL (clamped address +
  // multiplicative mask) so they pipeline, and double-buffered in
  // registers: loads for step k0+BK issue before the MFMAs of step k0.
  const int kk = threadIdx.x % CV2_BK;
  const int nn0 = (threadIdx.x / CV2_BK) * (CV2_BN / 8);
  const int amm = threadIdx.x / 4;
  const int ak0 = (threadIdx.x % 4) * 8;

  // hoisted n-decomposition for this thread's 16-column segment
  int nseg[2], ohseg[2], owseg[2];  // segments the 16 cols may span
  // (generic per-element decomposition below; segments unused when
  // OW>=16 — kept simple: recompute per load round, it is ALU not loads)

  short areg[8];
  short breg[CV2_BN / 8];  // 16 shorts; compiler packs pairs

  auto gather = [&](int k0) {
    // A
    int m = m0 + amm;
    const __hip_bfloat16* wrow = wc + (int64_t)min(m, g.OC - 1) * K;
    bool mvalid = m < g.OC;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int k = k0 + ak0 + j;
      float v = to_f32(wrow[min(k, K - 1)]);
      areg[j] = bf16_bits((mvalid && k < K) ? v : 0.f);
    }
    // B
    int k = k0 + kk;
    int kc = min(k, K - 1);
    int ic = kc / 9, r = kc % 9;
    int dh = r / 3, dw = r % 3;
    const __hip_bfloat16* plane = xc + (int64_t)ic * g.B * HW;
    bool kvalid = k < K;
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j) {
      int n = n0 + nn0 + j;
      int nc = min(n, N - 1);
      int b, oh, ow;
      n_decomp(g, nc, b, oh, ow);
      int ih = oh * g.stride + dh - 1, iw = ow * g.stride + dw - 1;
      bool ok = kvalid && n < N && ih >= 0 && ih < g.H && iw >= 0 && iw < g.W;
      int ihc = min(max(ih, 0), g.H - 1), iwc = min(max(iw, 0), g.W - 1);
      float v = to_f32(plane[(int64_t)b * HW + ihc * g.W + iwc]);
      breg[j] = ok ? bf16_bits(v) : (short)0;
    }
  };

  auto commit = [&]() {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      a_lds[amm * CV2_BK + ak0 + j] = areg[j];
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j)
      bT_lds[(nn0 + j) * (CV2_BK + CV2_PAD) + kk] = breg[j];
  };

  gather(0);
  for (int k0 = 0; k0 < K; k0 += CV2_BK) {
    commit();
    __syncthreads();
    if (k0 + CV2_BK < K) gather(k0 + CV2_BK);   // loads overlap the MFMAs
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[(wave * 16 + (lane & 15)) * CV2_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CV2_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[(nt * 16 + (lane & 15)) * (CV2_BK + CV2_PAD)
                  + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int nt = 0; nt < CV2_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.OC)
        yc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}


// ---------------------------------------------------------------------------
// forward v5 — the pipelined variant for the common shapes
// (IC*9 % 32 == 0, OW a power of two >= 16):
//   - A fragments load DIRECTLY from global: each lane reads its own
//     16-byte W row slice per K-step (L2-shared across the client''s
//     n-tile workgroups) — no A staging, no A barrier;
//   - B tile double-buffered in LDS with ONE barrier per K-step: the
//     gather for step k+1 issues before the MFMAs of step k, and the
//     commit of step k+1''s tile happens after them, so global latency
//     hides under compute (PMC on v2: 44% of wave time parked);
//   - the gather hoists the (b, oh) decomposition per 16-column row
//     segment (always within one output row when OW >= 16).
template <bool ROW_HOIST>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd_v5(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom g) {
  __shared__ short bT_lds[2][CV2_BN * (CV2_BK + CV2_PAD)];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CV2_BM;
  const int n0 = blockIdx.x * CV2_BN;
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HW = g.H * g.W;
  const __hip_bfloat16* xc = x + (int64_t)c * g.IC * g.B * HW;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);     // this lane's W row
  const __hip_bfloat16* wrow = wc + (int64_t)min(arow, g.OC - 1) * K;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV2_BN / 16];
#pragma unroll
  for (int i = 0; i < CV2_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV2_BK;
  const int nn0 = (threadIdx.x / CV2_BK) * (CV2_BN / 8);
  short breg[CV2_BN / 8];

  auto gather = [&](int k0) {
    int k = k0 + kk;                       // k < K by construction
    int ic = k / 9, r = k % 9;
    int dh = r / 3, dw = r % 3;
    const __hip_bfloat16* plane = xc + (int64_t)ic * g.B * HW;
    if (ROW_HOIST) {
      // the whole segment lies inside ONE output row: hoist (b, oh)
      int n = n0 + nn0;
      // clamp b: the unconditional clamped-address load must stay in
      // bounds even for fully-masked (n >= N) tail segments
      int b = min(n >> g.lg_ohw, g.B - 1);
      int q = n & ((1 << g.lg_ohw) - 1);
      int oh = q >> g.lg_ow;
      int ow0 = q & ((1 << g.lg_ow) - 1);
      int ih = oh * g.stride + dh - 1;
      bool row_ok = (ih >= 0) && (ih < g.H) && (n < N);
      const __hip_bfloat16* row =
          plane + (int64_t)b * HW + (int64_t)max(0, min(ih, g.H - 1)) * g.W;
#pragma unroll
      for (int j = 0; j < CV2_BN / 8; ++j) {
        int iw = (ow0 + j) * g.stride + dw - 1;
        bool ok = row_ok && iw >= 0 && iw < g.W;
        float v = to_f32(row[max(0, min(iw, g.W - 1))]);
        breg[j] = ok ? bf16_bits(v) : (short)0;
      }
    } else {
      // small planes (OW 4/8): per-element shift decomposition under
      // the same pipeline
#pragma unroll
      for (int j = 0; j < CV2_BN / 8; ++j) {
        int n = n0 + nn0 + j;
        int nc = min(n, N - 1);
        int b = nc >> g.lg_ohw;
        int q = nc & ((1 << g.lg_ohw) - 1);
        int oh = q >> g.lg_ow;
        int ow = q & ((1 << g.lg_ow) - 1);
        int ih = oh * g.stride + dh - 1, iw = ow * g.stride + dw - 1;
        bool ok = n < N && ih >= 0 && ih < g.H && iw >= 0 && iw < g.W;
        float v = to_f32(plane[(int64_t)b * HW
                               + (int64_t)max(0, min(ih, g.H - 1)) * g.W
                               + max(0, min(iw, g.W - 1))]);
        breg[j] = ok ? bf16_bits(v) : (short)0;
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV2_BK + CV2_PAD) + kk] = breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV2_BK) {
    __syncthreads();                       // buf[cur] ready for everyone
    if (k0 + CV2_BK < K) gather(k0 + CV2_BK);   // loads fly over the MFMAs
    // A fragment direct from global (16 B per lane; k0 8-aligned)
    bf16x8 a;
    {
      uint4 av = *reinterpret_cast<const uint4*>(wrow + k0 + 8 * (lane >> 4));
      a = *reinterpret_cast<const bf16x8*>(&av);
      if (!arow_ok) {
#pragma unroll
        for (int e = 0; e < 8; ++e) a[e] = 0;
      }
    }
#pragma unroll
    for (int nt = 0; nt < CV2_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV2_BK + CV2_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV2_BK < K) commit(cur ^ 1);  // waits the gather's loads
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV2_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.OC)
        yc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}


// ---------------------------------------------------------------------------
// dgrad v5: same pipeline as fwd v5 — A (flipped W) double-buffered in
// LDS, B (dy gather) register-double-buffered, ONE barrier per K-step.
template <int STRIDE>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_dgrad_v5(
    const __hip_bfloat16* __restrict__ dy, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom g) {
  __shared__ short a_lds[2][CV2_BM * CV2_BK];
  __shared__ short bT_lds[2][CV2_BN * (CV2_BK + CV2_PAD)];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CV2_BM;            // over IC
  const int n0 = blockIdx.x * CV2_BN;            // over B*H*W
  const int K = g.OC * 9;
  const int N = g.B * g.H * g.W;
  const int OHW = g.OH * g.OW;
  const int HW = g.H * g.W;
  const int lg_w = g.lg_ow + (STRIDE == 2 ? 1 : 0);   // log2(W) = log2(OW*s)
  const int lg_hw = lg_w * 2;                          // H == W planes
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * g.B * OHW;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * g.IC * 9;
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CV2_BN / 16];
#pragma unroll
  for (int i = 0; i < CV2_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV2_BK;
  const int nn0 = (threadIdx.x / CV2_BK) * (CV2_BN / 8);
  const int amm = threadIdx.x / 4;               // A staging row (ic)
  const int ak0 = (threadIdx.x % 4) * 8;
  short breg[CV2_BN / 8];
  short areg[8];

  auto gather = [&](int k0) {
    {
      int ic = m0 + amm;
      int icc = min(ic, g.IC - 1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int k = k0 + ak0 + j;                    // < K by construction
        int oc = k / 9, r2 = k % 9;
        int dh = r2 / 3, dw2 = r2 % 3;
        float v = to_f32(wc[((int64_t)oc * g.IC + icc) * 9
                            + (2 - dh) * 3 + (2 - dw2)]);
        areg[j] = (ic < g.IC) ? bf16_bits(v) : (short)0;
      }
    }
    {
      int k = k0 + kk;
      int oc = k / 9, r2 = k % 9;
      int dh = r2 / 3, dw2 = r2 % 3;
      const __hip_bfloat16* plane = dyc + (int64_t)oc * g.B * OHW;
      // the segment lies inside ONE input row when W >= seg width
      int n = n0 + nn0;
      int b = min(n >> lg_hw, g.B - 1);   // masked tails stay in bounds
      int q = n & ((1 << lg_hw) - 1);
      int h = q >> lg_w;
      int w0 = q & ((1 << lg_w) - 1);
      int num_h = h + dh - 1;
      bool row_ok = (n < N) && num_h >= 0 && num_h % STRIDE == 0
                    && (num_h / STRIDE) < g.OH;
      const __hip_bfloat16* row = plane
          + (int64_t)b * OHW
          + (int64_t)max(0, min(num_h / STRIDE, g.OH - 1)) * g.OW;
      if (g.W >= CV2_BN / 8) {
#pragma unroll
        for (int j = 0; j < CV2_BN / 8; ++j) {
          int num_w = w0 + j + dw2 - 1;
          bool ok = row_ok && num_w >= 0 && num_w % STRIDE == 0
                    && (num_w / STRIDE) < g.OW;
          float v = to_f32(row[max(0, min(num_w / STRIDE, g.OW - 1))]);
          breg[j] = ok ? bf16_bits(v) : (short)0;
        }
      } else {
#pragma unroll
        for (int j = 0; j < CV2_BN / 8; ++j) {
          int nj = min(n0 + nn0 + j, N - 1);
          int bj = min(nj >> lg_hw, g.B - 1);
          int qj = nj & ((1 << lg_hw) - 1);
          int hj = qj >> lg_w;
          int wj = qj & ((1 << lg_w) - 1);
          int nh = hj + dh - 1, nw = wj + dw2 - 1;
          bool ok = (n0 + nn0 + j) < N && nh >= 0 && nw >= 0
                    && nh % STRIDE == 0 && nw % STRIDE == 0
                    && (nh / STRIDE) < g.OH && (nw / STRIDE) < g.OW;
          float v = to_f32(plane[(int64_t)bj * OHW
              + (int64_t)max(0, min(nh / STRIDE, g.OH - 1)) * g.OW
              + max(0, min(nw / STRIDE, g.OW - 1))]);
          breg[j] = ok ? bf16_bits(v) : (short)0;
        }
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      a_lds[buf][amm * CV2_BK + ak0 + j] = areg[j];
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV2_BK + CV2_PAD) + kk] = breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV2_BK) {
    __syncthreads();
    if (k0 + CV2_BK < K) gather(k0 + CV2_BK);
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[cur][(wave * 16 + (lane & 15)) * CV2_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CV2_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV2_BK + CV2_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV2_BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV2_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.IC)
        dxc[(int64_t)m * N + n] = from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v5: A = dY rows DIRECT from global (contiguous, like fwd's W);
// B = patch tile register-double-buffered; one barrier per q-step.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad_v5(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom g) {
  __shared__ short bT_lds[2][CV2_BN * (CV2_BK + CV2_PAD)];
  const int c = blockIdx.z;
  const int m0 = blockIdx.y * CV2_BM;          // over OC
  const int n0 = blockIdx.x * CV2_BN;          // over IC*9
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;            // reduction dim (pow2 x B)
  const int OHW = g.OH * g.OW;
  const int HW = g.H * g.W;
  const __hip_bfloat16* xc = x + (int64_t)c * g.IC * g.B * HW;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);   // dY row (oc)
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV2_BN / 16];
#pragma unroll
  for (int i = 0; i < CV2_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int qq = threadIdx.x % CV2_BK;
  const int nn0 = (threadIdx.x / CV2_BK) * (CV2_BN / 8);
  short breg[CV2_BN / 8];

  auto gather = [&](int q0) {
    int q = q0 + qq;
    int qc = min(q, NN - 1);
    int b = qc >> g.lg_ohw;
    int p = qc & ((1 << g.lg_ohw) - 1);
    int oh = p >> g.lg_ow;
    int ow = p & ((1 << g.lg_ow) - 1);
    bool q_ok = q < NN;
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j) {
      int k = n0 + nn0 + j;
      float v = 0.f;
      bool ok = q_ok && k < K9;
      int kc = min(k, K9 - 1);
      int ic = kc / 9, r2 = kc % 9;
      int dh = r2 / 3, dw2 = r2 % 3;
      int ih = oh * g.stride + dh - 1, iw = ow * g.stride + dw2 - 1;
      ok = ok && ih >= 0 && ih < g.H && iw >= 0 && iw < g.W;
      v = to_f32(xc[((int64_t)ic * g.B + b) * HW
                    + (int64_t)max(0, min(ih, g.H - 1)) * g.W
                    + max(0, min(iw, g.W - 1))]);
      breg[j] = ok ? bf16_bits(v) : (short)0;
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV2_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV2_BK + CV2_PAD) + qq] = breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += CV2_BK) {
    __syncthreads();
    if (q0 + CV2_BK < NN) gather(q0 + CV2_BK);
    bf16x8 a;
    {
      // NN is B * pow2 — multiple of 32, so q0 + 8*(l>>4) is 8-aligned
      uint4 av = *reinterpret_cast<const uint4*>(dyrow + q0 + 8 * (lane >> 4));
      a = *reinterpret_cast<const bf16x8*>(&av);
      if (!arow_ok) {
#pragma unroll
        for (int e = 0; e < 8; ++e) a[e] = 0;
      }
    }
#pragma unroll
    for (int nt = 0; nt < CV2_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV2_BK + CV2_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (q0 + CV2_BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV2_BN / 16; ++nt) {
    int k = n0 + nt * 16 + (lane & 15);
    if (k >= K9) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.OC)
        dwc[(int64_t)m * K9 + k] = __float2bfloat16(acc[nt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" void ols_mfma_selftest(const void* A, const void* B, float* D,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(k_mfma_selftest, dim3(1), dim3(64), 0, stream,
                     (const __hip_bfloat16*)A, (const __hip_bfloat16*)B, D);
}

static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

extern "C" void ols_conv3x3_fwd(const void* x, const void* w, void* y, int C,
                                int IC, int OC, int B, int H, int W,
                                int stride, hipStream_t stream) {
  ConvGeom g{B, H, W, (H + stride - 1) / stride, (W + stride - 1) / stride,
             IC, OC, stride, 0, 0, 0};
  {
    auto is_p2 = [](int v) { return v > 0 && (v & (v - 1)) == 0; };
    auto lg = [](int v) { int l = 0; while ((1 << l) < v) ++l; return l; };
    if (is_p2(g.OW) && is_p2(g.OH)) {
      g.lg_ow = lg(g.OW);
      g.lg_ohw = lg(g.OW) + lg(g.OH);
      g.pow2 = 1;
    }
  }
  dim3 grid(ceil_div(B * g.OH * g.OW, CV2_BN), ceil_div(OC, CV2_BM), C);
  const bool v5_ok = g.pow2 && ((IC * 9) % CV2_BK == 0);
  if (v5_ok && g.OW >= CV2_BN / 8)
    hipLaunchKernelGGL((k_conv3x3_fwd_v5<true>), grid, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (__hip_bfloat16*)y, g);
  else if (v5_ok)
    hipLaunchKernelGGL((k_conv3x3_fwd_v5<false>), grid, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (__hip_bfloat16*)y, g);
  else
    hipLaunchKernelGGL(k_conv3x3_fwd_v2, grid, dim3(CONV_THREADS), 0, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                       (__hip_bfloat16*)y, g);
}

extern "C" void ols_conv3x3_dgrad(const void* dy, const void* w, void* dx,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream) {
  ConvGeom g{B, H, W, (H + stride - 1) / stride, (W + stride - 1) / stride,
             IC, OC, stride, 0, 0, 0};
  {
    auto is_p2 = [](int v) { return v > 0 && (v & (v - 1)) == 0; };
    auto lg = [](int v) { int l = 0; while ((1 << l) < v) ++l; return l; };
    if (is_p2(g.OW) && is_p2(g.OH)) {
      g.lg_ow = lg(g.OW);
      g.lg_ohw = lg(g.OW) + lg(g.OH);
      g.pow2 = 1;
    }
  }
  const bool v5_ok = g.pow2 && ((OC * 9) % CV2_BK == 0)
                     && (stride == 1 || stride == 2) && H == W;
  if (v5_ok) {
    dim3 grid5(ceil_div(B * H * W, CV2_BN), ceil_div(IC, CV2_BM), C);
    if (stride == 1)
      hipLaunchKernelGGL((k_conv3x3_dgrad_v5<1>), grid5, dim3(CONV_THREADS),
                         0, stream, (const __hip_bfloat16*)dy,
                         (const __hip_bfloat16*)w, (__hip_bfloat16*)dx, g);
    else
      hipLaunchKernelGGL((k_conv3x3_dgrad_v5<2>), grid5, dim3(CONV_THREADS),
                         0, stream, (const __hip_bfloat16*)dy,
                         (const __hip_bfloat16*)w, (__hip_bfloat16*)dx, g);
  } else {
    dim3 grid(ceil_div(B * H * W, CONV_BN), ceil_div(IC, CONV_BM), C);
    hipLaunchKernelGGL(k_conv3x3_dgrad, grid, dim3(CONV_THREADS), 0, stream,
                       (const __hip_bfloat16*)dy, (const __hip_bfloat16*)w,
                       (__hip_bfloat16*)dx, g);
  }
}

extern "C" void ols_conv3x3_wgrad(const void* x, const void* dy, void* dw,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream) {
  ConvGeom g{B, H, W, (H + stride - 1) / stride, (W + stride - 1) / stride,
             IC, OC, stride, 0, 0, 0};
  {
    auto is_p2 = [](int v) { return v > 0 && (v & (v - 1)) == 0; };
    auto lg = [](int v) { int l = 0; while ((1 << l) < v) ++l; return l; };
    if (is_p2(g.OW) && is_p2(g.OH)) {
      g.lg_ow = lg(g.OW);
      g.lg_ohw = lg(g.OW) + lg(g.OH);
      g.pow2 = 1;
    }
  }
  const int NN = B * g.OH * g.OW;
  const bool v5_ok = g.pow2 && (NN % CV2_BK == 0);
  if (v5_ok) {
    dim3 grid5(ceil_div(IC * 9, CV2_BN), ceil_div(OC, CV2_BM), C);
    hipLaunchKernelGGL(k_conv3x3_wgrad_v5, grid5, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)dy, (__hip_bfloat16*)dw, g);
  } else {
    dim3 grid(ceil_div(IC * 9, CONV_BN), ceil_div(OC, CONV_BM), C);
    hipLaunchKernelGGL(k_conv3x3_wgrad, grid, dim3(CONV_THREADS), 0, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       (__hip_bfloat16*)dw, g);
  }
}
