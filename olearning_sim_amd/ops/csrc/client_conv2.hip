// Client-batched 3x3 convolution, v6 family: padded-input implicit GEMM.
//
// What changed vs client_conv.hip's v5 pipeline (measured on MI355X,
// tools/convbench3.py):
//  - the input (activations for fwd/wgrad, dy for dgrad) arrives with a
//    1-element zero halo per plane ([.., H+2, W+2], F.pad on the host),
//    so the implicit-im2col gather has NO bounds masks, NO clamps and
//    NO float round-trips — each element is one ushort load at a
//    directly-addressed offset (v5 spent ~10 VALU ops per gathered
//    element on decompose+mask+select+cvt; PMC showed the kernels
//    issue-bound on that, MFMA busy <10%);
//  - XCD-aware block mapping: all (m,n)-tiles of one client land on one
//    XCD so the client's operand panels stay in that XCD's 4 MB L2
//    (deep layers re-read x / dy across tiles; dispatch places block b
//    on XCD b%8 — cdna_hip_programming.md T1);
//  - small planes (OW 4/8) hoist the (b,oh) decomposition per sub-row
//    instead of per element (template LG_OW);
//  - stride-2 dgrad is parity-decomposed into 4 dense classes (he,we):
//    output position (2i+he, 2j+we) only receives taps with
//    (he+dh-1)%2==0 / (we+dw-1)%2==0, so each class contracts over
//    K=OC*nh*nw with NO zero rows — v5's masked formulation spent 4x
//    the useful MFMA work (measured 21-29 TF vs ~90 for stride-1).
//
// GEMM structure per (client, m-tile, n-tile) workgroup is v5's:
// BM=64 (4 waves x 16 rows), BN=128, BK=32, A-operand read directly
// from global (16 B/lane/K-step), B tile double-buffered in LDS
// [n][k+pad] so fragments are aligned ds_read_b128, one barrier per
// K-step, register-double-buffered gather (loads for step k+1 issue
// before the MFMAs of step k).
//
// Replaces the role of MIOpen grouped conv on the simulator's hot path
// (reference delegates all compute to operator subprocesses,
// ols_core/taskMgr/utils/utils_run_task.py:496-514 — no kernels to
// port; this is the MI355X-native design).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CV6_BM 64
#define CV6_BN 128
#define CV6_BK 32
#define CV6_PAD 8
#define CONV_THREADS 256

struct ConvGeom6 {
  int B, Hp, Wp;        // padded input plane (Hp = H + 2)
  int OH, OW;           // output plane
  int IC, OC, stride;
  int C, tiles_m, tiles_n;
  int lg_ow, lg_ohw;    // log2(OW), log2(OH*OW) — pow2 planes only
};

// Map a linear block id to (client, tile) with all of one client's
// tiles on one XCD (dispatch: block b -> XCD b%8). The grid is padded
// to ceil(C/8)*8 clients; surplus blocks exit.
__device__ __forceinline__ bool xcd_remap6(int lid, int C, int T,
                                           int& c, int& tile) {
  const int xcd = lid & 7;
  const int s = lid >> 3;
  const int grp = s / T;
  tile = s - grp * T;
  c = xcd + 8 * grp;
  return c < C;
}

// ---------------------------------------------------------------------------
// fwd v6: y[c][oc][n] = sum_k W[c][oc][k] * x_pad[c][ic][b][oh*s+dh][ow*s+dw]
// (k = (ic,dh,dw), dw fastest).
//
// LG_OW: 2 or 3 hoist (b,oh) per sub-row of 4/8 columns; 4 = "OW >= 16",
// whole 16-column segment in one output row.
template <int LG_OW_T, int STRIDE>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd_v6(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom6 g) {
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short bT_lds[2][CV6_BN * (CV6_BK + CV6_PAD)];
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* wrow = wc + (int64_t)min(arow, g.OC - 1) * K;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV6_BK;
  const int nn0 = (threadIdx.x / CV6_BK) * (CV6_BN / 8);
  ushort breg[CV6_BN / 8];

  auto gather = [&](int k0) {
    const int k = k0 + kk;                  // < K (K % BK == 0)
    const int ic = k / 9, r = k - ic * 9;
    const int dh = r / 3, dw = r - dh * 3;
    const ushort* plane = xc + (int64_t)ic * planeB;
    if (LG_OW_T >= 4) {
      // whole 16-col segment inside one output row
      int n = min(n0 + nn0, N - 1);
      int b = n >> g.lg_ohw;
      int q = n & ((1 << g.lg_ohw) - 1);
      int oh = q >> g.lg_ow;
      int ow0 = min(q & ((1 << g.lg_ow) - 1), g.OW - CV6_BN / 8);
      const ushort* row = plane + (int64_t)b * HpWp
                          + (oh * STRIDE + dh) * g.Wp + ow0 * STRIDE + dw;
#pragma unroll
      for (int j = 0; j < CV6_BN / 8; ++j) breg[j] = row[j * STRIDE];
    } else {
      // OW = 4 or 8: hoist per sub-row of OW columns
      constexpr int SUBW = 1 << LG_OW_T;
      constexpr int NROW = (CV6_BN / 8) / SUBW;
#pragma unroll
      for (int rr = 0; rr < NROW; ++rr) {
        int n = min(n0 + nn0 + rr * SUBW, N - 1);
        int b = n >> g.lg_ohw;
        int q = n & ((1 << g.lg_ohw) - 1);
        int oh = q >> LG_OW_T;
        const ushort* row = plane + (int64_t)b * HpWp
                            + (oh * STRIDE + dh) * g.Wp + dw;
#pragma unroll
        for (int j = 0; j < SUBW; ++j)
          breg[rr * SUBW + j] = row[j * STRIDE];
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV6_BK + CV6_PAD) + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV6_BK) {
    __syncthreads();
    if (k0 + CV6_BK < K) gather(k0 + CV6_BK);
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
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV6_BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// fwd v8: BK=32 with a TWO-iteration gather lookahead — two register
// sets; the loads a commit waits for were issued a FULL iteration
// earlier.  MEASURED NEGATIVE (kept env-gated as the record of a
// rejected hypothesis): -2% (s0) to -42% (s3) vs the one-ahead v6
// (gpurun_out/v8_fwd.log vs v6_fwd.log) — the extra live registers
// and hipcc's conservative waits outweigh the longer load window; the
// 73% SQ_WAIT_ANY is saturation of the shared memory system, which a
// deeper per-wave pipeline cannot hide.
template <int LG_OW_T, int STRIDE>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd_v8(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom6 g) {
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short bT_lds[2][CV6_BN * (CV6_BK + CV6_PAD)];
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* wrow = wc + (int64_t)min(arow, g.OC - 1) * K;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV6_BK;
  const int nn0 = (threadIdx.x / CV6_BK) * (CV6_BN / 8);
  ushort brA[CV6_BN / 8], brB[CV6_BN / 8];

  auto gather = [&](ushort* breg, int k0) {
    const int k = k0 + kk;
    const int ic = k / 9, r = k - ic * 9;
    const int dh = r / 3, dw = r - dh * 3;
    const ushort* plane = xc + (int64_t)ic * planeB;
    if (LG_OW_T >= 4) {
      int n = min(n0 + nn0, N - 1);
      int b = n >> g.lg_ohw;
      int q = n & ((1 << g.lg_ohw) - 1);
      int oh = q >> g.lg_ow;
      int ow0 = min(q & ((1 << g.lg_ow) - 1), g.OW - CV6_BN / 8);
      const ushort* row = plane + (int64_t)b * HpWp
                          + (oh * STRIDE + dh) * g.Wp + ow0 * STRIDE + dw;
#pragma unroll
      for (int j = 0; j < CV6_BN / 8; ++j) breg[j] = row[j * STRIDE];
    } else {
      constexpr int SUBW = 1 << LG_OW_T;
      constexpr int NROW = (CV6_BN / 8) / SUBW;
#pragma unroll
      for (int rr = 0; rr < NROW; ++rr) {
        int n = min(n0 + nn0 + rr * SUBW, N - 1);
        int b = n >> g.lg_ohw;
        int q = n & ((1 << g.lg_ohw) - 1);
        int oh = q >> LG_OW_T;
        const ushort* row = plane + (int64_t)b * HpWp
                            + (oh * STRIDE + dh) * g.Wp + dw;
#pragma unroll
        for (int j = 0; j < SUBW; ++j)
          breg[rr * SUBW + j] = row[j * STRIDE];
      }
    }
  };
  auto commit = [&](const ushort* breg, int buf) {
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV6_BK + CV6_PAD) + kk] = (short)breg[j];
  };

  // prologue: k0 staged and published; k1 in flight in brB
  gather(brA, 0);
  commit(brA, 0);
  if (CV6_BK < K) gather(brB, CV6_BK);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV6_BK) {
    __syncthreads();                      // buf[cur] (= step k0) ready
    // issue step k0+2's loads into the set the MFMAs are done with
    ushort* nxt2 = ((k0 / CV6_BK) & 1) ? brB : brA;     // holds k0's data
    ushort* nxt1 = ((k0 / CV6_BK) & 1) ? brA : brB;     // holds k0+1's
    if (k0 + 2 * CV6_BK < K) gather(nxt2, k0 + 2 * CV6_BK);
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
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV6_BK < K) commit(nxt1, cur ^ 1);   // k0+1: issued LAST iter
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// fwd v7: BK=64 — two MFMA sub-steps per barrier (16 MFMAs/wave/step,
// amortising the stage/barrier latency the BK=32 pipeline pays every
// 8 MFMAs; cdna_hip_programming.md: BK 32->64 = +7..16% on the dense
// GEMM ladder), with s_setprio(1) around the MFMA cluster (T5).
// Requires K % 64 == 0 (IC a multiple of 64); launcher falls back to
// v6 otherwise.
template <int LG_OW_T, int STRIDE>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_fwd_v7(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, ConvGeom6 g) {
  constexpr int BK = 64;
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short bT_lds[2][CV6_BN * (BK + CV6_PAD)];
  const int K = g.IC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* wc = w + (int64_t)c * g.OC * K;
  __hip_bfloat16* yc = y + (int64_t)c * g.OC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* wrow = wc + (int64_t)min(arow, g.OC - 1) * K;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x & 63;           // staged k-row
  const int nn0 = (threadIdx.x >> 6) * 32;   // 32 cols per thread
  ushort breg[32];

  auto gather = [&](int k0) {
    const int k = k0 + kk;
    const int ic = k / 9, r = k - ic * 9;
    const int dh = r / 3, dw = r - dh * 3;
    const ushort* plane = xc + (int64_t)ic * planeB;
    constexpr int SUBW = 1 << (LG_OW_T < 5 ? LG_OW_T : 5);
    constexpr int NROW = 32 / SUBW;
#pragma unroll
    for (int rr = 0; rr < NROW; ++rr) {
      int n = min(n0 + nn0 + rr * SUBW, N - 1);
      int b = n >> g.lg_ohw;
      int q = n & ((1 << g.lg_ohw) - 1);
      int oh = q >> g.lg_ow;
      int ow0 = q & ((1 << g.lg_ow) - 1);
      if (SUBW < 32) ow0 = 0;                // sub-row segments are aligned
      else ow0 = min(ow0, g.OW - 32);
      const ushort* row = plane + (int64_t)b * HpWp
                          + (oh * STRIDE + dh) * g.Wp + ow0 * STRIDE + dw;
#pragma unroll
      for (int j = 0; j < SUBW; ++j)
        breg[rr * SUBW + j] = row[j * STRIDE];
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 32; ++j)
      bT_lds[buf][(nn0 + j) * (BK + CV6_PAD) + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    if (k0 + BK < K) gather(k0 + BK);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      bf16x8 a;
      {
        uint4 av = *reinterpret_cast<const uint4*>(
            wrow + k0 + sub * 32 + 8 * (lane >> 4));
        a = *reinterpret_cast<const bf16x8*>(&av);
        if (!arow_ok) {
#pragma unroll
          for (int e = 0; e < 8; ++e) a[e] = 0;
        }
      }
#pragma unroll
      for (int nt = 0; nt < CV6_BN / 16; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &bT_lds[cur][(nt * 16 + (lane & 15)) * (BK + CV6_PAD)
                         + sub * 32 + 8 * (lane >> 4)]);
        acc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    if (k0 + BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// dgrad v6 (stride 1): dX[ic][n] over K=(oc,dh,dw), with the weight
// flip folded into the A staging (A[ic][k] = W[oc][ic][8-(3dh+dw)]) —
// no host-side weight transform, W is read once through L2.
// B gather = fwd's padded gather over dy_pad (output plane = H,W of dX).
template <int LG_OW_T>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_dgrad_v6(
    const __hip_bfloat16* __restrict__ dyp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom6 g) {
  // geometry: OH/OW here are dX's plane (the gather output), Hp/Wp the
  // padded dy plane; IC = output rows (dX channels), OC = contracted.
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short a_lds[2][CV6_BM * CV6_BK];
  __shared__ short bT_lds[2][CV6_BN * (CV6_BK + CV6_PAD)];
  const int K = g.OC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* dyc =
      reinterpret_cast<const ushort*>(dyp) + (int64_t)c * g.OC * planeB;
  const ushort* wc =
      reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * g.IC * 9;
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV6_BK;
  const int nn0 = (threadIdx.x / CV6_BK) * (CV6_BN / 8);
  const int amm = threadIdx.x / 4;              // A row (ic) staged
  const int ak0 = (threadIdx.x % 4) * 8;
  const int aic = min(m0 + amm, g.IC - 1);
  const bool aic_ok = (m0 + amm) < g.IC;
  ushort breg[CV6_BN / 8];
  ushort areg[8];

  auto gather = [&](int k0) {
    {
      // A: 8 consecutive k = (oc, 3dh+dw); flip index 8-(3dh+dw) walks
      // backwards inside each oc's contiguous 9-element block
      const int kb = k0 + ak0;
      int oc0 = kb / 9;
      int r0 = kb - oc0 * 9;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int rr = r0 + j;
        int oc = oc0 + (rr >= 9);
        rr -= (rr >= 9) ? 9 : 0;
        ushort v = wc[((int64_t)oc * g.IC + aic) * 9 + (8 - rr)];
        areg[j] = aic_ok ? v : (ushort)0;
      }
    }
    {
      const int k = k0 + kk;
      const int oc = k / 9, r = k - oc * 9;
      const int dh = r / 3, dw = r - dh * 3;
      const ushort* plane = dyc + (int64_t)oc * planeB;
      if (LG_OW_T >= 4) {
        int n = min(n0 + nn0, N - 1);
        int b = n >> g.lg_ohw;
        int q = n & ((1 << g.lg_ohw) - 1);
        int oh = q >> g.lg_ow;
        int ow0 = min(q & ((1 << g.lg_ow) - 1), g.OW - CV6_BN / 8);
        const ushort* row = plane + (int64_t)b * HpWp + (oh + dh) * g.Wp
                            + ow0 + dw;
#pragma unroll
        for (int j = 0; j < CV6_BN / 8; ++j) breg[j] = row[j];
      } else {
        constexpr int SUBW = 1 << LG_OW_T;
        constexpr int NROW = (CV6_BN / 8) / SUBW;
#pragma unroll
        for (int rr = 0; rr < NROW; ++rr) {
          int n = min(n0 + nn0 + rr * SUBW, N - 1);
          int b = n >> g.lg_ohw;
          int q = n & ((1 << g.lg_ohw) - 1);
          int oh = q >> LG_OW_T;
          const ushort* row = plane + (int64_t)b * HpWp + (oh + dh) * g.Wp + dw;
#pragma unroll
          for (int j = 0; j < SUBW; ++j) breg[rr * SUBW + j] = row[j];
        }
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      a_lds[buf][amm * CV6_BK + ak0 + j] = (short)areg[j];
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV6_BK + CV6_PAD) + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV6_BK) {
    __syncthreads();
    if (k0 + CV6_BK < K) gather(k0 + CV6_BK);
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[cur][(wave * 16 + (lane & 15)) * CV6_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV6_BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// dgrad/wgrad v7: BK=64 (two MFMA sub-steps per barrier) — the fwd v7
// structure applied to the backward pipelines.  MEASURED NEUTRAL
// (dgrad, +9% only at s1d) to NEGATIVE (wgrad -5..-10%: the 37 KB LDS
// halves residency); kept env-gated (OLSIM_CONV_DW64=1) as the A/B
// record — gpurun_out/dw64.log vs v8_fwd.log baseline columns.
template <int LG_OW_T>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_dgrad_v7(
    const __hip_bfloat16* __restrict__ dyp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom6 g) {
  constexpr int BK = 64;
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short a_lds[2][CV6_BM * BK];
  __shared__ short bT_lds[2][CV6_BN * (BK + CV6_PAD)];
  const int K = g.OC * 9;
  const int N = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* dyc =
      reinterpret_cast<const ushort*>(dyp) + (int64_t)c * g.OC * planeB;
  const ushort* wc =
      reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * g.IC * 9;
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * N;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x & 63;            // B k-row
  const int nn0 = (threadIdx.x >> 6) * 32;    // 32 cols / thread
  const int amm = threadIdx.x / 4;            // A row (ic)
  const int ak0 = (threadIdx.x % 4) * 16;     // 16 k / thread
  const int aic = min(m0 + amm, g.IC - 1);
  const bool aic_ok = (m0 + amm) < g.IC;
  ushort breg[32];
  ushort areg[16];

  auto gather = [&](int k0) {
    {
      const int kb = k0 + ak0;
      int oc0 = kb / 9;
      int r0 = kb - oc0 * 9;
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        int rr = r0 + j;
        int carry = (rr >= 18) ? 2 : (rr >= 9 ? 1 : 0);
        int oc = oc0 + carry;
        rr -= 9 * carry;
        ushort v = wc[((int64_t)oc * g.IC + aic) * 9 + (8 - rr)];
        areg[j] = aic_ok ? v : (ushort)0;
      }
    }
    {
      const int k = k0 + kk;
      const int oc = k / 9, r = k - oc * 9;
      const int dh = r / 3, dw = r - dh * 3;
      const ushort* plane = dyc + (int64_t)oc * planeB;
      constexpr int SUBW = 1 << (LG_OW_T < 5 ? LG_OW_T : 5);
      constexpr int NROW = 32 / SUBW;
#pragma unroll
      for (int rr = 0; rr < NROW; ++rr) {
        int n = min(n0 + nn0 + rr * SUBW, N - 1);
        int b = n >> g.lg_ohw;
        int q = n & ((1 << g.lg_ohw) - 1);
        int oh = q >> g.lg_ow;
        int ow0 = q & ((1 << g.lg_ow) - 1);
        if (SUBW < 32) ow0 = 0;
        else ow0 = min(ow0, g.OW - 32);
        const ushort* row = plane + (int64_t)b * HpWp + (oh + dh) * g.Wp
                            + ow0 + dw;
#pragma unroll
        for (int j = 0; j < SUBW; ++j) breg[rr * SUBW + j] = row[j];
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 16; ++j)
      a_lds[buf][amm * BK + ak0 + j] = (short)areg[j];
#pragma unroll
    for (int j = 0; j < 32; ++j)
      bT_lds[buf][(nn0 + j) * (BK + CV6_PAD) + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    if (k0 + BK < K) gather(k0 + BK);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &a_lds[cur][(wave * 16 + (lane & 15)) * BK + sub * 32
                      + 8 * (lane >> 4)]);
#pragma unroll
      for (int nt = 0; nt < CV6_BN / 16; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &bT_lds[cur][(nt * 16 + (lane & 15)) * (BK + CV6_PAD)
                         + sub * 32 + 8 * (lane >> 4)]);
        acc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    if (k0 + BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// dgrad stride 2, parity-decomposed: one launch per class (HE, WE).
// Output positions (2i+HE, 2j+WE); valid taps dh with (HE+dh-1) even:
// HE=0 -> {1}, HE=1 -> {0,2} (same for WE/dw).  Class-k = (oc, th, tw)
// over K = OC*NH*NW (NH,NW in {1,2}); the dy read offset is
// (i + (HE+dh-1)/2, j + (WE+dw-1)/2) into the padded dy plane.
// Class output plane = OH x OW (= dy's plane: H/2 x W/2).
template <int HE, int WE, int LG_OW_T>
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_dgrad_s2(
    const __hip_bfloat16* __restrict__ dyp, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ dx, ConvGeom6 g) {
  constexpr int NH = HE ? 2 : 1;
  constexpr int NW = WE ? 2 : 1;
  constexpr int NT = NH * NW;
  // valid dh per class (flip index uses 2-dh); delta = (HE+dh-1)/2
  constexpr int DH0 = HE ? 0 : 1;     // first valid dh
  constexpr int DW0 = WE ? 0 : 1;
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;

  __shared__ short a_lds[2][CV6_BM * CV6_BK];
  __shared__ short bT_lds[2][CV6_BN * (CV6_BK + CV6_PAD)];
  const int K = g.OC * NT;
  const int N = g.B * g.OH * g.OW;          // class grid = dy plane
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* dyc =
      reinterpret_cast<const ushort*>(dyp) + (int64_t)c * g.OC * planeB;
  const ushort* wc =
      reinterpret_cast<const ushort*>(w) + (int64_t)c * g.OC * g.IC * 9;
  const int H = 2 * g.OH, W = 2 * g.OW;     // dX plane
  __hip_bfloat16* dxc = dx + (int64_t)c * g.IC * g.B * H * W;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kk = threadIdx.x % CV6_BK;
  const int nn0 = (threadIdx.x / CV6_BK) * (CV6_BN / 8);
  const int amm = threadIdx.x / 4;
  const int ak0 = (threadIdx.x % 4) * 8;
  const int aic = min(m0 + amm, g.IC - 1);
  const bool aic_ok = (m0 + amm) < g.IC;
  ushort breg[CV6_BN / 8];
  ushort areg[8];

  auto gather = [&](int k0) {
    {
      // A[ic][k=(oc,th,tw)] = W[oc][ic][(2-dh)*3+(2-dw)],
      // dh = DH0 + 2*th, dw = DW0 + 2*tw
      const int kb = k0 + ak0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int k = kb + j;
        int oc = k / NT;                    // NT pow2: shift
        int ti = k - oc * NT;
        int th = NW == 2 ? (ti >> 1) : ti;  // ti = th*NW + tw
        int tw = NW == 2 ? (ti & 1) : 0;
        int dh = DH0 + 2 * th;
        int dw = DW0 + 2 * tw;
        ushort v = wc[((int64_t)oc * g.IC + aic) * 9 + (2 - dh) * 3 + (2 - dw)];
        areg[j] = aic_ok ? v : (ushort)0;
      }
    }
    {
      const int k = k0 + kk;
      const int oc = k / NT;
      const int ti = k - oc * NT;
      const int th = NW == 2 ? (ti >> 1) : ti;
      const int tw = NW == 2 ? (ti & 1) : 0;
      const int dh = DH0 + 2 * th;
      const int dw = DW0 + 2 * tw;
      const int deh = (HE + dh - 1) >> 1;   // dy row offset (0 or 1)
      const int dew = (WE + dw - 1) >> 1;
      const ushort* plane = dyc + (int64_t)oc * planeB;
      if (LG_OW_T >= 4) {
        int n = min(n0 + nn0, N - 1);
        int b = n >> g.lg_ohw;
        int q = n & ((1 << g.lg_ohw) - 1);
        int i = q >> g.lg_ow;
        int j0 = min(q & ((1 << g.lg_ow) - 1), g.OW - CV6_BN / 8);
        const ushort* row = plane + (int64_t)b * HpWp + (i + deh + 1) * g.Wp
                            + j0 + dew + 1;
#pragma unroll
        for (int j = 0; j < CV6_BN / 8; ++j) breg[j] = row[j];
      } else {
        constexpr int SUBW = 1 << LG_OW_T;
        constexpr int NROW = (CV6_BN / 8) / SUBW;
#pragma unroll
        for (int rr = 0; rr < NROW; ++rr) {
          int n = min(n0 + nn0 + rr * SUBW, N - 1);
          int b = n >> g.lg_ohw;
          int q = n & ((1 << g.lg_ohw) - 1);
          int i = q >> LG_OW_T;
          const ushort* row = plane + (int64_t)b * HpWp
                              + (i + deh + 1) * g.Wp + dew + 1;
#pragma unroll
          for (int j = 0; j < SUBW; ++j) breg[rr * SUBW + j] = row[j];
        }
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      a_lds[buf][amm * CV6_BK + ak0 + j] = (short)areg[j];
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV6_BK + CV6_PAD) + kk] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += CV6_BK) {
    __syncthreads();
    if (k0 + CV6_BK < K) gather(k0 + CV6_BK);
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        &a_lds[cur][(wave * 16 + (lane & 15)) * CV6_BK + 8 * (lane >> 4)]);
#pragma unroll
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (k0 + CV6_BK < K) commit(cur ^ 1);
    cur ^= 1;
  }

  // scatter back to the strided class positions (2i+HE, 2j+WE)
#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
    int n = n0 + nt * 16 + (lane & 15);
    if (n >= N) continue;
    int b = n >> g.lg_ohw;
    int q = n & ((1 << g.lg_ohw) - 1);
    int i = q >> g.lg_ow;
    int j = q & ((1 << g.lg_ow) - 1);
    int64_t base = ((int64_t)b * H + 2 * i + HE) * W + 2 * j + WE;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      if (m < g.IC)
        dxc[(int64_t)m * g.B * H * W + base] =
            from_f32<__hip_bfloat16>(acc[nt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v6: dW[oc][k9] = sum_q dY[oc][q] * P[k9][q]; A = dY rows direct
// from global (unpadded, contiguous); B = patch gather from x_pad with
// the per-column offsets (ic*B*HpWp + dh*Wp + dw) hoisted out of the
// q-loop entirely — per step only the (b,oh,ow) base changes.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad_v6(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom6 g) {
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;                  // over OC
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;   // over IC*9
  __shared__ short bT_lds[2][CV6_BN * (CV6_PAD + CV6_BK)];
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;            // reduction
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int qq = threadIdx.x % CV6_BK;
  const int nn0 = (threadIdx.x / CV6_BK) * (CV6_BN / 8);
  ushort breg[CV6_BN / 8];

  // per-column gather offsets: off[j] = ic*B*HpWp + dh*Wp + dw (clamped
  // duplicates for the K9 tail; outputs there are write-masked)
  int64_t off[CV6_BN / 8];
#pragma unroll
  for (int j = 0; j < CV6_BN / 8; ++j) {
    int k = min(n0 + nn0 + j, K9 - 1);
    int ic = k / 9, r = k - ic * 9;
    int dh = r / 3, dw2 = r - dh * 3;
    off[j] = (int64_t)ic * planeB + dh * g.Wp + dw2;
  }

  auto gather = [&](int q0) {
    int q = min(q0 + qq, NN - 1);
    int b = q >> g.lg_ohw;
    int p = q & ((1 << g.lg_ohw) - 1);
    int oh = p >> g.lg_ow;
    int ow = p & ((1 << g.lg_ow) - 1);
    const ushort* base = xc + (int64_t)b * HpWp
                         + (oh * g.stride) * g.Wp + ow * g.stride;
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j) breg[j] = base[off[j]];
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < CV6_BN / 8; ++j)
      bT_lds[buf][(nn0 + j) * (CV6_BK + CV6_PAD) + qq] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += CV6_BK) {
    __syncthreads();
    if (q0 + CV6_BK < NN) gather(q0 + CV6_BK);
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
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (q0 + CV6_BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// wgrad v8 (stride 1): run-vectorised B gather.  PMC on v6
// (profiles/pmc_final_r02.md): 19% VALU + 26% instruction-wait, MFMA
// busy only 18% — the producer's 16 scalar u16 loads + 16 ds_write_b16
// per thread per step dominate the instruction stream.  Here one
// producer item owns a (ic, dh) "run" x 8 consecutive q: 8 consecutive
// stride-1 q positions stay inside one padded row, so THREE 8-byte
// loads cover the 10 u16 needed for all three dw columns, extracted
// branchlessly (E is always even: Wp, HpWp, planeB and ow are even)
// and committed as one ds_write_b128 per column.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad_v8(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom6 g) {
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;                  // over OC
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;   // over IC*9
  __shared__ short bT_lds[2][CV6_BN * (CV6_PAD + CV6_BK)];
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;            // reduction
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  // producer item = (run, q-octet): run = (ic, dh) pair of the window
  const int runs_lo = n0 / 3;
  const int runs_hi = (min(n0 + CV6_BN, K9) + 2) / 3;
  const int nitems = (runs_hi - runs_lo) * (CV6_BK / 8);
  const bool prod = threadIdx.x < nitems;
  const int run = runs_lo + threadIdx.x / (CV6_BK / 8);
  const int qv = (threadIdx.x % (CV6_BK / 8)) * 8;
  const int ric = run / 3, rdh = run - ric * 3;
  const int64_t runoff = (int64_t)ric * planeB + rdh * g.Wp;

  uint2 r0, r1, r2;
  int tpar = 0;

  auto gather = [&](int q0) {
    if (!prod) return;
    const int q = q0 + qv;
    const int b = q >> g.lg_ohw;
    const int p = q & ((1 << g.lg_ohw) - 1);
    const int oh = p >> g.lg_ow;
    const int ow = p & ((1 << g.lg_ow) - 1);
    const int64_t E = runoff + (int64_t)b * HpWp + oh * g.Wp + ow;
    // clamp so the 12-u16 window never reads past the tensor (only the
    // final element of the final plane can clamp, by exactly 2)
    const int64_t E4 = min(E & ~3LL, (int64_t)g.IC * planeB - 12);
    tpar = (int)(E - E4) >> 1;               // 0 or 1 (E is even)
    const uint2* src = reinterpret_cast<const uint2*>(xc + E4);
    r0 = src[0];
    r1 = src[1];
    r2 = src[2];
  };
  auto commit = [&](int buf) {
    if (!prod) return;
    // s[k] = u32 of u16 elements (E + 2k, E + 2k + 1)
    uint32_t s[5];
    s[0] = tpar ? r0.y : r0.x;
    s[1] = tpar ? r1.x : r0.y;
    s[2] = tpar ? r1.y : r1.x;
    s[3] = tpar ? r2.x : r1.y;
    s[4] = tpar ? r2.y : r2.x;
#pragma unroll
    for (int dw2 = 0; dw2 < 3; ++dw2) {
      const int k9 = run * 3 + dw2;
      const int col = k9 - n0;
      if (col < 0 || col >= CV6_BN || k9 >= K9) continue;
      uint4 o;
      if (dw2 == 0) {
        o.x = s[0]; o.y = s[1]; o.z = s[2]; o.w = s[3];
      } else if (dw2 == 2) {
        o.x = s[1]; o.y = s[2]; o.z = s[3]; o.w = s[4];
      } else {
        o.x = __builtin_amdgcn_alignbit(s[1], s[0], 16);
        o.y = __builtin_amdgcn_alignbit(s[2], s[1], 16);
        o.z = __builtin_amdgcn_alignbit(s[3], s[2], 16);
        o.w = __builtin_amdgcn_alignbit(s[4], s[3], 16);
      }
      *reinterpret_cast<uint4*>(
          &bT_lds[buf][col * (CV6_BK + CV6_PAD) + qv]) = o;
    }
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += CV6_BK) {
    __syncthreads();
    if (q0 + CV6_BK < NN) gather(q0 + CV6_BK);
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
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (q0 + CV6_BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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


// wgrad v8s: run-vectorised producer for STRIDE 2 (OW >= 8).  Same
// item structure as v8, but the 8 q positions map to every-2nd input
// pixel, so the 3 dw columns need 17 consecutive u16 (5 8-byte loads)
// and the extraction picks alternate elements with constant-selector
// v_perm after one branchless word-shift.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad_v8s(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom6 g) {
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;
  __shared__ short bT_lds[2][CV6_BN * (CV6_PAD + CV6_BK)];
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int runs_lo = n0 / 3;
  const int runs_hi = (min(n0 + CV6_BN, K9) + 2) / 3;
  const int nitems = (runs_hi - runs_lo) * (CV6_BK / 8);
  const bool prod = threadIdx.x < nitems;
  const int run = runs_lo + threadIdx.x / (CV6_BK / 8);
  const int qv = (threadIdx.x % (CV6_BK / 8)) * 8;
  const int ric = run / 3, rdh = run - ric * 3;
  const int64_t runoff = (int64_t)ric * planeB + rdh * g.Wp;

  uint2 r0, r1, r2, r3, r4;
  int tpar = 0;

  auto gather = [&](int q0) {
    if (!prod) return;
    const int q = q0 + qv;
    const int b = q >> g.lg_ohw;
    const int p = q & ((1 << g.lg_ohw) - 1);
    const int oh = p >> g.lg_ow;
    const int ow = p & ((1 << g.lg_ow) - 1);
    const int64_t E = runoff + (int64_t)b * HpWp + (oh * 2) * g.Wp + ow * 2;
    const int64_t E4 = min(E & ~3LL, (int64_t)g.IC * planeB - 20);
    tpar = (int)(E - E4) >> 1;               // 0 or 1 (E is even)
    const uint2* src = reinterpret_cast<const uint2*>(xc + E4);
    r0 = src[0];
    r1 = src[1];
    r2 = src[2];
    r3 = src[3];
    r4 = src[4];
  };
  auto commit = [&](int buf) {
    if (!prod) return;
    uint32_t w[10] = {r0.x, r0.y, r1.x, r1.y, r2.x,
                      r2.y, r3.x, r3.y, r4.x, r4.y};
    // s[k] = u32 of u16 elements (E + 2k, E + 2k + 1), k 0..8
    uint32_t s[9];
#pragma unroll
    for (int k = 0; k < 9; ++k) s[k] = tpar ? w[k + 1] : w[k];
    // col dw needs elements E + dw + 2j (j 0..7): alternate u16s
#pragma unroll
    for (int dw2 = 0; dw2 < 3; ++dw2) {
      const int k9 = run * 3 + dw2;
      const int col = k9 - n0;
      if (col < 0 || col >= CV6_BN || k9 >= K9) continue;
      // o[m] = (elem dw+4m, elem dw+4m+2); dw even -> lo16 halves of
      // s[dw/2+2m], s[dw/2+2m+1]; dw odd -> hi16 halves of s[2m], s[2m+1]
      const int b0 = (dw2 + 1) >> 1;   // 0, 0, 1 -> s-base per pair step
      const uint32_t sel = (dw2 & 1) ? 0x07060302u : 0x05040100u;
      const int base = (dw2 == 2) ? 1 : 0;
      uint4 o;
      o.x = __builtin_amdgcn_perm(s[base + 1], s[base + 0], sel);
      o.y = __builtin_amdgcn_perm(s[base + 3], s[base + 2], sel);
      o.z = __builtin_amdgcn_perm(s[base + 5], s[base + 4], sel);
      o.w = __builtin_amdgcn_perm(s[base + 7], s[base + 6], sel);
      (void)b0;
      *reinterpret_cast<uint4*>(
          &bT_lds[buf][col * (CV6_BK + CV6_PAD) + qv]) = o;
    }
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += CV6_BK) {
    __syncthreads();
    if (q0 + CV6_BK < NN) gather(q0 + CV6_BK);
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
    for (int nt = 0; nt < CV6_BN / 16; ++nt) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &bT_lds[cur][(nt * 16 + (lane & 15)) * (CV6_BK + CV6_PAD)
                       + 8 * (lane >> 4)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
    if (q0 + CV6_BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// wgrad v7: BK=64 reduction steps (two MFMA sub-steps per barrier),
// int32 gather offsets.  Requires NN % 64 == 0.
__global__ __launch_bounds__(CONV_THREADS) void k_conv3x3_wgrad_v7(
    const __hip_bfloat16* __restrict__ xp, const __hip_bfloat16* __restrict__ dy,
    __hip_bfloat16* __restrict__ dw, ConvGeom6 g) {
  constexpr int BK = 64;
  int c, tile;
  if (!xcd_remap6(blockIdx.x, g.C, g.tiles_m * g.tiles_n, c, tile)) return;
  const int mt = tile / g.tiles_n;
  const int m0 = mt * CV6_BM;                  // over OC
  const int n0 = (tile - mt * g.tiles_n) * CV6_BN;   // over IC*9
  __shared__ short bT_lds[2][CV6_BN * (BK + CV6_PAD)];
  const int K9 = g.IC * 9;
  const int NN = g.B * g.OH * g.OW;            // reduction
  const int HpWp = g.Hp * g.Wp;
  const int64_t planeB = (int64_t)g.B * HpWp;
  const ushort* xc =
      reinterpret_cast<const ushort*>(xp) + (int64_t)c * g.IC * planeB;
  const __hip_bfloat16* dyc = dy + (int64_t)c * g.OC * NN;
  __hip_bfloat16* dwc = dw + (int64_t)c * g.OC * K9;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int arow = m0 + wave * 16 + (lane & 15);
  const __hip_bfloat16* dyrow = dyc + (int64_t)min(arow, g.OC - 1) * NN;
  const bool arow_ok = arow < g.OC;

  f32x4 acc[CV6_BN / 16];
#pragma unroll
  for (int i = 0; i < CV6_BN / 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int qq = threadIdx.x & 63;
  const int nn0 = (threadIdx.x >> 6) * 32;     // 32 K9-cols / thread
  ushort breg[32];

  int off[32];                                 // plane offsets fit int32
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    int k = min(n0 + nn0 + j, K9 - 1);
    int ic = k / 9, r = k - ic * 9;
    int dh = r / 3, dw2 = r - dh * 3;
    off[j] = (int)((int64_t)ic * planeB + dh * g.Wp + dw2);
  }

  auto gather = [&](int q0) {
    int q = min(q0 + qq, NN - 1);
    int b = q >> g.lg_ohw;
    int p = q & ((1 << g.lg_ohw) - 1);
    int oh = p >> g.lg_ow;
    int ow = p & ((1 << g.lg_ow) - 1);
    const ushort* base = xc + (int64_t)b * HpWp
                         + (oh * g.stride) * g.Wp + ow * g.stride;
#pragma unroll
    for (int j = 0; j < 32; ++j) breg[j] = base[off[j]];
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 32; ++j)
      bT_lds[buf][(nn0 + j) * (BK + CV6_PAD) + qq] = (short)breg[j];
  };

  gather(0);
  commit(0);
  int cur = 0;
  for (int q0 = 0; q0 < NN; q0 += BK) {
    __syncthreads();
    if (q0 + BK < NN) gather(q0 + BK);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      bf16x8 a;
      {
        uint4 av = *reinterpret_cast<const uint4*>(
            dyrow + q0 + sub * 32 + 8 * (lane >> 4));
        a = *reinterpret_cast<const bf16x8*>(&av);
        if (!arow_ok) {
#pragma unroll
          for (int e = 0; e < 8; ++e) a[e] = 0;
        }
      }
#pragma unroll
      for (int nt = 0; nt < CV6_BN / 16; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &bT_lds[cur][(nt * 16 + (lane & 15)) * (BK + CV6_PAD)
                         + sub * 32 + 8 * (lane >> 4)]);
        acc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    if (q0 + BK < NN) commit(cur ^ 1);
    cur ^= 1;
  }

#pragma unroll
  for (int nt = 0; nt < CV6_BN / 16; ++nt) {
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
// launchers

static inline int cdiv6(int a, int b) { return (a + b - 1) / b; }

static inline void fill_geom6(ConvGeom6& g, int C, int IC, int OC, int B,
                              int H, int W, int stride) {
  g.B = B; g.Hp = H + 2; g.Wp = W + 2;
  g.OH = (H + stride - 1) / stride; g.OW = (W + stride - 1) / stride;
  g.IC = IC; g.OC = OC; g.stride = stride; g.C = C;
  auto lg = [](int v) { int l = 0; while ((1 << l) < v) ++l; return l; };
  g.lg_ow = lg(g.OW);
  g.lg_ohw = lg(g.OW) + lg(g.OH);
}

static inline int xcd_blocks6(int C, int T) {
  return ((C + 7) / 8 * 8) * T;
}

// 1 when the v6 padded path supports the shape (pow2 planes, K%32==0)
extern "C" int ols_conv3x3_v6_ok(int IC, int OC, int B, int H, int W,
                                 int stride) {
  auto p2 = [](int v) { return v > 0 && (v & (v - 1)) == 0; };
  int OH = (H + stride - 1) / stride, OW = (W + stride - 1) / stride;
  if (!p2(OH) || !p2(OW) || OW < 4) return 0;
  if ((IC * 9) % CV6_BK != 0) return 0;             // fwd K
  if ((OC * 9) % CV6_BK != 0) return 0;             // dgrad K (s=1)
  if (stride == 2 && OC % CV6_BK != 0) return 0;    // smallest s2 class K=OC
  if (stride != 1 && stride != 2) return 0;
  if ((B * OH * OW) % CV6_BK != 0) return 0;        // wgrad A-direct rows
  return 1;
}

extern "C" void ols_conv3x3_fwd_p(const void* xp, const void* w, void* y,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream) {
  ConvGeom6 g;
  fill_geom6(g, C, IC, OC, B, H, W, stride);
  g.tiles_m = cdiv6(OC, CV6_BM);
  g.tiles_n = cdiv6(B * g.OH * g.OW, CV6_BN);
  dim3 grid(xcd_blocks6(C, g.tiles_m * g.tiles_n));
  auto xb = (const __hip_bfloat16*)xp;
  auto wb = (const __hip_bfloat16*)w;
  auto yb = (__hip_bfloat16*)y;
#define LAUNCH_FWD6(LG, ST) \
  hipLaunchKernelGGL((k_conv3x3_fwd_v6<LG, ST>), grid, dim3(CONV_THREADS), \
                     0, stream, xb, wb, yb, g)
#define LAUNCH_FWD7(LG, ST) \
  hipLaunchKernelGGL((k_conv3x3_fwd_v7<LG, ST>), grid, dim3(CONV_THREADS), \
                     0, stream, xb, wb, yb, g)
  // BK=64 (v7) wins +5..10% at OW >= 8 stride 1 (A/B: gpurun_out/
  // v7_fwd.log vs v6_fwd.log at C=250); the 4x4-plane deep layers and
  // stride 2 measured flat-to-worse, so they stay on BK=32.
  // OLSIM_CONV_V8 forces the two-ahead BK=32 pipeline everywhere (A/B).
#define LAUNCH_FWD8(LG, ST) \
  hipLaunchKernelGGL((k_conv3x3_fwd_v8<LG, ST>), grid, dim3(CONV_THREADS), \
                     0, stream, xb, wb, yb, g)
  if (getenv("OLSIM_CONV_V8") != nullptr) {
    if (stride == 1) {
      if (g.lg_ow >= 4) LAUNCH_FWD8(4, 1);
      else if (g.lg_ow == 3) LAUNCH_FWD8(3, 1);
      else LAUNCH_FWD8(2, 1);
    } else {
      if (g.lg_ow >= 4) LAUNCH_FWD8(4, 2);
      else if (g.lg_ow == 3) LAUNCH_FWD8(3, 2);
      else LAUNCH_FWD8(2, 2);
    }
    return;
  }
  const bool v7 = ((IC * 9) % 64 == 0) && stride == 1 && g.lg_ow >= 3;
  if (stride == 1) {
    if (v7 && g.lg_ow >= 5) LAUNCH_FWD7(5, 1);
    else if (v7 && g.lg_ow == 4) LAUNCH_FWD7(4, 1);
    else if (v7) LAUNCH_FWD7(3, 1);
    else if (g.lg_ow >= 4) LAUNCH_FWD6(4, 1);
    else if (g.lg_ow == 3) LAUNCH_FWD6(3, 1);
    else LAUNCH_FWD6(2, 1);
  } else {
    if (g.lg_ow >= 4) LAUNCH_FWD6(4, 2);
    else if (g.lg_ow == 3) LAUNCH_FWD6(3, 2);
    else LAUNCH_FWD6(2, 2);
  }
#undef LAUNCH_FWD6
#undef LAUNCH_FWD7
#undef LAUNCH_FWD8
}

extern "C" void ols_conv3x3_dgrad_p(const void* dyp, const void* w, void* dx,
                                    int C, int IC, int OC, int B, int H,
                                    int W, int stride, hipStream_t stream) {
  if (stride == 1) {
    // plane geometry: padded dy plane = H+2 x W+2 (dy is H x W at s=1);
    // output plane = H x W
    ConvGeom6 g;
    fill_geom6(g, C, IC, OC, B, H, W, 1);
    g.tiles_m = cdiv6(IC, CV6_BM);
    g.tiles_n = cdiv6(B * H * W, CV6_BN);
    dim3 grid(xcd_blocks6(C, g.tiles_m * g.tiles_n));
    auto db = (const __hip_bfloat16*)dyp;
    auto wb = (const __hip_bfloat16*)w;
    auto xb = (__hip_bfloat16*)dx;
    if (getenv("OLSIM_CONV_DW64") != nullptr && (OC * 9) % 64 == 0) {
      if (g.lg_ow >= 5)
        hipLaunchKernelGGL((k_conv3x3_dgrad_v7<5>), grid, dim3(CONV_THREADS),
                           0, stream, db, wb, xb, g);
      else if (g.lg_ow == 4)
        hipLaunchKernelGGL((k_conv3x3_dgrad_v7<4>), grid, dim3(CONV_THREADS),
                           0, stream, db, wb, xb, g);
      else if (g.lg_ow == 3)
        hipLaunchKernelGGL((k_conv3x3_dgrad_v7<3>), grid, dim3(CONV_THREADS),
                           0, stream, db, wb, xb, g);
      else
        hipLaunchKernelGGL((k_conv3x3_dgrad_v7<2>), grid, dim3(CONV_THREADS),
                           0, stream, db, wb, xb, g);
      return;
    }
    if (g.lg_ow >= 4)
      hipLaunchKernelGGL((k_conv3x3_dgrad_v6<4>), grid, dim3(CONV_THREADS),
                         0, stream, db, wb, xb, g);
    else if (g.lg_ow == 3)
      hipLaunchKernelGGL((k_conv3x3_dgrad_v6<3>), grid, dim3(CONV_THREADS),
                         0, stream, db, wb, xb, g);
    else
      hipLaunchKernelGGL((k_conv3x3_dgrad_v6<2>), grid, dim3(CONV_THREADS),
                         0, stream, db, wb, xb, g);
    return;
  }
  // stride 2: 4 parity classes; geometry carries the dy plane (OH x OW,
  // padded Hp x Wp) and the class grid N = B*OH*OW
  ConvGeom6 g;
  int OH = H / 2, OW = W / 2;
  g.B = B; g.Hp = OH + 2; g.Wp = OW + 2;
  g.OH = OH; g.OW = OW;
  g.IC = IC; g.OC = OC; g.stride = 2; g.C = C;
  auto lg = [](int v) { int l = 0; while ((1 << l) < v) ++l; return l; };
  g.lg_ow = lg(OW);
  g.lg_ohw = lg(OW) + lg(OH);
  g.tiles_m = cdiv6(IC, CV6_BM);
  g.tiles_n = cdiv6(B * OH * OW, CV6_BN);
  dim3 grid(xcd_blocks6(C, g.tiles_m * g.tiles_n));
  auto db = (const __hip_bfloat16*)dyp;
  auto wb = (const __hip_bfloat16*)w;
  auto xb = (__hip_bfloat16*)dx;
#define LAUNCH_S2(HE, WE, LG) \
  hipLaunchKernelGGL((k_conv3x3_dgrad_s2<HE, WE, LG>), grid, \
                     dim3(CONV_THREADS), 0, stream, db, wb, xb, g)
  if (g.lg_ow >= 4) {
    LAUNCH_S2(0, 0, 4); LAUNCH_S2(0, 1, 4); LAUNCH_S2(1, 0, 4); LAUNCH_S2(1, 1, 4);
  } else if (g.lg_ow == 3) {
    LAUNCH_S2(0, 0, 3); LAUNCH_S2(0, 1, 3); LAUNCH_S2(1, 0, 3); LAUNCH_S2(1, 1, 3);
  } else {
    LAUNCH_S2(0, 0, 2); LAUNCH_S2(0, 1, 2); LAUNCH_S2(1, 0, 2); LAUNCH_S2(1, 1, 2);
  }
#undef LAUNCH_S2
}

extern "C" void ols_conv3x3_wgrad_p(const void* xp, const void* dy, void* dw,
                                    int C, int IC, int OC, int B, int H,
                                    int W, int stride, hipStream_t stream) {
  ConvGeom6 g;
  fill_geom6(g, C, IC, OC, B, H, W, stride);
  g.tiles_m = cdiv6(OC, CV6_BM);
  g.tiles_n = cdiv6(IC * 9, CV6_BN);
  dim3 grid(xcd_blocks6(C, g.tiles_m * g.tiles_n));
  const int NN = B * g.OH * g.OW;
  if (stride == 2 && g.OW >= 8 && getenv("OLSIM_CONV_DW8") == nullptr) {
    hipLaunchKernelGGL(k_conv3x3_wgrad_v8s, grid, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)xp,
                       (const __hip_bfloat16*)dy, (__hip_bfloat16*)dw, g);
    return;
  }
  if (stride == 1 && g.OW >= 8 && getenv("OLSIM_CONV_DW8") == nullptr) {
    // run-vectorised producer (v8): 8 consecutive stride-1 q stay in
    // one padded row; OLSIM_CONV_DW8=0 restores the scalar gather
    hipLaunchKernelGGL(k_conv3x3_wgrad_v8, grid, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)xp,
                       (const __hip_bfloat16*)dy, (__hip_bfloat16*)dw, g);
    return;
  }
  if (getenv("OLSIM_CONV_DW64") != nullptr && NN % 64 == 0) {
    hipLaunchKernelGGL(k_conv3x3_wgrad_v7, grid, dim3(CONV_THREADS), 0,
                       stream, (const __hip_bfloat16*)xp,
                       (const __hip_bfloat16*)dy, (__hip_bfloat16*)dw, g);
    return;
  }
  hipLaunchKernelGGL(k_conv3x3_wgrad_v6, grid, dim3(CONV_THREADS), 0, stream,
                     (const __hip_bfloat16*)xp, (const __hip_bfloat16*)dy,
                     (__hip_bfloat16*)dw, g);
}
