// Fused SGD / FedProx update over the flat client-replica buffer.
//
// Replaces the optimiser step of every simulated device's local training
// (the work the reference delegates to a subprocess per phone,
// ols_core/taskMgr/utils/utils_run_task.py:496-514) with ONE launch over
// all co-resident clients:
//
//   buf[p] -= lr * (grad[p] + mu * (buf[p] - g[jmap(p)]))
//
// Layout (engine/client_manager.py): buf is the concatenation over
// parameter blocks b of [C, n_b]; block b spans buf positions
// C*offs[b] .. C*offs[b+1] and corresponds to master elements
// offs[b] .. offs[b+1].  jmap(p) = offs[b] + (p - C*offs[b]) % n_b.
//
// mu == 0 (FedAvg) needs no master map and runs as a pure 16-B/lane
// streaming saxpy: 3 streams (read w, read g, write w) -> HBM-bound.

#include "common.h"

template <typename T, int V>
__global__ __launch_bounds__(OLS_THREADS) void k_sgd_plain(
    T* __restrict__ buf, const T* __restrict__ grad, int64_t npacks,
    float lr) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < npacks; i += stride) {
    Pack<T, V> w = reinterpret_cast<const Pack<T, V>*>(buf)[i];
    Pack<T, V> g = reinterpret_cast<const Pack<T, V>*>(grad)[i];
#pragma unroll
    for (int k = 0; k < V; ++k)
      w.v[k] = from_f32<T>(to_f32(w.v[k]) - lr * to_f32(g.v[k]));
    reinterpret_cast<Pack<T, V>*>(buf)[i] = w;
  }
}

template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_sgd_tail(
    T* __restrict__ buf, const T* __restrict__ grad, int64_t start,
    int64_t total, float lr) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < total)
    buf[i] = from_f32<T>(to_f32(buf[i]) - lr * to_f32(grad[i]));
}

// FedProx: per-element master lookup through the block table.
template <typename T>
__global__ __launch_bounds__(OLS_THREADS) void k_sgd_prox(
    T* __restrict__ buf, const T* __restrict__ grad,
    const T* __restrict__ master, const int64_t* __restrict__ offs,
    int nblocks, int64_t clients, int64_t total, float lr, float mu) {
  constexpr int V = 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * V;
  for (int64_t base = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * V;
       base < total; base += stride) {
    // buf position p lies in block b iff C*offs[b] <= p < C*offs[b+1],
    // i.e. offs[b] <= p/C < offs[b+1] (boundaries are multiples of C)
    int b = find_block(offs, nblocks, base / clients);
    int64_t n = offs[b + 1] - offs[b];
    int64_t blk_base = clients * offs[b];
#pragma unroll
    for (int k = 0; k < V; ++k) {
      int64_t p = base + k;
      if (p >= total) break;
      if (p >= clients * offs[b + 1]) {
        ++b;
        n = offs[b + 1] - offs[b];
        blk_base = clients * offs[b];
      }
      int64_t j = offs[b] + (p - blk_base) % n;
      float w = to_f32(buf[p]);
      float upd = to_f32(grad[p]) + mu * (w - to_f32(master[j]));
      buf[p] = from_f32<T>(w - lr * upd);
    }
  }
}

template <typename T>
static void launch_sgd(T* buf, const T* grad, const T* master,
                       const int64_t* offs, int nblocks, int64_t clients,
                       int64_t total, float lr, float mu,
                       hipStream_t stream) {
  if (mu == 0.0f || master == nullptr) {
    constexpr int V = sizeof(T) == 4 ? 4 : 8;  // 16 B per lane
    int64_t npacks = total / V;
    if (npacks > 0) {
      hipLaunchKernelGGL((k_sgd_plain<T, V>),
                         dim3(ols_grid(npacks, OLS_THREADS)),
                         dim3(OLS_THREADS), 0, stream, buf, grad, npacks, lr);
    }
    int64_t done = npacks * V;
    if (done < total) {
      hipLaunchKernelGGL((k_sgd_tail<T>), dim3(1), dim3(OLS_THREADS), 0,
                         stream, buf, grad, done, total, lr);
    }
  } else {
    hipLaunchKernelGGL((k_sgd_prox<T>),
                       dim3(ols_grid(total, OLS_THREADS * 8)),
                       dim3(OLS_THREADS), 0, stream, buf, grad, master, offs,
                       nblocks, clients, total, lr, mu);
  }
}

extern "C" void ols_fused_sgd_update_flat(
    void* buf, const void* grad, const void* master, const int64_t* offs,
    int nblocks, int64_t clients, int64_t total, float lr, float mu,
    int dtype /*0=f32 1=bf16 2=f16*/, hipStream_t stream) {
  switch (dtype) {
    case 0:
      launch_sgd<float>((float*)buf, (const float*)grad,
                        (const float*)master, offs, nblocks, clients, total,
                        lr, mu, stream);
      break;
    case 1:
      launch_sgd<__hip_bfloat16>((__hip_bfloat16*)buf,
                                 (const __hip_bfloat16*)grad,
                                 (const __hip_bfloat16*)master, offs, nblocks,
                                 clients, total, lr, mu, stream);
      break;
    default:
      launch_sgd<__half>((__half*)buf, (const __half*)grad,
                         (const __half*)master, offs, nblocks, clients, total,
                         lr, mu, stream);
  }
}
