// Torch bindings for the olearning_sim_amd gfx950 kernels.
// Registered as torch.ops.olsim_hip.* (loaded from the in-tree .so by
// olearning_sim_amd/ops/fused.py).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

extern "C" void ols_fused_sgd_update_flat(
    void* buf, const void* grad, const void* master, const int64_t* offs,
    int nblocks, int64_t clients, int64_t total, float lr, float mu,
    int dtype, hipStream_t stream);

extern "C" void ols_weighted_delta_accum_flat(
    float* delta, const void* buf, const void* master, const float* weights,
    const int64_t* offs, int nblocks, int64_t clients, int64_t pglobal,
    float wsum, int dtype, hipStream_t stream);

extern "C" void ols_cross_entropy_fwd_bwd(
    const void* logits, const int64_t* labels, float* loss, void* dlogits,
    int64_t nrows, int64_t K, float inv_n, int dtype, hipStream_t stream);

extern "C" void ols_groupnorm_fwd(const void* x, const void* res, void* y,
                                  float* mean, float* rstd, const void* gamma,
                                  const void* beta, int B, int C, int ch,
                                  int G, int HW, float eps, bool relu,
                                  int layout, int dtype, hipStream_t stream);

extern "C" void ols_mfma_selftest(const void* A, const void* B, float* D,
                                  hipStream_t stream);

extern "C" void ols_conv3x3_fwd(const void* x, const void* w, void* y, int C,
                                int IC, int OC, int B, int H, int W,
                                int stride, hipStream_t stream);
extern "C" void ols_conv3x3_dgrad(const void* dy, const void* w, void* dx,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream);
extern "C" void ols_conv3x3_wgrad(const void* x, const void* dy, void* dw,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream);

extern "C" int ols_conv3x3_v6_ok(int IC, int OC, int B, int H, int W,
                                 int stride);
extern "C" void ols_conv3x3_fwd_p(const void* xp, const void* w, void* y,
                                  int C, int IC, int OC, int B, int H, int W,
                                  int stride, hipStream_t stream);
extern "C" void ols_conv3x3_dgrad_p(const void* dyp, const void* w, void* dx,
                                    int C, int IC, int OC, int B, int H,
                                    int W, int stride, hipStream_t stream);
extern "C" void ols_conv3x3_wgrad_p(const void* xp, const void* dy, void* dw,
                                    int C, int IC, int OC, int B, int H,
                                    int W, int stride, hipStream_t stream);

extern "C" void ols_conv5x5_fwd(const void* x, const void* w, const void* b,
                                void* y, const int* ntab, int C, int IC,
                                int OC, int B, int H, int W, int relu,
                                hipStream_t stream);
extern "C" void ols_conv5x5_dgrad(const void* dyp, const void* w, void* dx,
                                  const int* ntab, int C, int IC, int OC,
                                  int B, int H, int W, hipStream_t stream);
extern "C" void ols_conv5x5_wgrad(const void* x, const void* dy, void* dw,
                                  float* part, const int* ntab, int C,
                                  int IC, int OC, int B, int H, int W,
                                  hipStream_t stream);

extern "C" void ols_pool2x2_fwd(const void* x, void* y, unsigned char* arg,
                                int64_t planes, int OH, int OW, int dtype,
                                hipStream_t stream);
extern "C" void ols_pool2x2_bwd(const void* dy, const unsigned char* arg,
                                void* dx, int64_t planes, int OH, int OW,
                                int dtype, hipStream_t stream);

extern "C" void ols_subsample2_fwd(const void* x, void* y, int64_t planes,
                                   int OH, int OW, int dtype,
                                   hipStream_t stream);
extern "C" void ols_subsample2_bwd(const void* dy, void* dx, int64_t planes,
                                   int H, int W, int dtype,
                                   hipStream_t stream);

extern "C" void ols_transpose2d(const void* in, void* out, int64_t B, int M,
                                int N, int dtype, hipStream_t stream);

extern "C" void ols_pad2d(const void* in, void* out, int64_t planes, int H,
                          int W, int pad, int dtype, hipStream_t stream);

extern "C" void ols_replicate(const void* src, void* dst, int64_t clients,
                              int64_t n, int dtype, hipStream_t stream);

extern "C" void ols_synth_batch(const float* x, const int64_t* y, void* out,
                                int64_t rows, int64_t batch, int64_t n,
                                float s, float t, hipStream_t stream);

extern "C" void ols_relu_mask(const void* dy, const void* y, void* out,
                              int64_t n, int dtype, hipStream_t stream);

extern "C" void ols_layernorm_fwd(const void* x, const void* gamma,
                                  const void* beta, void* y, float* mean,
                                  float* rstd, int64_t rows, int H,
                                  int64_t rows_per_client, float eps,
                                  int dtype, hipStream_t stream);
extern "C" void ols_layernorm_bwd(const void* x, const void* dy,
                                  const void* gamma, const float* mean,
                                  const float* rstd, void* dx, float* dgamma,
                                  float* dbeta, int64_t rows, int H,
                                  int64_t rows_per_client, int dtype,
                                  hipStream_t stream);

extern "C" void ols_groupnorm_bwd(const void* x, const void* y,
                                  const void* dy, void* dx, void* dres,
                                  const float* mean, const float* rstd,
                                  const void* gamma, float* dgamma,
                                  float* dbeta, int B, int C, int ch, int G,
                                  int HW, bool relu, int layout, int dtype,
                                  hipStream_t stream);

namespace {

int dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default:
      TORCH_CHECK(false, "olsim_hip: unsupported dtype ", t.scalar_type());
  }
}

void fused_sgd_update_flat(at::Tensor buf, at::Tensor grad,
                           at::Tensor global_flat, at::Tensor offsets,
                           int64_t clients, double lr, double mu) {
  TORCH_CHECK(buf.is_cuda() && buf.is_contiguous(), "buf must be GPU+contig");
  TORCH_CHECK(grad.sizes() == buf.sizes() && grad.scalar_type() == buf.scalar_type());
  const bool prox = mu != 0.0 && global_flat.numel() > 0;
  if (prox) {
    TORCH_CHECK(offsets.numel() >= 2 && offsets.scalar_type() == at::kLong);
    TORCH_CHECK(global_flat.scalar_type() == buf.scalar_type());
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  ols_fused_sgd_update_flat(
      buf.data_ptr(), grad.data_ptr(),
      prox ? global_flat.data_ptr() : nullptr,
      prox ? offsets.data_ptr<int64_t>() : nullptr,
      prox ? (int)(offsets.numel() - 1) : 0, clients, buf.numel(),
      (float)lr, (float)mu, dtype_code(buf), stream.stream());
}

void weighted_delta_accum_flat(at::Tensor delta, at::Tensor buf,
                               at::Tensor global_flat, at::Tensor weights,
                               at::Tensor offsets, int64_t clients,
                               double wsum) {
  TORCH_CHECK(delta.is_cuda() && delta.scalar_type() == at::kFloat &&
              delta.is_contiguous());
  TORCH_CHECK(buf.is_contiguous() && global_flat.is_contiguous());
  TORCH_CHECK(buf.scalar_type() == global_flat.scalar_type());
  TORCH_CHECK(weights.scalar_type() == at::kFloat && weights.numel() == clients);
  TORCH_CHECK(offsets.scalar_type() == at::kLong && offsets.numel() >= 2);
  TORCH_CHECK(buf.numel() == clients * global_flat.numel());
  auto stream = at::cuda::getCurrentCUDAStream();
  ols_weighted_delta_accum_flat(
      delta.data_ptr<float>(), buf.data_ptr(), global_flat.data_ptr(),
      weights.data_ptr<float>(), offsets.data_ptr<int64_t>(),
      (int)(offsets.numel() - 1), clients, global_flat.numel(), (float)wsum,
      dtype_code(buf), stream.stream());
}

std::tuple<at::Tensor, at::Tensor> cross_entropy_fwd_bwd(at::Tensor logits,
                                                         at::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.scalar_type() == at::kLong &&
              labels.numel() == logits.size(0));
  int64_t n = logits.size(0), k = logits.size(1);
  auto loss = at::empty({n}, logits.options().dtype(at::kFloat));
  auto dlogits = at::empty_like(logits);
  auto stream = at::cuda::getCurrentCUDAStream();
  ols_cross_entropy_fwd_bwd(
      logits.data_ptr(), labels.contiguous().data_ptr<int64_t>(),
      loss.data_ptr<float>(), dlogits.data_ptr(), n, k, 1.0f / (float)n,
      dtype_code(logits), stream.stream());
  return {loss, dlogits};
}

int gn_dtype(const at::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == at::kFloat || t.scalar_type() == at::kBFloat16,
              "groupnorm: f32/bf16 only");
  return t.scalar_type() == at::kFloat ? 0 : 1;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> groupnorm_fwd(
    at::Tensor x, at::Tensor res, at::Tensor gamma, at::Tensor beta,
    int64_t clients, int64_t groups, double eps, bool relu) {
  // dim 4: [B, C*ch, H, W] (layout 0); dim 5: [C, ch, B, H, W] (layout 1)
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              (x.dim() == 4 || x.dim() == 5));
  const int layout = x.dim() == 4 ? 0 : 1;
  int64_t B, ch, HW;
  if (layout == 0) {
    B = x.size(0); ch = x.size(1) / clients; HW = x.size(2) * x.size(3);
    TORCH_CHECK(x.size(1) % clients == 0);
  } else {
    TORCH_CHECK(x.size(0) == clients);
    ch = x.size(1); B = x.size(2); HW = x.size(3) * x.size(4);
  }
  TORCH_CHECK(ch % groups == 0);
  TORCH_CHECK(ch / groups <= 128, "groupnorm: ch/G > 128 unsupported");
  TORCH_CHECK(gamma.is_contiguous() && beta.is_contiguous());
  const bool has_res = res.numel() > 0;
  if (has_res) TORCH_CHECK(res.is_contiguous() && res.sizes() == x.sizes());
  auto y = at::empty_like(x);
  const int64_t ngroups = B * clients * groups;
  auto mean = at::empty({ngroups}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({ngroups}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  ols_groupnorm_fwd(x.data_ptr(), has_res ? res.data_ptr() : nullptr,
                    y.data_ptr(), mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), gamma.data_ptr(), beta.data_ptr(),
                    (int)B, (int)clients, (int)ch, (int)groups, (int)HW,
                    (float)eps, relu, layout, gn_dtype(x), stream.stream());
  return {y, mean, rstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> groupnorm_bwd(
    at::Tensor x, at::Tensor y, at::Tensor dy, at::Tensor mean,
    at::Tensor rstd, at::Tensor gamma, int64_t clients, int64_t groups,
    bool has_res, bool relu) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  auto dyc = dy.contiguous();
  const int layout = x.dim() == 4 ? 0 : 1;
  int64_t B, ch, HW;
  if (layout == 0) {
    B = x.size(0); ch = x.size(1) / clients; HW = x.size(2) * x.size(3);
  } else {
    ch = x.size(1); B = x.size(2); HW = x.size(3) * x.size(4);
  }
  auto dx = at::empty_like(x);
  auto dres = has_res ? at::empty_like(x)
                      : at::empty({0}, x.options());
  auto dgamma = at::zeros({clients, ch}, x.options().dtype(at::kFloat));
  auto dbeta = at::zeros({clients, ch}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  ols_groupnorm_bwd(x.data_ptr(), y.data_ptr(), dyc.data_ptr(),
                    dx.data_ptr(), has_res ? dres.data_ptr() : nullptr,
                    mean.data_ptr<float>(), rstd.data_ptr<float>(),
                    gamma.data_ptr(), dgamma.data_ptr<float>(),
                    dbeta.data_ptr<float>(), (int)B, (int)clients, (int)ch,
                    (int)groups, (int)HW, relu, layout, gn_dtype(x),
                    stream.stream());
  return {dx, dres, dgamma, dbeta};
}

at::Tensor mfma_selftest(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) &&
              B.sizes() == at::IntArrayRef({32, 16}));
  auto D = at::empty({16, 16}, A.options().dtype(at::kFloat));
  ols_mfma_selftest(A.contiguous().data_ptr(), B.contiguous().data_ptr(),
                    D.data_ptr<float>(),
                    at::cuda::getCurrentCUDAStream().stream());
  return D;
}

// x: [C, IC, B, H, W] bf16; w: [C, OC, IC, 3, 3] bf16 -> y [C, OC, B, OH, OW]
at::Tensor conv3x3_fwd(at::Tensor x, at::Tensor w, int64_t stride) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 5);
  TORCH_CHECK(w.is_contiguous() && w.dim() == 5 && w.size(3) == 3);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  int C = x.size(0), IC = x.size(1), B = x.size(2), H = x.size(3),
      W = x.size(4), OC = w.size(1);
  TORCH_CHECK(w.size(0) == C && w.size(2) == IC);
  int OH = (H + stride - 1) / stride, OW = (W + stride - 1) / stride;
  auto y = at::empty({C, OC, B, OH, OW}, x.options());
  ols_conv3x3_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), C, IC, OC, B, H,
                  W, (int)stride, at::cuda::getCurrentCUDAStream().stream());
  return y;
}

at::Tensor conv3x3_dgrad(at::Tensor dy, at::Tensor w, int64_t H, int64_t W,
                         int64_t stride) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 5);
  int C = dy.size(0), OC = dy.size(1), B = dy.size(2);
  int IC = w.size(2);
  auto dx = at::empty({C, IC, B, H, W}, dy.options());
  ols_conv3x3_dgrad(dy.data_ptr(), w.contiguous().data_ptr(), dx.data_ptr(),
                    C, IC, OC, B, (int)H, (int)W, (int)stride,
                    at::cuda::getCurrentCUDAStream().stream());
  return dx;
}

at::Tensor conv3x3_wgrad(at::Tensor x, at::Tensor dy, int64_t stride) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  int C = x.size(0), IC = x.size(1), B = x.size(2), H = x.size(3),
      W = x.size(4), OC = dy.size(1);
  auto dw = at::empty({C, OC, IC, 3, 3}, x.options());
  ols_conv3x3_wgrad(x.data_ptr(), dy.data_ptr(), dw.data_ptr(), C, IC,
                    OC, B, H, W, (int)stride,
                    at::cuda::getCurrentCUDAStream().stream());
  return dw;
}

// ---- v6 padded family (client_conv2.hip) --------------------------------
// x_pad: [C, IC, B, H+2, W+2] bf16 (1-element zero halo per plane)

bool conv3x3_v6_ok(int64_t IC, int64_t OC, int64_t B, int64_t H, int64_t W,
                   int64_t stride) {
  return ols_conv3x3_v6_ok((int)IC, (int)OC, (int)B, (int)H, (int)W,
                           (int)stride) != 0;
}

at::Tensor conv3x3_fwd_p(at::Tensor xp, at::Tensor w, int64_t stride) {
  TORCH_CHECK(xp.is_cuda() && xp.is_contiguous() && xp.dim() == 5);
  TORCH_CHECK(w.is_contiguous() && w.dim() == 5 && w.size(3) == 3);
  TORCH_CHECK(xp.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  int C = xp.size(0), IC = xp.size(1), B = xp.size(2),
      H = xp.size(3) - 2, W = xp.size(4) - 2, OC = w.size(1);
  TORCH_CHECK(w.size(0) == C && w.size(2) == IC);
  TORCH_CHECK(ols_conv3x3_v6_ok(IC, OC, B, H, W, (int)stride),
              "shape unsupported by the v6 conv path");
  int OH = (H + stride - 1) / stride, OW = (W + stride - 1) / stride;
  auto y = at::empty({C, OC, B, OH, OW}, xp.options());
  ols_conv3x3_fwd_p(xp.data_ptr(), w.data_ptr(), y.data_ptr(), C, IC, OC, B,
                    H, W, (int)stride,
                    at::cuda::getCurrentCUDAStream().stream());
  return y;
}

// dy_pad: [C, OC, B, OH+2, OW+2]; returns dx [C, IC, B, H, W]
at::Tensor conv3x3_dgrad_p(at::Tensor dyp, at::Tensor w, int64_t H, int64_t W,
                           int64_t stride) {
  TORCH_CHECK(dyp.is_cuda() && dyp.is_contiguous() && dyp.dim() == 5);
  TORCH_CHECK(w.is_contiguous() && w.dim() == 5);
  int C = dyp.size(0), OC = dyp.size(1), B = dyp.size(2);
  int IC = w.size(2);
  TORCH_CHECK(ols_conv3x3_v6_ok(IC, OC, B, H, W, (int)stride),
              "shape unsupported by the v6 conv path");
  auto dx = at::empty({C, IC, B, H, W}, dyp.options());
  ols_conv3x3_dgrad_p(dyp.data_ptr(), w.data_ptr(), dx.data_ptr(), C, IC, OC,
                      B, (int)H, (int)W, (int)stride,
                      at::cuda::getCurrentCUDAStream().stream());
  return dx;
}

// ---- 5x5 VALID family (client_conv5.hip, LeNet) -------------------------
// ntab: int32 [B*OH*OW] plane offsets (b*H*W + oh*W + ow) built host-side

at::Tensor conv5x5_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                       at::Tensor ntab, bool relu) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 5);
  TORCH_CHECK(w.is_contiguous() && w.dim() == 5 && w.size(3) == 5);
  TORCH_CHECK(ntab.is_cuda() && ntab.scalar_type() == at::kInt &&
              ntab.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  int C = x.size(0), IC = x.size(1), B = x.size(2), H = x.size(3),
      W = x.size(4), OC = w.size(1);
  TORCH_CHECK(w.size(0) == C && w.size(2) == IC && OC <= 16);
  TORCH_CHECK(ntab.numel() == (int64_t)B * (H - 4) * (W - 4));
  auto y = at::empty({C, OC, B, H - 4, W - 4}, x.options());
  ols_conv5x5_fwd(x.data_ptr(), w.data_ptr(), bias.contiguous().data_ptr(),
                  y.data_ptr(), ntab.data_ptr<int>(), C, IC, OC, B, H, W,
                  relu ? 1 : 0, at::cuda::getCurrentCUDAStream().stream());
  return y;
}

// dyp: [C,OC,B,OH+8,OW+8] (pad 4); returns dx [C,IC,B,H,W]
at::Tensor conv5x5_dgrad(at::Tensor dyp, at::Tensor w, at::Tensor ntab,
                         int64_t H, int64_t W) {
  TORCH_CHECK(dyp.is_cuda() && dyp.is_contiguous() && dyp.dim() == 5);
  TORCH_CHECK(ntab.is_cuda() && ntab.scalar_type() == at::kInt);
  int C = dyp.size(0), OC = dyp.size(1), B = dyp.size(2);
  int IC = w.size(2);
  TORCH_CHECK(IC <= 16 && ntab.numel() == (int64_t)B * H * W);
  auto dx = at::empty({C, IC, B, H, W}, dyp.options());
  ols_conv5x5_dgrad(dyp.data_ptr(), w.contiguous().data_ptr(), dx.data_ptr(),
                    ntab.data_ptr<int>(), C, IC, OC, B, (int)H, (int)W,
                    at::cuda::getCurrentCUDAStream().stream());
  return dx;
}

at::Tensor conv5x5_wgrad(at::Tensor x, at::Tensor dy, at::Tensor ntab) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  TORCH_CHECK(ntab.is_cuda() && ntab.scalar_type() == at::kInt);
  int C = x.size(0), IC = x.size(1), B = x.size(2), H = x.size(3),
      W = x.size(4), OC = dy.size(1);
  TORCH_CHECK(OC <= 16 && ((int64_t)B * (H - 4) * (W - 4)) % 32 == 0);
  auto dw = at::empty({C, OC, IC, 5, 5}, x.options());
  // two-stage direct path (client_conv5.hip k_conv5x5_wgrad_part):
  // per-(client, b) fp32 partials + a B-reduction.  MEASURED NEGATIVE
  // (fedprox 96.9 vs 49.4 ms/round with the MFMA wgrad): the [C, B,
  // ntaps] fp32 partial buffer is ~1 GB per call and the per-tap
  // q-chains stay latency-serial.  wgrad is the one conv5 direction
  // with a real GEMM shape (K = B*OH*OW = 1600) — the MFMA kernel
  // stays.  Kept behind OLSIM_CONV5=direct for the record.
  const size_t lds1 = ((size_t)IC * H * W
                       + (size_t)OC * (H - 4) * (W - 4)) * sizeof(short);
  const char* c5 = getenv("OLSIM_CONV5");
  at::Tensor part;
  float* partp = nullptr;
  if (lds1 <= 32768 && c5 != nullptr && c5[0] == 'd') {
    part = at::empty({(int64_t)C * B * OC * IC * 25},
                     x.options().dtype(at::kFloat));
    partp = part.data_ptr<float>();
  }
  ols_conv5x5_wgrad(x.data_ptr(), dy.data_ptr(), dw.data_ptr(), partp,
                    ntab.data_ptr<int>(), C, IC, OC, B, H, W,
                    at::cuda::getCurrentCUDAStream().stream());
  return dw;
}

at::Tensor conv3x3_wgrad_p(at::Tensor xp, at::Tensor dy, int64_t stride) {
  TORCH_CHECK(xp.is_cuda() && xp.is_contiguous() && dy.is_contiguous());
  int C = xp.size(0), IC = xp.size(1), B = xp.size(2),
      H = xp.size(3) - 2, W = xp.size(4) - 2, OC = dy.size(1);
  TORCH_CHECK(ols_conv3x3_v6_ok(IC, OC, B, H, W, (int)stride),
              "shape unsupported by the v6 conv path");
  auto dw = at::empty({C, OC, IC, 3, 3}, xp.options());
  ols_conv3x3_wgrad_p(xp.data_ptr(), dy.data_ptr(), dw.data_ptr(), C,
                      IC, OC, B, H, W, (int)stride,
                      at::cuda::getCurrentCUDAStream().stream());
  return dw;
}

// ---- 2x2 max pool (pool2x2.hip) -----------------------------------------
// x: [..., H, W] with H, W even; pooled over the last two dims

std::tuple<at::Tensor, at::Tensor> pool2x2_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() >= 2);
  int H = x.size(-2), W = x.size(-1);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0);
  int64_t planes = x.numel() / ((int64_t)H * W);
  auto sizes = x.sizes().vec();
  sizes[sizes.size() - 2] = H / 2;
  sizes[sizes.size() - 1] = W / 2;
  auto y = at::empty(sizes, x.options());
  auto arg = at::empty(sizes, x.options().dtype(at::kByte));
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_pool2x2_fwd(x.data_ptr(), y.data_ptr(),
                  (unsigned char*)arg.data_ptr(), planes, H / 2, W / 2, dt,
                  at::cuda::getCurrentCUDAStream().stream());
  return {y, arg};
}

at::Tensor pool2x2_bwd(at::Tensor dy, at::Tensor arg) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() >= 2);
  int OH = dy.size(-2), OW = dy.size(-1);
  int64_t planes = dy.numel() / ((int64_t)OH * OW);
  auto sizes = dy.sizes().vec();
  sizes[sizes.size() - 2] = OH * 2;
  sizes[sizes.size() - 1] = OW * 2;
  auto dx = at::empty(sizes, dy.options());
  int dt = dy.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_pool2x2_bwd(dy.data_ptr(), (unsigned char*)arg.contiguous().data_ptr(),
                  dx.data_ptr(), planes, OH, OW, dt,
                  at::cuda::getCurrentCUDAStream().stream());
  return dx;
}

// ---- per-client LayerNorm (layernorm.hip) -------------------------------
// x [C, N, H] (N rows per client); gamma/beta [C, H]

std::tuple<at::Tensor, at::Tensor, at::Tensor> layernorm_fwd(
    at::Tensor x, at::Tensor gamma, at::Tensor beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(gamma.is_contiguous() && beta.is_contiguous());
  int64_t C = x.size(0), N = x.size(1);
  int H = x.size(2);
  TORCH_CHECK(H % 8 == 0 && N % 4 == 0, "layernorm_fwd: H%8, N%4");
  int64_t rows = C * N;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_layernorm_fwd(x.data_ptr(), gamma.data_ptr(), beta.data_ptr(),
                    y.data_ptr(), mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), rows, H, N, (float)eps, dt,
                    at::cuda::getCurrentCUDAStream().stream());
  return {y, mean, rstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> layernorm_bwd(
    at::Tensor x, at::Tensor dy, at::Tensor gamma, at::Tensor mean,
    at::Tensor rstd) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  int64_t C = x.size(0), N = x.size(1);
  int H = x.size(2);
  int64_t rows = C * N;
  auto dx = at::empty_like(x);
  auto dgamma = at::zeros({C, (int64_t)H}, x.options().dtype(at::kFloat));
  auto dbeta = at::zeros({C, (int64_t)H}, x.options().dtype(at::kFloat));
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_layernorm_bwd(x.data_ptr(), dy.data_ptr(),
                    gamma.contiguous().data_ptr(), mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), dx.data_ptr(),
                    dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), rows,
                    H, N, dt, at::cuda::getCurrentCUDAStream().stream());
  return {dx, dgamma, dbeta};
}

// ---- 2x stride subsample (pool2x2.hip) ----------------------------------

at::Tensor subsample2(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() >= 2);
  int H = x.size(-2), W = x.size(-1);
  TORCH_CHECK(H % 2 == 0 && W % 4 == 0);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 ||
              x.scalar_type() == at::kFloat);
  int64_t planes = x.numel() / ((int64_t)H * W);
  auto sizes = x.sizes().vec();
  sizes[sizes.size() - 2] = H / 2;
  sizes[sizes.size() - 1] = W / 2;
  auto y = at::empty(sizes, x.options());
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_subsample2_fwd(x.data_ptr(), y.data_ptr(), planes, H / 2, W / 2, dt,
                     at::cuda::getCurrentCUDAStream().stream());
  return y;
}

at::Tensor subsample2_bwd(at::Tensor dy, int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() >= 2);
  TORCH_CHECK(dy.size(-2) == H / 2 && dy.size(-1) == W / 2);
  int64_t planes = dy.numel() / ((int64_t)(H / 2) * (W / 2));
  auto sizes = dy.sizes().vec();
  sizes[sizes.size() - 2] = H;
  sizes[sizes.size() - 1] = W;
  auto dx = at::empty(sizes, dy.options());
  int dt = dy.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_subsample2_bwd(dy.data_ptr(), dx.data_ptr(), planes, (int)H, (int)W,
                     dt, at::cuda::getCurrentCUDAStream().stream());
  return dx;
}

// ---- client replica broadcast (replicate.hip) ---------------------------

at::Tensor replicate(at::Tensor src, int64_t clients) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous() && clients >= 1);
  TORCH_CHECK(src.numel() % 8 == 0);
  TORCH_CHECK(src.scalar_type() == at::kBFloat16 ||
              src.scalar_type() == at::kFloat);
  auto sizes = src.sizes().vec();
  sizes.insert(sizes.begin(), clients);
  auto dst = at::empty(sizes, src.options());
  int dt = src.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_replicate(src.data_ptr(), dst.data_ptr(), clients, src.numel(), dt,
                at::cuda::getCurrentCUDAStream().stream());
  return dst;
}

at::Tensor synth_batch(at::Tensor x, at::Tensor y, double s, double t) {
  // x [B, n] fp32, y [rows] int64 (rows = C*B) -> out [rows, n] bf16
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2);
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.size(1) % 8 == 0);
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() &&
              y.scalar_type() == at::kLong);
  int64_t rows = y.numel(), batch = x.size(0), n = x.size(1);
  TORCH_CHECK(rows % batch == 0);
  auto out = at::empty({rows, n}, x.options().dtype(at::kBFloat16));
  ols_synth_batch(x.data_ptr<float>(), y.data_ptr<int64_t>(),
                  out.data_ptr(), rows, batch, n, (float)s, (float)t,
                  at::cuda::getCurrentCUDAStream().stream());
  return out;
}

// ---- fused relu-mask backward (replicate.hip) ---------------------------

at::Tensor relu_mask(at::Tensor dy, at::Tensor y) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(dy.sizes() == y.sizes() && dy.scalar_type() == y.scalar_type());
  TORCH_CHECK(dy.numel() % 8 == 0);
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 ||
              dy.scalar_type() == at::kFloat);
  auto out = at::empty_like(dy);
  int dt = dy.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_relu_mask(dy.data_ptr(), y.data_ptr(), out.data_ptr(), dy.numel(), dt,
                at::cuda::getCurrentCUDAStream().stream());
  return out;
}

// ---- zero-pad trailing 2 dims (pad2d.hip) -------------------------------

at::Tensor pad2d(at::Tensor x, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() >= 2);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 ||
              x.scalar_type() == at::kHalf);
  int H = x.size(-2), W = x.size(-1);
  TORCH_CHECK(W % 2 == 0 && pad >= 1);
  int64_t planes = x.numel() / ((int64_t)H * W);
  auto sizes = x.sizes().vec();
  sizes[sizes.size() - 2] = H + 2 * pad;
  sizes[sizes.size() - 1] = W + 2 * pad;
  auto y = at::empty(sizes, x.options());
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 2;
  ols_pad2d(x.data_ptr(), y.data_ptr(), planes, H, W, (int)pad, dt,
            at::cuda::getCurrentCUDAStream().stream());
  return y;
}

// ---- batched 2-D transpose (transpose.hip) ------------------------------

at::Tensor transpose2d(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 ||
              x.scalar_type() == at::kFloat);
  int64_t B = x.size(0);
  int M = x.size(1), N = x.size(2);
  auto y = at::empty({B, (int64_t)N, (int64_t)M}, x.options());
  int dt = x.scalar_type() == at::kBFloat16 ? 1 : 0;
  ols_transpose2d(x.data_ptr(), y.data_ptr(), B, M, N, dt,
                  at::cuda::getCurrentCUDAStream().stream());
  return y;
}

}  // namespace

TORCH_LIBRARY(olsim_hip, m) {
  m.def("fused_sgd_update_flat(Tensor(a!) buf, Tensor grad, Tensor global_flat, "
        "Tensor offsets, int clients, float lr, float mu) -> ()");
  m.def("weighted_delta_accum_flat(Tensor(a!) delta, Tensor buf, "
        "Tensor global_flat, Tensor weights, Tensor offsets, int clients, float wsum) -> ()");
  m.def("cross_entropy_fwd_bwd(Tensor logits, Tensor labels) -> (Tensor, Tensor)");
  m.def("groupnorm_fwd(Tensor x, Tensor res, Tensor gamma, Tensor beta, "
        "int clients, int groups, float eps, bool relu) -> (Tensor, Tensor, Tensor)");
  m.def("groupnorm_bwd(Tensor x, Tensor y, Tensor dy, Tensor mean, "
        "Tensor rstd, Tensor gamma, int clients, int groups, bool has_res, "
        "bool relu) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("mfma_selftest(Tensor A, Tensor B) -> Tensor");
  m.def("conv3x3_fwd(Tensor x, Tensor w, int stride) -> Tensor");
  m.def("conv3x3_dgrad(Tensor dy, Tensor w, int H, int W, int stride) -> Tensor");
  m.def("conv3x3_wgrad(Tensor x, Tensor dy, int stride) -> Tensor");
  m.def("conv3x3_v6_ok(int IC, int OC, int B, int H, int W, int stride) -> bool",
        &conv3x3_v6_ok);
  m.def("conv3x3_fwd_p(Tensor xp, Tensor w, int stride) -> Tensor");
  m.def("conv3x3_dgrad_p(Tensor dyp, Tensor w, int H, int W, int stride) -> Tensor");
  m.def("conv3x3_wgrad_p(Tensor xp, Tensor dy, int stride) -> Tensor");
  m.def("conv5x5_fwd(Tensor x, Tensor w, Tensor bias, Tensor ntab, bool relu) -> Tensor");
  m.def("conv5x5_dgrad(Tensor dyp, Tensor w, Tensor ntab, int H, int W) -> Tensor");
  m.def("conv5x5_wgrad(Tensor x, Tensor dy, Tensor ntab) -> Tensor");
  m.def("pool2x2_fwd(Tensor x) -> (Tensor, Tensor)");
  m.def("pool2x2_bwd(Tensor dy, Tensor arg) -> Tensor");
  m.def("transpose2d(Tensor x) -> Tensor");
  m.def("pad2d(Tensor x, int pad) -> Tensor");
  m.def("relu_mask(Tensor dy, Tensor y) -> Tensor");
  m.def("replicate(Tensor src, int clients) -> Tensor");
  m.def("subsample2(Tensor x) -> Tensor");
  m.def("subsample2_bwd(Tensor dy, int H, int W) -> Tensor");
  m.def("synth_batch(Tensor x, Tensor y, float s, float t) -> Tensor");
  m.def("layernorm_fwd(Tensor x, Tensor gamma, Tensor beta, float eps) -> (Tensor, Tensor, Tensor)");
  m.def("layernorm_bwd(Tensor x, Tensor dy, Tensor gamma, Tensor mean, Tensor rstd) -> (Tensor, Tensor, Tensor)");
}

TORCH_LIBRARY_IMPL(olsim_hip, CUDA, m) {
  m.impl("fused_sgd_update_flat", &fused_sgd_update_flat);
  m.impl("weighted_delta_accum_flat", &weighted_delta_accum_flat);
  m.impl("cross_entropy_fwd_bwd", &cross_entropy_fwd_bwd);
  m.impl("groupnorm_fwd", &groupnorm_fwd);
  m.impl("groupnorm_bwd", &groupnorm_bwd);
  m.impl("mfma_selftest", &mfma_selftest);
  m.impl("conv3x3_fwd", &conv3x3_fwd);
  m.impl("conv3x3_dgrad", &conv3x3_dgrad);
  m.impl("conv3x3_wgrad", &conv3x3_wgrad);
  m.impl("conv3x3_fwd_p", &conv3x3_fwd_p);
  m.impl("conv3x3_dgrad_p", &conv3x3_dgrad_p);
  m.impl("conv3x3_wgrad_p", &conv3x3_wgrad_p);
  m.impl("conv5x5_fwd", &conv5x5_fwd);
  m.impl("conv5x5_dgrad", &conv5x5_dgrad);
  m.impl("conv5x5_wgrad", &conv5x5_wgrad);
  m.impl("pool2x2_fwd", &pool2x2_fwd);
  m.impl("pool2x2_bwd", &pool2x2_bwd);
  m.impl("transpose2d", &transpose2d);
  m.impl("pad2d", &pad2d);
  m.impl("relu_mask", &relu_mask);
  m.impl("replicate", &replicate);
  m.impl("subsample2", &subsample2);
  m.impl("subsample2_bwd", &subsample2_bwd);
  m.impl("synth_batch", &synth_batch);
  m.impl("layernorm_fwd", &layernorm_fwd);
  m.impl("layernorm_bwd", &layernorm_bwd);
}
