// Torch extension bindings for the MI355X HIP kernels (module `_C`).
// Compiled under PyTorch-ROCm; kernels live in the .hip translation units.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

extern "C" void launch_normalize_u8_to_bf16(
    const uint8_t* in, void* out, const float* scale_dev,
    const float* shift_dev, long long n, hipStream_t stream);

// in:  uint8 contiguous [N,H,W,3] (NHWC)
// out: bf16 tensor with identical memory order (channels_last NCHW view
//      is constructed Python-side)
// scale/shift: float32 device tensors of 3 elements:
//      scale[c] = 1/(255*std[c]), shift[c] = -mean[c]/std[c]
void normalize_u8_to_bf16(torch::Tensor in, torch::Tensor out,
                          torch::Tensor scale, torch::Tensor shift) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda(), "tensors must be on GPU");
  TORCH_CHECK(in.scalar_type() == torch::kUInt8, "input must be uint8");
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "output must be bf16");
  TORCH_CHECK(in.is_contiguous(), "input must be contiguous (NHWC)");
  TORCH_CHECK(in.numel() == out.numel(), "element count mismatch");
  TORCH_CHECK(scale.is_cuda() && shift.is_cuda() && scale.numel() == 3 &&
                  shift.numel() == 3,
              "scale/shift must be 3-element device float tensors");
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32 &&
              shift.scalar_type() == torch::kFloat32,
              "scale/shift must be float32");
  auto stream = at::cuda::getCurrentHIPStream();
  launch_normalize_u8_to_bf16(
      in.data_ptr<uint8_t>(), out.data_ptr(),
      scale.data_ptr<float>(), shift.data_ptr<float>(),
      (long long)in.numel(), stream.stream());
}

extern "C" void launch_groupfit_eval(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2, const int* orders,
    float* mse, unsigned char* statusv,
    const float* b0, const float* b1, const float* b2,
    const float* w0, const float* w1, const float* w2,
    int T, long long G, int S, int C,
    int KX, hipStream_t stream);

extern "C" void launch_diff_center(const float* yT, float* wc, float* wm,
                                   int T, long long G, int S, int d,
                                   hipStream_t stream);

extern "C" void launch_exog_project_mfma(const float* P, const float* wc,
                                         float* beta, int n, long long G,
                                         int KX, hipStream_t stream);

extern "C" void launch_groupfit_final(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2,
    const int* best_order, float* fitted, float* params,
    unsigned char* statusv,
    const float* b0, const float* b1, const float* b2,
    const float* w0, const float* w1, const float* w2,
    int T, long long G, int KX, hipStream_t stream);

extern "C" void launch_stem_conv_fwd(const void* x, const void* w, void* out,
                                     void* wp_scratch,
                                     int Nb, int H, int W, int HO, int WO,
                                     hipStream_t stream, int phase_mask);
extern "C" void launch_stem_conv_wrw(const void* x, const void* dy,
                                     float* dw_f32, void* dw_bf16,
                                     int Nb, int H, int W, int HO, int WO,
                                     hipStream_t stream, int phase_mask);
extern "C" void launch_conv3x3_wrw(const void* x, const void* dy,
                                   float* part, void* dw_bf16,
                                   int Nb, int H, int W, int C,
                                   int chunks, hipStream_t stream,
                                   int phase_mask);
extern "C" int conv3x3_wrw_chunks(int Nb, int H, int C, int chunks_req);

static void _check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous f32 on GPU");
}

// yT: [T][G] time-major; xc{d}: [T-d][KX]; pj{d}: [KX][S-d];
// orders: [C][3] i32; mse: [C][G] f32 out; status: [C][G] u8 out
void groupfit_eval(torch::Tensor yT, torch::Tensor xc0, torch::Tensor xc1,
                   torch::Tensor xc2, torch::Tensor pj0, torch::Tensor pj1,
                   torch::Tensor pj2, torch::Tensor orders,
                   torch::Tensor mse, torch::Tensor status, int64_t S,
                   torch::Tensor b0, torch::Tensor b1, torch::Tensor b2,
                   torch::Tensor w0, torch::Tensor w1, torch::Tensor w2) {
  _check_f32(yT, "yT"); _check_f32(xc0, "xc0"); _check_f32(xc1, "xc1");
  _check_f32(xc2, "xc2"); _check_f32(pj0, "pj0"); _check_f32(pj1, "pj1");
  _check_f32(pj2, "pj2"); _check_f32(mse, "mse");
  TORCH_CHECK(orders.scalar_type() == torch::kInt32 && orders.is_cuda());
  TORCH_CHECK(status.scalar_type() == torch::kUInt8 && status.is_cuda());
  int T = yT.size(0);
  long long G = yT.size(1);
  int C = orders.size(0);
  TORCH_CHECK(S > 8 && S <= T, "need 8 < S <= T");
  TORCH_CHECK(mse.size(0) == C && mse.size(1) == G);
  auto stream = at::cuda::getCurrentHIPStream();
  auto fp = [](torch::Tensor& t) -> const float* {
    return t.numel() ? t.data_ptr<float>() : nullptr;
  };
  launch_groupfit_eval(
      yT.data_ptr<float>(), xc0.data_ptr<float>(), xc1.data_ptr<float>(),
      xc2.data_ptr<float>(), pj0.data_ptr<float>(), pj1.data_ptr<float>(),
      pj2.data_ptr<float>(), orders.data_ptr<int>(), mse.data_ptr<float>(),
      status.data_ptr<uint8_t>(),
      fp(b0), fp(b1), fp(b2), fp(w0), fp(w1), fp(w2),
      T, G, (int)S, C, (int)xc0.size(1),
      stream.stream());
}

void groupfit_final(torch::Tensor yT, torch::Tensor xc0, torch::Tensor xc1,
                    torch::Tensor xc2, torch::Tensor pj0, torch::Tensor pj1,
                    torch::Tensor pj2, torch::Tensor best_order,
                    torch::Tensor fitted, torch::Tensor params,
                    torch::Tensor status,
                    torch::Tensor b0, torch::Tensor b1, torch::Tensor b2,
                    torch::Tensor w0, torch::Tensor w1, torch::Tensor w2) {
  _check_f32(yT, "yT"); _check_f32(fitted, "fitted");
  _check_f32(params, "params");
  TORCH_CHECK(best_order.scalar_type() == torch::kInt32 &&
              best_order.is_cuda());
  int T = yT.size(0);
  long long G = yT.size(1);
  TORCH_CHECK(fitted.size(0) == T && fitted.size(1) == G);
  auto stream = at::cuda::getCurrentHIPStream();
  auto fp = [](torch::Tensor& t) -> const float* {
    return t.numel() ? t.data_ptr<float>() : nullptr;
  };
  launch_groupfit_final(
      yT.data_ptr<float>(), xc0.data_ptr<float>(), xc1.data_ptr<float>(),
      xc2.data_ptr<float>(), pj0.data_ptr<float>(), pj1.data_ptr<float>(),
      pj2.data_ptr<float>(), best_order.data_ptr<int>(),
      fitted.data_ptr<float>(), params.data_ptr<float>(),
      status.data_ptr<uint8_t>(),
      fp(b0), fp(b1), fp(b2), fp(w0), fp(w1), fp(w2),
      T, G, (int)xc0.size(1), stream.stream());
}

// Wc [n][G], wm [G] from the time-major panel (differenced d times,
// centered with the per-group train mean).
void diff_center(torch::Tensor yT, torch::Tensor wc, torch::Tensor wm,
                 int64_t S, int64_t d) {
  _check_f32(yT, "yT"); _check_f32(wc, "wc"); _check_f32(wm, "wm");
  auto stream = at::cuda::getCurrentHIPStream();
  launch_diff_center(yT.data_ptr<float>(), wc.data_ptr<float>(),
                     wm.data_ptr<float>(), yT.size(0), yT.size(1),
                     (int)S, (int)d, stream.stream());
}

// beta[KX][G] = P[KX][n] @ wc[n][G] on the f32 matrix cores.
void exog_project_mfma(torch::Tensor P, torch::Tensor wc,
                       torch::Tensor beta) {
  _check_f32(P, "P"); _check_f32(wc, "wc"); _check_f32(beta, "beta");
  TORCH_CHECK(P.size(1) == wc.size(0), "K mismatch");
  TORCH_CHECK(beta.size(0) == P.size(0) && beta.size(1) == wc.size(1));
  TORCH_CHECK(P.size(0) <= 16, "KX must fit one MFMA row tile");
  auto stream = at::cuda::getCurrentHIPStream();
  launch_exog_project_mfma(P.data_ptr<float>(), wc.data_ptr<float>(),
                           beta.data_ptr<float>(), P.size(1), wc.size(1),
                           P.size(0), stream.stream());
}

// ---------------------------------------------------------------- fused BN
// NHWC bf16 fused BatchNorm(+residual)(+ReLU); see fused_bn.hip.

extern "C" int bn_reduce_blocks(long long rows, int C);
extern "C" void launch_bn_fwd_reduce(const void* x, float* partial,
                                     int nblocks, long long rows, int C,
                                     hipStream_t stream);
extern "C" void launch_bn_fwd_finalize(const float* partial, int nblocks,
                                       float* mean, float* invstd,
                                       float* running_mean,
                                       float* running_var, float momentum,
                                       float eps, long long rows, int C,
                                       int update_running, hipStream_t stream);
extern "C" void launch_bn_fwd_apply(const void* x, const void* res, void* y,
                                    const float* mean, const float* invstd,
                                    const float* weight, const float* bias,
                                    long long rows, int C, int relu,
                                    hipStream_t stream);
extern "C" void launch_bn_bwd_reduce(const void* dz, const void* y,
                                     const void* x, const float* mean,
                                     const float* invstd, float* partial,
                                     void* dym, int nblocks, long long rows,
                                     int C, int relu, hipStream_t stream);
extern "C" void launch_bn_bwd_reduce_rm(const void* dz, const void* x,
                                        const float* mean,
                                        const float* invstd,
                                        const float* weight,
                                        const float* bias, float* partial,
                                        void* dym, int nblocks,
                                        long long rows, int C,
                                        hipStream_t stream);
extern "C" void launch_bn_bwd_apply_rm(const void* dz, const void* x,
                                       const float* mean,
                                       const float* invstd,
                                       const float* bias, const float* k,
                                       void* dx, long long rows, int C,
                                       hipStream_t stream);
extern "C" void launch_bn_bwd_apply_dym(const void* dym, const void* x,
                                        const float* mean,
                                        const float* invstd, const float* k,
                                        void* dx, long long rows, int C,
                                        hipStream_t stream);
extern "C" void launch_bn_bwd_finalize(const float* partial, int nblocks,
                                       const float* invstd,
                                       const float* weight, float* dweight,
                                       float* dbias, float* k, long long rows,
                                       int C, int accumulate,
                                       hipStream_t stream);
extern "C" void launch_bn_bwd_apply(const void* dz, const void* y,
                                    const void* x, const float* mean,
                                    const float* invstd, const float* k,
                                    void* dx, void* dres, long long rows,
                                    int C, int relu, hipStream_t stream);

// Activations: bf16, NHWC memory order (a channels_last NCHW tensor viewed
// flat). rows = N*H*W; C must be 8*2^k (ResNet uses 64..512).
static void _check_bn_act(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16,
              name, " must be bf16 on GPU");
}

static long long _bn_rows(const torch::Tensor& x, int64_t C) {
  TORCH_CHECK(C % 8 == 0 && 256 % (C / 8) == 0,
              "C must be 8*2^k and <= 2048, got ", C);
  TORCH_CHECK(x.numel() % C == 0, "numel not divisible by C");
  return (long long)(x.numel() / C);
}

torch::Tensor bn_fwd_reduce(torch::Tensor x, int64_t C) {
  _check_bn_act(x, "x");
  long long rows = _bn_rows(x, C);
  int nb = bn_reduce_blocks(rows, (int)C);
  auto partial = torch::empty(
      {(int64_t)nb * 2 * C},
      torch::TensorOptions().dtype(torch::kFloat32).device(x.device()));
  launch_bn_fwd_reduce(x.data_ptr(), partial.data_ptr<float>(), nb, rows,
                       (int)C, at::cuda::getCurrentHIPStream().stream());
  return partial;
}

void bn_fwd_finalize(torch::Tensor partial, torch::Tensor mean,
                     torch::Tensor invstd, torch::Tensor running_mean,
                     torch::Tensor running_var, double momentum, double eps,
                     int64_t rows, int64_t C, bool update_running) {
  _check_f32(partial, "partial"); _check_f32(mean, "mean");
  _check_f32(invstd, "invstd");
  _check_f32(running_mean, "running_mean");
  _check_f32(running_var, "running_var");
  TORCH_CHECK(partial.numel() % (2 * C) == 0, "partial must be [nb*2C]");
  int nb = (int)(partial.numel() / (2 * C));
  launch_bn_fwd_finalize(partial.data_ptr<float>(), nb,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         running_mean.data_ptr<float>(),
                         running_var.data_ptr<float>(), (float)momentum,
                         (float)eps, rows, (int)C, update_running ? 1 : 0,
                         at::cuda::getCurrentHIPStream().stream());
}

void bn_fwd_apply(torch::Tensor x, torch::Tensor res, torch::Tensor y,
                  torch::Tensor mean, torch::Tensor invstd,
                  torch::Tensor weight, torch::Tensor bias, int64_t C,
                  bool relu) {
  _check_bn_act(x, "x"); _check_bn_act(y, "y");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(weight, "weight"); _check_f32(bias, "bias");
  const void* res_p = nullptr;
  if (res.defined() && res.numel()) {
    _check_bn_act(res, "res");
    TORCH_CHECK(res.numel() == x.numel(), "residual shape mismatch");
    res_p = res.data_ptr();
  }
  launch_bn_fwd_apply(x.data_ptr(), res_p, y.data_ptr(),
                      mean.data_ptr<float>(), invstd.data_ptr<float>(),
                      weight.data_ptr<float>(), bias.data_ptr<float>(),
                      _bn_rows(x, C), (int)C, relu ? 1 : 0,
                      at::cuda::getCurrentHIPStream().stream());
}

// dym (optional, pass an empty tensor to skip): receives the relu-masked
// upstream gradient — for residual layers it IS the residual grad and
// feeds bn_bwd_apply_dym.
torch::Tensor bn_bwd_reduce(torch::Tensor dz, torch::Tensor y,
                            torch::Tensor x, torch::Tensor mean,
                            torch::Tensor invstd, torch::Tensor dym,
                            int64_t C, bool relu) {
  _check_bn_act(dz, "dz"); _check_bn_act(y, "y"); _check_bn_act(x, "x");
  void* dym_p = nullptr;
  if (dym.defined() && dym.numel()) {
    _check_bn_act(dym, "dym");
    TORCH_CHECK(dym.numel() == x.numel(), "dym shape mismatch");
    dym_p = dym.data_ptr();
  }
  long long rows = _bn_rows(x, C);
  int nb = bn_reduce_blocks(rows, (int)C);
  auto partial = torch::empty(
      {(int64_t)nb * 2 * C},
      torch::TensorOptions().dtype(torch::kFloat32).device(x.device()));
  launch_bn_bwd_reduce(dz.data_ptr(), y.data_ptr(), x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       partial.data_ptr<float>(), dym_p, nb, rows, (int)C,
                       relu ? 1 : 0,
                       at::cuda::getCurrentHIPStream().stream());
  return partial;
}

// Recompute-mask variants (non-residual RELU backward): mask from
// (w*xhat + b) > 0 — no y read, y not saved for backward.
torch::Tensor bn_bwd_reduce_rm(torch::Tensor dz, torch::Tensor x,
                               torch::Tensor mean, torch::Tensor invstd,
                               torch::Tensor weight, torch::Tensor bias,
                               torch::Tensor dym, int64_t C) {
  _check_bn_act(dz, "dz"); _check_bn_act(x, "x");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(weight, "weight"); _check_f32(bias, "bias");
  void* dym_p = nullptr;
  if (dym.defined() && dym.numel()) {
    _check_bn_act(dym, "dym");
    TORCH_CHECK(dym.numel() == x.numel(), "dym shape mismatch");
    dym_p = dym.data_ptr();
  }
  long long rows = _bn_rows(x, C);
  int nb = bn_reduce_blocks(rows, (int)C);
  auto partial = torch::empty(
      {(int64_t)nb * 2 * C},
      torch::TensorOptions().dtype(torch::kFloat32).device(x.device()));
  launch_bn_bwd_reduce_rm(dz.data_ptr(), x.data_ptr(),
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          weight.data_ptr<float>(), bias.data_ptr<float>(),
                          partial.data_ptr<float>(), dym_p, nb, rows,
                          (int)C, at::cuda::getCurrentHIPStream().stream());
  return partial;
}

void bn_bwd_apply_rm(torch::Tensor dz, torch::Tensor x, torch::Tensor mean,
                     torch::Tensor invstd, torch::Tensor bias,
                     torch::Tensor k, torch::Tensor dx, int64_t C) {
  _check_bn_act(dz, "dz"); _check_bn_act(x, "x"); _check_bn_act(dx, "dx");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(bias, "bias"); _check_f32(k, "k");
  launch_bn_bwd_apply_rm(dz.data_ptr(), x.data_ptr(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         bias.data_ptr<float>(), k.data_ptr<float>(),
                         dx.data_ptr(), _bn_rows(x, C), (int)C,
                         at::cuda::getCurrentHIPStream().stream());
}

void bn_bwd_apply_dym(torch::Tensor dym, torch::Tensor x,
                      torch::Tensor mean, torch::Tensor invstd,
                      torch::Tensor k, torch::Tensor dx, int64_t C) {
  _check_bn_act(dym, "dym"); _check_bn_act(x, "x"); _check_bn_act(dx, "dx");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(k, "k");
  launch_bn_bwd_apply_dym(dym.data_ptr(), x.data_ptr(),
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          k.data_ptr<float>(), dx.data_ptr(),
                          _bn_rows(x, C), (int)C,
                          at::cuda::getCurrentHIPStream().stream());
}

void bn_bwd_finalize(torch::Tensor partial, torch::Tensor invstd,
                     torch::Tensor weight, torch::Tensor dweight,
                     torch::Tensor dbias, torch::Tensor k, int64_t rows,
                     int64_t C, bool accumulate) {
  _check_f32(partial, "partial"); _check_f32(invstd, "invstd");
  _check_f32(weight, "weight"); _check_f32(dweight, "dweight");
  _check_f32(dbias, "dbias"); _check_f32(k, "k");
  TORCH_CHECK(k.numel() == 3 * C, "k must be [3C]");
  TORCH_CHECK(partial.numel() % (2 * C) == 0, "partial must be [nb*2C]");
  int nb = (int)(partial.numel() / (2 * C));
  launch_bn_bwd_finalize(partial.data_ptr<float>(), nb,
                         invstd.data_ptr<float>(),
                         weight.data_ptr<float>(), dweight.data_ptr<float>(),
                         dbias.data_ptr<float>(), k.data_ptr<float>(), rows,
                         (int)C, accumulate ? 1 : 0,
                         at::cuda::getCurrentHIPStream().stream());
}

void bn_bwd_apply(torch::Tensor dz, torch::Tensor y, torch::Tensor x,
                  torch::Tensor mean, torch::Tensor invstd, torch::Tensor k,
                  torch::Tensor dx, torch::Tensor dres, int64_t C,
                  bool relu) {
  _check_bn_act(dz, "dz"); _check_bn_act(y, "y"); _check_bn_act(x, "x");
  _check_bn_act(dx, "dx");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(k, "k");
  void* dres_p = nullptr;
  if (dres.defined() && dres.numel()) {
    _check_bn_act(dres, "dres");
    dres_p = dres.data_ptr();
  }
  launch_bn_bwd_apply(dz.data_ptr(), y.data_ptr(), x.data_ptr(),
                      mean.data_ptr<float>(), invstd.data_ptr<float>(),
                      k.data_ptr<float>(), dx.data_ptr(), dres_p,
                      _bn_rows(x, C), (int)C, relu ? 1 : 0,
                      at::cuda::getCurrentHIPStream().stream());
}

// ---------------------------------------------------------------- arma gen
extern "C" void launch_arma_gen(const float* ar, const float* ma,
                                const float* eps, float* out, int na,
                                int nb, int burn, int T, long long G,
                                hipStream_t stream);

// Batched ARMA sample paths: ar [G][na] (leading 1), ma [G][nb],
// eps [burn+T][G] time-major noise, out [T][G].
void arma_generate(torch::Tensor ar, torch::Tensor ma, torch::Tensor eps,
                   torch::Tensor out, int64_t burn) {
  _check_f32(ar, "ar"); _check_f32(ma, "ma"); _check_f32(eps, "eps");
  _check_f32(out, "out");
  long long G = ar.size(0);
  int na = (int)ar.size(1), nb = (int)ma.size(1);
  TORCH_CHECK(na >= 1 && na <= 5 && nb >= 1 && nb <= 5,
              "orders up to 4 (leading-1 length 5)");
  TORCH_CHECK(ma.size(0) == G && out.size(1) == G && eps.size(1) == G,
              "G mismatch");
  int T = (int)out.size(0);
  TORCH_CHECK(eps.size(0) == burn + T, "eps must be [burn+T][G]");
  launch_arma_gen(ar.data_ptr<float>(), ma.data_ptr<float>(),
                  eps.data_ptr<float>(), out.data_ptr<float>(), na, nb,
                  (int)burn, T, G,
                  at::cuda::getCurrentHIPStream().stream());
}

// ----------------------------------------------------------------- maxpool
extern "C" void launch_maxpool_fwd(const void* x, void* y,
                                   unsigned char* code, int N, int H, int W,
                                   int Ho, int Wo, int C,
                                   hipStream_t stream);
extern "C" void launch_bn_maxpool_fwd(const void* x, const float* mean,
                                      const float* invstd,
                                      const float* weight,
                                      const float* bias, void* y,
                                      void* code, int N, int H, int W,
                                      int Ho, int Wo, int C,
                                      hipStream_t stream);
extern "C" void launch_maxpool_bwd(const void* dy,
                                   const unsigned char* code, void* dx,
                                   int N, int H, int W, int Ho, int Wo,
                                   int C, hipStream_t stream);

// x: bf16 channels_last [N,C,H,W]; y: [N,C,Ho,Wo]; code: u8 with y's
// NHWC memory order. 3x3/stride-2/pad-1 only (the ResNet stem pool).
void maxpool3x3s2_fwd(torch::Tensor x, torch::Tensor y,
                      torch::Tensor code) {
  _check_bn_act(x, "x"); _check_bn_act(y, "y");
  TORCH_CHECK(code.scalar_type() == torch::kUInt8 && code.is_cuda());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int Ho = y.size(2), Wo = y.size(3);
  TORCH_CHECK(C % 8 == 0 && 256 % (C / 8) == 0, "bad C");
  TORCH_CHECK(Ho == (H + 1) / 2 && Wo == (W + 1) / 2, "3x3 s2 p1 shape");
  TORCH_CHECK(code.numel() == y.numel());
  launch_maxpool_fwd(x.data_ptr(), y.data_ptr(), code.data_ptr<uint8_t>(),
                     N, H, W, Ho, Wo, C,
                     at::cuda::getCurrentHIPStream().stream());
}

// Fused stem forward: relu(bn(x)) applied inline while pooling — the
// normalized activation map is never materialized (see maxpool.hip).
void bn_maxpool3x3s2_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor invstd, torch::Tensor weight,
                         torch::Tensor bias, torch::Tensor y,
                         torch::Tensor code) {
  _check_bn_act(x, "x"); _check_bn_act(y, "y");
  _check_f32(mean, "mean"); _check_f32(invstd, "invstd");
  _check_f32(weight, "weight"); _check_f32(bias, "bias");
  TORCH_CHECK(code.scalar_type() == torch::kUInt8 && code.is_cuda());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int Ho = y.size(2), Wo = y.size(3);
  TORCH_CHECK(C % 8 == 0 && 256 % (C / 8) == 0, "bad C");
  TORCH_CHECK(Ho == (H + 1) / 2 && Wo == (W + 1) / 2, "3x3 s2 p1 shape");
  TORCH_CHECK(code.numel() == y.numel());
  launch_bn_maxpool_fwd(x.data_ptr(), mean.data_ptr<float>(),
                        invstd.data_ptr<float>(),
                        weight.data_ptr<float>(), bias.data_ptr<float>(),
                        y.data_ptr(), code.data_ptr<uint8_t>(),
                        N, H, W, Ho, Wo, C,
                        at::cuda::getCurrentHIPStream().stream());
}

void maxpool3x3s2_bwd(torch::Tensor dy, torch::Tensor code,
                      torch::Tensor dx) {
  _check_bn_act(dy, "dy"); _check_bn_act(dx, "dx");
  TORCH_CHECK(code.scalar_type() == torch::kUInt8 && code.is_cuda());
  int N = dx.size(0), C = dx.size(1), H = dx.size(2), W = dx.size(3);
  int Ho = dy.size(2), Wo = dy.size(3);
  TORCH_CHECK(code.numel() == dy.numel());
  launch_maxpool_bwd(dy.data_ptr(), code.data_ptr<uint8_t>(),
                     dx.data_ptr(), N, H, W, Ho, Wo, C,
                     at::cuda::getCurrentHIPStream().stream());
}

// ---------------------------------------------------------------- flat Adam
extern "C" void launch_adam_step(float* p, const float* g, float* m,
                                 float* v, int* step, float lr, float beta1,
                                 float beta2, float eps, float weight_decay,
                                 long long n, hipStream_t stream);

// One fused Adam step over the flat fp32 param/grad/moment buffers.
// step: int32 device scalar (bumped on-device; hipGraph-replayable).
void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor step, double lr, double beta1,
               double beta2, double eps, double weight_decay) {
  _check_f32(p, "p"); _check_f32(g, "g"); _check_f32(m, "m");
  _check_f32(v, "v");
  TORCH_CHECK(step.is_cuda() && step.scalar_type() == torch::kInt32 &&
                  step.numel() == 1,
              "step must be a 1-element int32 device tensor");
  long long n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n,
              "flat buffer length mismatch");
  launch_adam_step(p.data_ptr<float>(), g.data_ptr<float>(),
                   m.data_ptr<float>(), v.data_ptr<float>(),
                   step.data_ptr<int>(), (float)lr, (float)beta1,
                   (float)beta2, (float)eps, (float)weight_decay, n,
                   at::cuda::getCurrentHIPStream().stream());
}

extern "C" void launch_adam_step_mixed(float* master, const void* gb,
                                       const float* gf, float* m, float* v,
                                       void* pb, int* step, float lr,
                                       float beta1, float beta2, float eps,
                                       float weight_decay, long long nb,
                                       long long n, hipStream_t stream);

// bf16-parameter FlatAdam step: master fp32 [n]; gb bf16 grads for the
// first nb elements (pb = bf16 working params, rewritten here); gf fp32
// grads for the rest.
void adam_step_mixed(torch::Tensor master, torch::Tensor gb,
                     torch::Tensor gf, torch::Tensor m, torch::Tensor v,
                     torch::Tensor pb, torch::Tensor step, double lr,
                     double beta1, double beta2, double eps,
                     double weight_decay) {
  _check_f32(master, "master"); _check_f32(m, "m"); _check_f32(v, "v");
  TORCH_CHECK(gb.scalar_type() == torch::kBFloat16 &&
              pb.scalar_type() == torch::kBFloat16, "gb/pb must be bf16");
  TORCH_CHECK(step.is_cuda() && step.scalar_type() == torch::kInt32 &&
              step.numel() == 1, "step must be int32[1] on device");
  long long n = master.numel();
  long long nb = gb.numel();
  TORCH_CHECK(pb.numel() == nb && gf.numel() == n - nb &&
              m.numel() == n && v.numel() == n, "buffer length mismatch");
  const float* gf_p = gf.numel() ? gf.data_ptr<float>() : nullptr;
  launch_adam_step_mixed(master.data_ptr<float>(), gb.data_ptr(), gf_p,
                         m.data_ptr<float>(), v.data_ptr<float>(),
                         pb.data_ptr(), step.data_ptr<int>(), (float)lr,
                         (float)beta1, (float)beta2, (float)eps,
                         (float)weight_decay, nb, n,
                         at::cuda::getCurrentHIPStream().stream());
}

// Stem conv (7x7 s2 p3, 3->64, NHWC bf16): out = conv(x, w).
// x [N,H,W,3] bf16 channels-last storage, w [64,3,7,7] channels_last,
// out [N,HO,WO,64].
void stem_conv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor out,
                   torch::Tensor wp_scratch, int64_t phase_mask = 7) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(out.is_cuda() && out.scalar_type() == torch::kBFloat16);
  int Nb = x.size(0), H = x.size(1), W = x.size(2);
  TORCH_CHECK(x.size(3) == 3 && out.size(3) == 64, "stem is 3->64");
  int HO = out.size(1), WO = out.size(2);
  TORCH_CHECK(WO <= 128, "stem kernel handles output rows up to 128 px");
  TORCH_CHECK(w.numel() == 64 * 7 * 7 * 3);
  TORCH_CHECK(wp_scratch.scalar_type() == torch::kBFloat16 &&
              wp_scratch.numel() == 64 * 160);
  launch_stem_conv_fwd(x.data_ptr(), w.data_ptr(), out.data_ptr(),
                       wp_scratch.data_ptr(),
                       Nb, H, W, HO, WO,
                       at::cuda::getCurrentHIPStream().stream(),
                       (int)phase_mask);
}

// Weight grad: dw (bf16, weight layout) from x and dy; dw_f32 is the
// [64][160] fp32 accumulation scratch.
void stem_conv_wrw(torch::Tensor x, torch::Tensor dy, torch::Tensor dw_f32,
                   torch::Tensor dw, int64_t phase_mask = 7) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dw_f32.scalar_type() == torch::kFloat32 &&
              dw_f32.numel() == 64 * 160);
  TORCH_CHECK(dw.scalar_type() == torch::kBFloat16 &&
              dw.numel() == 64 * 7 * 7 * 3);
  int Nb = x.size(0), H = x.size(1), W = x.size(2);
  int HO = dy.size(1), WO = dy.size(2);
  launch_stem_conv_wrw(x.data_ptr(), dy.data_ptr(),
                       dw_f32.data_ptr<float>(), dw.data_ptr(),
                       Nb, H, W, HO, WO,
                       at::cuda::getCurrentHIPStream().stream(),
                       (int)phase_mask);
}

// 3x3/s1/p1 CxC weight grad: part fp32 [chunks][C][9][C] scratch
// (size via conv3x3_wrw_nchunks), dw bf16 out in channels_last weight
// layout (may be the flat grad view).
void conv3x3_wrw(torch::Tensor x, torch::Tensor dy, torch::Tensor part,
                 torch::Tensor dw, int64_t phase_mask = 7,
                 int64_t chunks = 0) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  TORCH_CHECK(dy.size(1) == H && dy.size(2) == W && dy.size(3) == C,
              "same-shape 3x3 s1 conv only");
  TORCH_CHECK(C % 64 == 0 && C <= 2048, "C must be a multiple of 64");
  TORCH_CHECK(W >= 3 && W <= 64, "row length outside kernel bounds");
  const int nch = conv3x3_wrw_chunks(Nb, H, C, (int)chunks);
  TORCH_CHECK(part.scalar_type() == torch::kFloat32 &&
              part.numel() >= (long long)nch * C * 9 * C,
              "partial buffer too small for ", nch, " chunks");
  TORCH_CHECK(dw.scalar_type() == torch::kBFloat16 &&
              dw.numel() == (long long)C * 9 * C);
  launch_conv3x3_wrw(x.data_ptr(), dy.data_ptr(), part.data_ptr<float>(),
                     dw.data_ptr(), Nb, H, W, C, nch,
                     at::cuda::getCurrentHIPStream().stream(),
                     (int)phase_mask);
}

int64_t conv3x3_wrw_nchunks(int64_t Nb, int64_t H, int64_t C,
                            int64_t chunks) {
  return conv3x3_wrw_chunks((int)Nb, (int)H, (int)C, (int)chunks);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv3x3_wrw", &conv3x3_wrw,
        "MFMA 3x3/s1 CxC weight grad (fp32 accum, bf16 cast to view)",
        pybind11::arg("x"), pybind11::arg("dy"), pybind11::arg("part"),
        pybind11::arg("dw"), pybind11::arg("phase_mask") = 7,
        pybind11::arg("chunks") = 0);
  m.def("conv3x3_wrw_nchunks", &conv3x3_wrw_nchunks,
        "partial-buffer chunk count the wrw launcher will use");
  m.def("stem_conv_fwd", &stem_conv_fwd,
        "MFMA stem conv fwd (7x7 s2, 3->64, NHWC bf16)",
        pybind11::arg("x"), pybind11::arg("w"), pybind11::arg("out"),
        pybind11::arg("wp_scratch"), pybind11::arg("phase_mask") = 7);
  m.def("stem_conv_wrw", &stem_conv_wrw,
        "MFMA stem conv weight-grad (fp32 accum + bf16 cast)",
        pybind11::arg("x"), pybind11::arg("dy"), pybind11::arg("dw_f32"),
        pybind11::arg("dw"), pybind11::arg("phase_mask") = 7);
  m.def("normalize_u8_to_bf16", &normalize_u8_to_bf16,
        "fused uint8 NHWC -> normalized bf16 (same memory order)");
  m.def("groupfit_eval", &groupfit_eval,
        "batched per-group ARIMAX candidate evaluation (validation MSE)");
  m.def("groupfit_final", &groupfit_final,
        "batched per-group final fit (fitted values + params)");
  m.def("diff_center", &diff_center,
        "difference + per-group centering of the time-major panel");
  m.def("exog_project_mfma", &exog_project_mfma,
        "design-matrix GEMM beta = P @ Wc on f32 MFMA");
  m.def("bn_fwd_reduce", &bn_fwd_reduce);
  m.def("bn_fwd_finalize", &bn_fwd_finalize);
  m.def("bn_fwd_apply", &bn_fwd_apply);
  m.def("bn_bwd_reduce", &bn_bwd_reduce);
  m.def("bn_bwd_reduce_rm", &bn_bwd_reduce_rm);
  m.def("bn_bwd_apply_rm", &bn_bwd_apply_rm);
  m.def("bn_bwd_finalize", &bn_bwd_finalize);
  m.def("bn_bwd_apply", &bn_bwd_apply);
  m.def("bn_bwd_apply_dym", &bn_bwd_apply_dym);
  m.def("adam_step", &adam_step,
        "fused flat Adam step (one kernel over p/g/m/v)");
  m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd);
  m.def("bn_maxpool3x3s2_fwd", &bn_maxpool3x3s2_fwd);
  m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd);
  m.def("adam_step_mixed", &adam_step_mixed,
        "fused flat Adam with bf16 params/grads + fp32 master");
  m.def("arma_generate", &arma_generate,
        "batched ARMA(p,q) sample paths, one group per lane");
}
