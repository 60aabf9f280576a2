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
    unsigned char* statusv, int T, long long G, int KX, hipStream_t stream);

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
                    torch::Tensor status) {
  _check_f32(yT, "yT"); _check_f32(fitted, "fitted");
  _check_f32(params, "params");
  TORCH_CHECK(best_order.scalar_type() == torch::kInt32 &&
              best_order.is_cuda());
  int T = yT.size(0);
  long long G = yT.size(1);
  TORCH_CHECK(fitted.size(0) == T && fitted.size(1) == G);
  auto stream = at::cuda::getCurrentHIPStream();
  launch_groupfit_final(
      yT.data_ptr<float>(), xc0.data_ptr<float>(), xc1.data_ptr<float>(),
      xc2.data_ptr<float>(), pj0.data_ptr<float>(), pj1.data_ptr<float>(),
      pj2.data_ptr<float>(), best_order.data_ptr<int>(),
      fitted.data_ptr<float>(), params.data_ptr<float>(),
      status.data_ptr<uint8_t>(), T, G, (int)xc0.size(1), stream.stream());
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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
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
}
