// Stage-1 of the batched group fit as an MFMA GEMM (gfx950).
//
// The exog regression coefficients of EVERY group are one matrix product
//   beta[KX, G] = P[KX, n] @ Wc[n, G]
// where P is the host-precomputed train pseudo-inverse (shared by all
// groups, forecast/batched.py make_exog_designs) and Wc is the
// differenced+centered demand panel. This is the "design-matrix GEMM" of
// the north star, computed on the f32 matrix cores
// (v_mfma_f32_16x16x4_f32 — exact f32 at the f32 vector rate, guide §3).
//
// Two kernels:
//   diff_center_kernel : lane-per-group pass producing Wc [n][G] and the
//                        per-group mean wm[G] from the time-major panel.
//   exog_project_mfma  : one wave per 16-column group tile; A-fragment
//                        rows are P (rows >= KX are zero-padded),
//                        B-fragment is the Wc tile, K accumulated 4 per
//                        MFMA; C written for rows < KX.

#include <hip/hip_runtime.h>

#define WAVE 64

typedef float f32x4 __attribute__((ext_vector_type(4)));

extern "C" __global__ __launch_bounds__(256) void diff_center_kernel(
    const float* __restrict__ yT,   // [T][G]
    float* __restrict__ wc,         // [n][G], n = S - d
    float* __restrict__ wm,         // [G]
    int T, long long G, int S, int d) {
  const long long g = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= G) return;
  const int n = S - d;
  float y0 = 0.f, y1 = 0.f, sum = 0.f;
  for (int t = 0; t < S; ++t) {
    const float yv = yT[(long long)t * G + g];
    float w;
    if (d == 0) w = yv;
    else if (d == 1) w = yv - y0;
    else w = yv - 2.0f * y0 + y1;
    y1 = y0; y0 = yv;
    if (t >= d) {
      wc[(long long)(t - d) * G + g] = w;
      sum += w;
    }
  }
  const float mean = sum / (float)n;
  wm[g] = mean;
  for (int t = 0; t < n; ++t)
    wc[(long long)t * G + g] -= mean;
}

extern "C" __global__ __launch_bounds__(WAVE) void exog_project_mfma_kernel(
    const float* __restrict__ P,    // [KX][n]
    const float* __restrict__ wc,   // [n][G] (centered)
    float* __restrict__ beta,       // [KX][G]
    int n, long long G, int KX) {
  const int lane = threadIdx.x;
  const long long j0 = (long long)blockIdx.x * 16;   // column tile
  const int arow = lane & 15;        // A row  (P row, padded to 16)
  const int ak = lane >> 4;          // A k within the 4-wide K step
  const int brow = lane >> 4;        // B k-row within the step
  const long long bcol = j0 + (lane & 15);
  const bool colok = bcol < G;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  int k = 0;
  for (; k + 4 <= n; k += 4) {
    const float a =
        (arow < KX) ? P[(long long)arow * n + (k + ak)] : 0.0f;
    const float b =
        colok ? wc[(long long)(k + brow) * G + bcol] : 0.0f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  // K tail (n % 4): feed zeros in the out-of-range k lanes
  if (k < n) {
    const int kr = n - k;
    const float a = (arow < KX && ak < kr)
                        ? P[(long long)arow * n + (k + ak)] : 0.0f;
    const float b = (colok && brow < kr)
                        ? wc[(long long)(k + brow) * G + bcol] : 0.0f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  // C layout (16x16x4f32): col = lane&15, row = (lane>>4)*4 + reg
  if (!colok) return;
  const int rbase = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = rbase + r;
    if (row < KX) beta[(long long)row * G + bcol] = acc[r];
  }
}

extern "C" void launch_diff_center(const float* yT, float* wc, float* wm,
                                   int T, long long G, int S, int d,
                                   hipStream_t stream) {
  const int block = 256;
  const long long grid = (G + block - 1) / block;
  hipLaunchKernelGGL(diff_center_kernel, dim3((unsigned)grid), dim3(block),
                     0, stream, yT, wc, wm, T, G, S, d);
}

extern "C" void launch_exog_project_mfma(const float* P, const float* wc,
                                         float* beta, int n, long long G,
                                         int KX, hipStream_t stream) {
  const long long grid = (G + 15) / 16;
  hipLaunchKernelGGL(exog_project_mfma_kernel, dim3((unsigned)grid),
                     dim3(WAVE), 0, stream, P, wc, beta, n, G, KX);
}
