// Batched ARMA(p,q) sample-path generator for gfx950.
//
// The reference synthesizes each SKU's demand series with statsmodels
// arma_generate_sample (a C lfilter) inside a pandas UDF
// (group_apply/_resources/01-data-generator.py:242-254, burn-in 3000).
// At the 100k-group benchmark scale that is ~316M recurrence steps of
// host work; here one kernel runs one group per lane (same layout as
// the groupfit kernels: time-major [T][G] panels, coalesced across
// lanes at every time step).
//
// Convention matches lfilter(ma, ar, eps) with ar[0] == 1:
//   x[t] = sum_j ma[j]*eps[t-j]  -  sum_{i>=1} ar[i]*x[t-i]
// Orders up to 4 (the reference grid ends at (4,2,4)).

#include <hip/hip_runtime.h>

#define AMAXO 4  // max AR/MA order

__global__ __launch_bounds__(256) void arma_gen_kernel(
    const float* __restrict__ ar,    // [G][na]  (ar[g][0] == 1)
    const float* __restrict__ ma,    // [G][nb]  (ma[g][0] == 1 usually)
    const float* __restrict__ eps,   // [Ttot][G] time-major noise
    float* __restrict__ out,         // [T][G] time-major
    int na, int nb, int burn, int T, long long G) {
  const long long g = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= G) return;

  float arc[AMAXO], mac[AMAXO + 1];
  const int p = na - 1, q = nb - 1;
  for (int i = 0; i < AMAXO; ++i) arc[i] = (i < p) ? ar[g * na + 1 + i] : 0.f;
  mac[0] = ma[g * nb];
  for (int j = 0; j < AMAXO; ++j)
    mac[j + 1] = (j < q) ? ma[g * nb + 1 + j] : 0.f;

  float xl[AMAXO] = {0.f, 0.f, 0.f, 0.f};
  float el[AMAXO] = {0.f, 0.f, 0.f, 0.f};
  const int Ttot = burn + T;
  for (int t = 0; t < Ttot; ++t) {
    const float e = eps[(long long)t * G + g];
    float acc = mac[0] * e;
#pragma unroll
    for (int j = 0; j < AMAXO; ++j) acc += mac[j + 1] * el[j];
#pragma unroll
    for (int i = 0; i < AMAXO; ++i) acc -= arc[i] * xl[i];
#pragma unroll
    for (int k = AMAXO - 1; k > 0; --k) {
      xl[k] = xl[k - 1];
      el[k] = el[k - 1];
    }
    xl[0] = acc;
    el[0] = e;
    if (t >= burn) out[(long long)(t - burn) * G + g] = acc;
  }
}

extern "C" void launch_arma_gen(const float* ar, const float* ma,
                                const float* eps, float* out, int na,
                                int nb, int burn, int T, long long G,
                                hipStream_t stream) {
  const int block = 256;
  long long grid = (G + block - 1) / block;
  hipLaunchKernelGGL(arma_gen_kernel, dim3((unsigned)grid), dim3(block), 0,
                     stream, ar, ma, eps, out, na, nb, burn, T, G);
}
