// Batched per-group ARIMAX fit for gfx950 (MI355X) — the N2 kernel.
//
// Replaces the reference's one-statsmodels-fit-per-Spark-task hot loop
// (group_apply/02_Fine_Grained_Demand_Forecasting.py:441-450,472-481)
// with a fixed-schedule estimator (see forecast/batched.py — the numpy
// oracle these kernels are tested against):
//   stage 1: OLS regression on shared exog design (host-precomputed
//            pseudo-inverse -> per-group matvec)
//   stage 2: Yule-Walker + Levinson-Durbin long-AR -> innovations
//   stage 3: lag-matrix normal equations (<=8x8), ridge + Cholesky,
//            stationarity shrinkage; one refinement pass (stage 4)
//   stage 5: H-step validation forecast, integrated to y-units, MSE
//
// MI355X-first decomposition: ONE GROUP PER LANE, 64 groups per wave,
// one wave per workgroup. Global series are TIME-MAJOR [T][G] so every
// per-timestep access is a fully coalesced 64-lane load. Per-group
// series (u, eps) live in LDS slabs [n][64] (bank = (t*64+lane)%64 —
// all 64 lanes hit distinct banks every access); per-group small
// matrices/vectors (A, r, coefs) live in a per-lane LDS state block
// because runtime-indexed register arrays would spill to scratch
// (guide §5.4 rule 20). Loop trip counts are uniform across the wave in
// the eval kernel (one candidate order at a time), so the hot loops have
// no divergence; the final-fit kernel predicates on per-lane (p,d,q).
//
// Design note on MFMA: the per-group normal equations are at most 8x8
// built from ~120-step series — batching 64 groups per wave makes the
// whole fit bandwidth/latency-bound (µs-scale per candidate for 100k
// groups), so matrix cores have nothing to accelerate in this op.

#include <hip/hip_runtime.h>
#include <math.h>

#define WAVE 64
#define MAXK 8          // p + q <= 8
#define MAXM 7          // long-AR order cap
#define RIDGE 1e-6f
#define SHRINK 0.98f

// per-lane LDS state layout (floats), stride WAVE
#define ST_A 0          // 8x8 normal-equation matrix (row-major)
#define ST_C 64         // rhs / solution (8)
#define ST_R 72         // autocovariances (8)
#define ST_AL 80        // long-AR coefs (8)
#define ST_TMP 88       // levinson scratch (8)
#define ST_BETA 96      // exog coefs (8)
#define ST_PHI 104      // AR coefs (4)
#define ST_THETA 108    // MA coefs (4)
#define ST_Z 112        // lag-regressor vector (8)
#define NSTATE 120

#define SIDX(k) ((k) * WAVE + lane)
#define PHI(i) st[SIDX(ST_PHI + (i))]
#define THETA(i) st[SIDX(ST_THETA + (i))]
#define BETA(k) st[SIDX(ST_BETA + (k))]

// Cholesky solve (per-lane LDS): A x = b, A KxK SPD at ST_A, b at ST_C on
// entry, x at ST_C on exit. Runtime K <= 8.
__device__ void chol_solve(float* st, int lane, int K) {
  for (int k = 0; k < K; ++k) {
    float akk = st[SIDX(ST_A + k * 8 + k)];
    for (int j = 0; j < k; ++j) {
      float lkj = st[SIDX(ST_A + k * 8 + j)];
      akk -= lkj * lkj;
    }
    akk = sqrtf(fmaxf(akk, 1e-20f));
    st[SIDX(ST_A + k * 8 + k)] = akk;
    float inv = 1.0f / akk;
    for (int i = k + 1; i < K; ++i) {
      float v = st[SIDX(ST_A + i * 8 + k)];
      for (int j = 0; j < k; ++j)
        v -= st[SIDX(ST_A + i * 8 + j)] * st[SIDX(ST_A + k * 8 + j)];
      st[SIDX(ST_A + i * 8 + k)] = v * inv;
    }
  }
  for (int i = 0; i < K; ++i) {
    float v = st[SIDX(ST_C + i)];
    for (int j = 0; j < i; ++j)
      v -= st[SIDX(ST_A + i * 8 + j)] * st[SIDX(ST_C + j)];
    st[SIDX(ST_C + i)] = v / st[SIDX(ST_A + i * 8 + i)];
  }
  for (int i = K - 1; i >= 0; --i) {
    float v = st[SIDX(ST_C + i)];
    for (int j = i + 1; j < K; ++j)
      v -= st[SIDX(ST_A + j * 8 + i)] * st[SIDX(ST_C + j)];
    st[SIDX(ST_C + i)] = v / st[SIDX(ST_A + i * 8 + i)];
  }
}

// Levinson-Durbin: AR(M) from autocovariances at ST_R -> coefs at ST_AL.
__device__ void levinson(float* st, int lane, int M) {
  for (int i = 0; i < MAXM + 1; ++i) st[SIDX(ST_AL + i)] = 0.0f;
  float e = st[SIDX(ST_R + 0)];
  e = (e > 0.0f) ? e : 1.0f;
  for (int k = 1; k <= M; ++k) {
    float acc = st[SIDX(ST_R + k)];
    for (int j = 0; j < k - 1; ++j)
      acc -= st[SIDX(ST_AL + j)] * st[SIDX(ST_R + (k - 1 - j))];
    float lam = acc / e;
    for (int j = 0; j < k - 1; ++j) st[SIDX(ST_TMP + j)] = st[SIDX(ST_AL + j)];
    for (int j = 0; j < k - 1; ++j)
      st[SIDX(ST_AL + j)] =
          st[SIDX(ST_TMP + j)] - lam * st[SIDX(ST_TMP + (k - 2 - j))];
    st[SIDX(ST_AL + (k - 1))] = lam;
    e *= (1.0f - lam * lam);
    if (e <= 0.0f) e = 1e-12f;
  }
}

// Core fit for one (group-lane, order): fills U (regression residuals)
// and E (innovations) slabs and the ST_BETA/ST_PHI/ST_THETA state.
// Returns the mean of the differenced series; *ok_out false on
// non-finite state.
__device__ __forceinline__ float fit_lane(
    const float* __restrict__ yT, const float* __restrict__ xc,
    const float* __restrict__ PJ, float* U, float* E, float* st,
    int lane, long long g, long long G, int S, int KX, int p, int d, int q,
    bool* ok_out,
    const float* __restrict__ beta_pre,  // [KX][G] from the MFMA
    const float* __restrict__ wm_pre) {  // [G]     projection, or null
  const int n = S - d;
  // ---- stage 0: difference + mean (stream y, write w into U)
  float y0 = 0.f, y1 = 0.f;
  float wsum = 0.0f;
  for (int t = 0; t < S; ++t) {
    float yv = yT[(long long)t * G + g];
    float w;
    if (d == 0) w = yv;
    else if (d == 1) w = yv - y0;
    else w = yv - 2.0f * y0 + y1;
    y1 = y0; y0 = yv;
    if (t >= d) {
      U[(t - d) * WAVE + lane] = w;
      wsum += w;
    }
  }
  const float wm = wm_pre ? wm_pre[g] : (wsum / (float)n);
  // ---- stage 1: beta = P @ (w - wm); u = wc - Xc beta. beta comes from
  // the MFMA design-matrix GEMM (ops/csrc/mfma_project.hip) when
  // precomputed, else from the per-lane matvec.
#pragma unroll
  for (int k = 0; k < MAXK; ++k) BETA(k) = 0.0f;
  if (beta_pre) {
    for (int k = 0; k < KX; ++k) BETA(k) = beta_pre[(long long)k * G + g];
  } else {
    for (int k = 0; k < KX; ++k) {
      float acc = 0.0f;
      const float* Pk = PJ + (long long)k * n;
      for (int t = 0; t < n; ++t)
        acc += Pk[t] * (U[t * WAVE + lane] - wm);
      BETA(k) = acc;
    }
  }
  for (int t = 0; t < n; ++t) {
    float reg = 0.0f;
    for (int k = 0; k < KX; ++k) reg += xc[(long long)t * KX + k] * BETA(k);
    U[t * WAVE + lane] = (U[t * WAVE + lane] - wm) - reg;
  }
  // ---- stage 2: long-AR innovations
  if (q > 0) {
    int M = max(p, q) + 3;
    if (M > MAXM) M = MAXM;
    if (M > n / 4) M = max(1, n / 4);
    for (int k = 0; k <= M; ++k) {
      float acc = 0.0f;
      for (int t = k; t < n; ++t)
        acc += U[t * WAVE + lane] * U[(t - k) * WAVE + lane];
      st[SIDX(ST_R + k)] = acc / (float)n;
    }
    levinson(st, lane, M);
    for (int t = 0; t < n; ++t) {
      float acc = U[t * WAVE + lane];
      for (int i = 1; i <= M; ++i)
        if (t - i >= 0)
          acc -= st[SIDX(ST_AL + (i - 1))] * U[(t - i) * WAVE + lane];
      E[t * WAVE + lane] = acc;
    }
  } else {
    for (int t = 0; t < n; ++t) E[t * WAVE + lane] = U[t * WAVE + lane];
  }
  // ---- stages 3+4: lag OLS, refine once
#pragma unroll
  for (int i = 0; i < 4; ++i) { PHI(i) = 0.0f; THETA(i) = 0.0f; }
  const int K = p + q;
  if (K > 0) {
    const int m = max(max(p, q), 1);
    for (int pass = 0; pass < 2; ++pass) {
      for (int i = 0; i < 64; ++i) st[SIDX(ST_A + i)] = 0.0f;
      for (int i = 0; i < 8; ++i) st[SIDX(ST_C + i)] = 0.0f;
      for (int t = m; t < n; ++t) {
        for (int i = 0; i < p; ++i)
          st[SIDX(ST_Z + i)] = U[(t - 1 - i) * WAVE + lane];
        for (int j = 0; j < q; ++j)
          st[SIDX(ST_Z + p + j)] = E[(t - 1 - j) * WAVE + lane];
        const float ut = U[t * WAVE + lane];
        for (int i = 0; i < K; ++i) {
          const float zi = st[SIDX(ST_Z + i)];
          for (int j = i; j < K; ++j)
            st[SIDX(ST_A + i * 8 + j)] += zi * st[SIDX(ST_Z + j)];
          st[SIDX(ST_C + i)] += zi * ut;
        }
      }
      float tr = 0.0f;
      for (int i = 0; i < K; ++i) tr += st[SIDX(ST_A + i * 8 + i)];
      const float lam = RIDGE * fmaxf(1.0f, tr / (float)K);
      for (int i = 0; i < K; ++i) {
        st[SIDX(ST_A + i * 8 + i)] += lam;
        for (int j = i + 1; j < K; ++j)
          st[SIDX(ST_A + j * 8 + i)] = st[SIDX(ST_A + i * 8 + j)];
      }
      chol_solve(st, lane, K);
      float sp = 0.0f, sq = 0.0f;
      for (int i = 0; i < 4; ++i) PHI(i) = 0.0f;
      for (int j = 0; j < 4; ++j) THETA(j) = 0.0f;
      for (int i = 0; i < p; ++i) PHI(i) = st[SIDX(ST_C + i)];
      for (int j = 0; j < q; ++j) THETA(j) = st[SIDX(ST_C + p + j)];
#pragma unroll
      for (int i = 0; i < 4; ++i) sp += fabsf(PHI(i));
#pragma unroll
      for (int j = 0; j < 4; ++j) sq += fabsf(THETA(j));
      if (sp > SHRINK) {
        const float s = SHRINK / sp;
        for (int i = 0; i < 4; ++i) PHI(i) *= s;
      }
      if (sq > SHRINK) {
        const float s = SHRINK / sq;
        for (int j = 0; j < 4; ++j) THETA(j) *= s;
      }
      // recompute innovations under (phi, theta)
      const float ph0 = PHI(0), ph1 = PHI(1), ph2 = PHI(2), ph3 = PHI(3);
      const float th0 = THETA(0), th1 = THETA(1), th2 = THETA(2),
                  th3 = THETA(3);
      float e0 = 0.f, e1 = 0.f, e2 = 0.f, e3 = 0.f;
      for (int t = 0; t < n; ++t) {
        float acc = U[t * WAVE + lane];
        if (t - 1 >= 0) acc -= ph0 * U[(t - 1) * WAVE + lane];
        if (t - 2 >= 0) acc -= ph1 * U[(t - 2) * WAVE + lane];
        if (t - 3 >= 0) acc -= ph2 * U[(t - 3) * WAVE + lane];
        if (t - 4 >= 0) acc -= ph3 * U[(t - 4) * WAVE + lane];
        acc -= th0 * e0 + th1 * e1 + th2 * e2 + th3 * e3;
        e3 = e2; e2 = e1; e1 = e0; e0 = acc;
        E[t * WAVE + lane] = acc;
      }
      if (q == 0) break;
    }
  }
  float chk = E[(n - 1) * WAVE + lane] + wm;
#pragma unroll
  for (int i = 0; i < 4; ++i) chk += PHI(i) + THETA(i);
  for (int k = 0; k < KX; ++k) chk += BETA(k);
  *ok_out = isfinite(chk);
  return wm;
}

extern "C" __global__ __launch_bounds__(WAVE) void groupfit_eval_kernel(
    const float* __restrict__ yT,
    const float* __restrict__ xc0, const float* __restrict__ xc1,
    const float* __restrict__ xc2,
    const float* __restrict__ pj0, const float* __restrict__ pj1,
    const float* __restrict__ pj2,
    const int* __restrict__ orders,       // [C][3]
    float* __restrict__ mse,              // [C][G]
    unsigned char* __restrict__ statusv,  // [C][G]
    const float* __restrict__ b0, const float* __restrict__ b1,
    const float* __restrict__ b2,         // [KX][G] per d, or null
    const float* __restrict__ w0, const float* __restrict__ w1,
    const float* __restrict__ w2,         // [G] per d, or null
    int T, long long G, int S, int C, int KX) {
  extern __shared__ float lds[];
  float* U = lds;
  float* E = lds + (long long)S * WAVE;
  float* st = lds + 2 * (long long)S * WAVE;
  const int lane = threadIdx.x;
  const long long g = (long long)blockIdx.x * WAVE + lane;
  const bool active = g < G;
  const long long gg = active ? g : (G - 1);
  const int H = T - S;

  for (int c = 0; c < C; ++c) {
    const int p = orders[c * 3 + 0];
    const int d = orders[c * 3 + 1];
    const int q = orders[c * 3 + 2];
    const float* xc = (d == 0) ? xc0 : (d == 1) ? xc1 : xc2;
    const float* PJ = (d == 0) ? pj0 : (d == 1) ? pj1 : pj2;
    const int n = S - d;

    const float* bpre = (d == 0) ? b0 : (d == 1) ? b1 : b2;
    const float* wpre = (d == 0) ? w0 : (d == 1) ? w1 : w2;
    bool ok;
    const float wm = fit_lane(yT, xc, PJ, U, E, st, lane, gg, G, S, KX,
                              p, d, q, &ok, bpre, wpre);

    // ---- stage 5: validation forecast + MSE
    float u0 = U[(n - 1) * WAVE + lane];
    float u1 = (n - 2 >= 0) ? U[(n - 2) * WAVE + lane] : 0.f;
    float u2 = (n - 3 >= 0) ? U[(n - 3) * WAVE + lane] : 0.f;
    float u3 = (n - 4 >= 0) ? U[(n - 4) * WAVE + lane] : 0.f;
    float e0 = E[(n - 1) * WAVE + lane];
    float e1 = (n - 2 >= 0) ? E[(n - 2) * WAVE + lane] : 0.f;
    float e2 = (n - 3 >= 0) ? E[(n - 3) * WAVE + lane] : 0.f;
    float e3 = (n - 4 >= 0) ? E[(n - 4) * WAVE + lane] : 0.f;
    const float ph0 = PHI(0), ph1 = PHI(1), ph2 = PHI(2), ph3 = PHI(3);
    const float th0 = THETA(0), th1 = THETA(1), th2 = THETA(2),
                th3 = THETA(3);
    float yl1 = yT[(long long)(S - 1) * G + gg];
    float yl2 = (S - 2 >= 0) ? yT[(long long)(S - 2) * G + gg] : 0.f;
    float sse = 0.0f;
    for (int h = 0; h < H; ++h) {
      float reg = 0.0f;
      for (int k = 0; k < KX; ++k)
        reg += xc[(long long)(n + h) * KX + k] * BETA(k);
      const float acc = ph0 * u0 + ph1 * u1 + ph2 * u2 + ph3 * u3 +
                        th0 * e0 + th1 * e1 + th2 * e2 + th3 * e3;
      const float w_pred = wm + reg + acc;
      u3 = u2; u2 = u1; u1 = u0; u0 = acc;
      e3 = e2; e2 = e1; e1 = e0; e0 = 0.0f;
      float y_pred;
      if (d == 0) y_pred = w_pred;
      else if (d == 1) { y_pred = yl1 + w_pred; yl2 = yl1; yl1 = y_pred; }
      else { y_pred = w_pred + 2.0f * yl1 - yl2; yl2 = yl1; yl1 = y_pred; }
      const float yv = yT[(long long)(S + h) * G + gg];
      const float err = yv - y_pred;
      sse += err * err;
    }
    const float m = sse / (float)H;
    if (active) {
      const bool fin = ok && isfinite(m);
      mse[(long long)c * G + g] = fin ? m : INFINITY;
      statusv[(long long)c * G + g] = fin ? 1 : 0;
    }
    __syncthreads();  // LDS slabs reused by the next candidate
  }
}

// Final fit: per-group chosen order (may differ per lane). Writes
// one-step-ahead fitted values [T][G] in y-units, per-group coefficient
// vector (wm, beta[KX], phi[4], theta[4]) and status.
extern "C" __global__ __launch_bounds__(WAVE) void groupfit_final_kernel(
    const float* __restrict__ yT,
    const float* __restrict__ xc0, const float* __restrict__ xc1,
    const float* __restrict__ xc2,
    const float* __restrict__ pj0, const float* __restrict__ pj1,
    const float* __restrict__ pj2,
    const int* __restrict__ best_order,  // [G][3]
    float* __restrict__ fitted,          // [T][G]
    float* __restrict__ params,          // [G][1+KX+8]
    unsigned char* __restrict__ statusv, // [G]
    int T, long long G, int KX) {
  extern __shared__ float lds[];
  float* U = lds;
  float* E = lds + (long long)T * WAVE;
  float* st = lds + 2 * (long long)T * WAVE;
  const int lane = threadIdx.x;
  const long long g = (long long)blockIdx.x * WAVE + lane;
  const bool active = g < G;
  const long long gg = active ? g : (G - 1);
  const int p = best_order[gg * 3 + 0];
  const int d = best_order[gg * 3 + 1];
  const int q = best_order[gg * 3 + 2];
  const float* xc = (d == 0) ? xc0 : (d == 1) ? xc1 : xc2;
  const float* PJ = (d == 0) ? pj0 : (d == 1) ? pj1 : pj2;
  const int S = T;  // final fit uses the whole series
  const int n = S - d;

  bool ok;
  const float wm = fit_lane(yT, xc, PJ, U, E, st, lane, gg, G, S, KX,
                            p, d, q, &ok, nullptr, nullptr);

  if (active) {
    const float ph0 = PHI(0), ph1 = PHI(1), ph2 = PHI(2), ph3 = PHI(3);
    const float th0 = THETA(0), th1 = THETA(1), th2 = THETA(2),
                th3 = THETA(3);
    float yprev1 = 0.f, yprev2 = 0.f;
    for (int t = 0; t < T; ++t) {
      const float yv = yT[(long long)t * G + g];
      float out;
      if (t < d) {
        out = yv;
      } else {
        const int td = t - d;
        float reg = wm;
        for (int k = 0; k < KX; ++k)
          reg += xc[(long long)td * KX + k] * BETA(k);
        float acc = 0.0f;
        if (td - 1 >= 0) acc += ph0 * U[(td - 1) * WAVE + lane] +
                                th0 * E[(td - 1) * WAVE + lane];
        if (td - 2 >= 0) acc += ph1 * U[(td - 2) * WAVE + lane] +
                                th1 * E[(td - 2) * WAVE + lane];
        if (td - 3 >= 0) acc += ph2 * U[(td - 3) * WAVE + lane] +
                                th2 * E[(td - 3) * WAVE + lane];
        if (td - 4 >= 0) acc += ph3 * U[(td - 4) * WAVE + lane] +
                                th3 * E[(td - 4) * WAVE + lane];
        const float what = reg + acc;
        if (d == 0) out = what;
        else if (d == 1) out = yprev1 + what;
        else out = 2.0f * yprev1 - yprev2 + what;
      }
      fitted[(long long)t * G + g] = out;
      yprev2 = yprev1; yprev1 = yv;
    }
    float* pp = params + g * (1 + KX + 8);
    pp[0] = wm;
    for (int k = 0; k < KX; ++k) pp[1 + k] = BETA(k);
    for (int i = 0; i < 4; ++i) {
      pp[1 + KX + i] = PHI(i);
      pp[1 + KX + 4 + i] = THETA(i);
    }
    statusv[g] = ok ? 1 : 0;
  }
}

static void _set_lds_limit(const void* func, size_t shmem) {
  static size_t done_eval = 0, done_final = 0;
  (void)done_eval; (void)done_final;
  hipFuncSetAttribute(func, hipFuncAttributeMaxDynamicSharedMemorySize,
                      (int)shmem);
}

extern "C" void launch_groupfit_eval(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2, const int* orders,
    float* mse, unsigned char* statusv,
    const float* b0, const float* b1, const float* b2,
    const float* w0, const float* w1, const float* w2,
    int T, long long G, int S, int C,
    int KX, hipStream_t stream) {
  const int blocks = (int)((G + WAVE - 1) / WAVE);
  const size_t shmem =
      (size_t)(2 * S * WAVE + NSTATE * WAVE) * sizeof(float);
  _set_lds_limit(reinterpret_cast<const void*>(groupfit_eval_kernel), shmem);
  hipLaunchKernelGGL(groupfit_eval_kernel, dim3(blocks), dim3(WAVE), shmem,
                     stream, yT, xc0, xc1, xc2, pj0, pj1, pj2, orders, mse,
                     statusv, b0, b1, b2, w0, w1, w2, T, G, S, C, KX);
}

extern "C" void launch_groupfit_final(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2,
    const int* best_order, float* fitted, float* params,
    unsigned char* statusv, int T, long long G, int KX, hipStream_t stream) {
  const int blocks = (int)((G + WAVE - 1) / WAVE);
  const size_t shmem =
      (size_t)(2 * T * WAVE + NSTATE * WAVE) * sizeof(float);
  _set_lds_limit(reinterpret_cast<const void*>(groupfit_final_kernel), shmem);
  hipLaunchKernelGGL(groupfit_final_kernel, dim3(blocks), dim3(WAVE), shmem,
                     stream, yT, xc0, xc1, xc2, pj0, pj1, pj2, best_order,
                     fitted, params, statusv, T, G, KX);
}
