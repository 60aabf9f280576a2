// Batched per-group ARIMAX fit for gfx950 (MI355X) — the N2 kernel.
//
// Replaces the reference's one-statsmodels-fit-per-Spark-task hot loop
// (group_apply/02_Fine_Grained_Demand_Forecasting.py:441-450,472-481)
// with a fixed-schedule estimator (see forecast/batched.py — the numpy
// oracle these kernels are tested against):
//   stage 1: OLS regression on shared exog design (host-precomputed
//            pseudo-inverse / MFMA design GEMM -> per-group beta)
//   stage 2: Yule-Walker + Levinson-Durbin long-AR -> innovations
//   stage 3: lag-matrix normal equations (<=8x8), ridge + Cholesky,
//            stationarity shrinkage; one refinement pass (stage 4)
//   stage 5: H-step validation forecast, integrated to y-units, MSE
//
// MI355X-first decomposition: ONE GROUP PER LANE, 64 groups per wave,
// one wave per workgroup. Global series are TIME-MAJOR [T][G] so every
// per-timestep access is a fully coalesced 64-lane load.
//
// STREAMING (round 2): every stage has <= 8-lag lookback, so the u/eps
// series never materialize. Each pass re-derives u_t on the fly from
// the y stream (1 coalesced load + KX fma) and carries 8-deep per-lane
// REGISTER rings (constant-indexed under full unrolls — runtime-indexed
// register arrays would spill, guide §5.4). The round-1 version staged
// u/eps in LDS slabs (2 x S x 64 lanes fp32 ≈ 60 KB/block) and ran ONE
// wave per CU; dropping the slabs leaves only the per-lane state block
// (small matrices/vectors, 120 floats x 64 lanes = 30 KB) and lifts
// occupancy to ~5 waves/CU. The extra y re-reads are noise: ~4 passes
// x n x 4 B x G per candidate ≈ 190 MB at 8 TB/s HBM3E.
//
// Design note on MFMA: the per-group normal equations are at most 8x8
// built from ~120-step series — batching 64 groups per wave makes the
// whole fit latency-bound, so matrix cores only help stage 1 (the
// [KX,n]x[n,G] design GEMM, ops/csrc/mfma_project.hip), which both the
// eval and final kernels take precomputed.

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

// ---------------------------------------------------------------------
// Streaming pass machinery. Each pass walks t = 0..S-1, re-deriving
//   w_t  = diff^d(y)_t            (y0/y1 register ring)
//   u_td = (w_t - wm) - x_td . beta
// and carries ur[j] = u_{td-1-j}, er[j] = e_{td-1-j} register rings
// (zero-filled: pre-sample lags are zero, matching the oracle).
// ---------------------------------------------------------------------

#define RING_SHIFT(r, v)                        \
  _Pragma("unroll") for (int _j = 7; _j > 0; --_j) r[_j] = r[_j - 1]; \
  r[0] = (v)

// One streaming u-producer step. Returns u_td (only valid when t >= d).
__device__ __forceinline__ float u_step(
    const float* __restrict__ yT, const float* __restrict__ xc,
    const float* st, int lane, long long g, long long G, int KX,
    int d, int t, float wm, float* y0, float* y1) {
  const float yv = yT[(long long)t * G + g];
  float w;
  if (d == 0) w = yv;
  else if (d == 1) w = yv - *y0;
  else w = yv - 2.0f * (*y0) + (*y1);
  *y1 = *y0; *y0 = yv;
  if (t < d) return 0.0f;
  const int td = t - d;
  float reg = 0.0f;
  for (int k = 0; k < KX; ++k)
    reg += xc[(long long)td * KX + k] * st[SIDX(ST_BETA + k)];
  return (w - wm) - reg;
}

// Core fit for one (group-lane, order): leaves ST_BETA/ST_PHI/ST_THETA
// state and the end-of-series rings ur/er (u_{n-1-j}, e_{n-1-j}).
// *ok_out false on non-finite state.
__device__ float fit_lane(
    const float* __restrict__ yT, const float* __restrict__ xc,
    const float* __restrict__ PJ, float* st,
    int lane, long long g, long long G, int S, int KX, int p, int d, int q,
    bool* ok_out, float* ur, float* er,
    const float* __restrict__ beta_pre,  // [KX][G] from the MFMA
    const float* __restrict__ wm_pre) {  // [G]     projection, or null
  const int n = S - d;
  float y0, y1;

  // ---- stage 0: mean of the differenced series (pass only if needed)
  float wm;
  if (wm_pre) {
    wm = wm_pre[g];
  } else {
    float wsum = 0.0f;
    y0 = y1 = 0.0f;
    for (int t = 0; t < S; ++t) {
      const float yv = yT[(long long)t * G + g];
      float w;
      if (d == 0) w = yv;
      else if (d == 1) w = yv - y0;
      else w = yv - 2.0f * y0 + y1;
      y1 = y0; y0 = yv;
      if (t >= d) wsum += w;
    }
    wm = wsum / (float)n;
  }

  // ---- stage 1: beta = P @ (w - wm) (pass only if not precomputed)
#pragma unroll
  for (int k = 0; k < MAXK; ++k) BETA(k) = 0.0f;
  if (beta_pre) {
    for (int k = 0; k < KX; ++k) BETA(k) = beta_pre[(long long)k * G + g];
  } else {
    float bacc[MAXK];
#pragma unroll
    for (int k = 0; k < MAXK; ++k) bacc[k] = 0.0f;
    y0 = y1 = 0.0f;
    for (int t = 0; t < S; ++t) {
      const float yv = yT[(long long)t * G + g];
      float w;
      if (d == 0) w = yv;
      else if (d == 1) w = yv - y0;
      else w = yv - 2.0f * y0 + y1;
      y1 = y0; y0 = yv;
      if (t < d) continue;
      const int td = t - d;
      const float wcv = w - wm;
#pragma unroll
      for (int k = 0; k < MAXK; ++k)
        if (k < KX) bacc[k] += PJ[(long long)k * n + td] * wcv;
    }
#pragma unroll
    for (int k = 0; k < MAXK; ++k)
      if (k < KX) BETA(k) = bacc[k];
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) { PHI(i) = 0.0f; THETA(i) = 0.0f; }
#pragma unroll
  for (int j = 0; j < 8; ++j) { ur[j] = 0.0f; er[j] = 0.0f; }

  const int K = p + q;
  const bool lane_q = q > 0;
  const bool lane_K = K > 0;
  int M = 0;

  // ---- stage 2: autocovariance pass + Levinson (q > 0 lanes)
  if (__any(lane_q)) {
    M = max(p, q) + 3;
    if (M > MAXM) M = MAXM;
    if (M > n / 4) M = max(1, n / 4);
    float r[MAXM + 1];
#pragma unroll
    for (int k = 0; k <= MAXM; ++k) r[k] = 0.0f;
    y0 = y1 = 0.0f;
    for (int t = 0; t < S; ++t) {
      const float u = u_step(yT, xc, st, lane, g, G, KX, d, t, wm, &y0, &y1);
      if (t < d) continue;
      r[0] += u * u;
#pragma unroll
      for (int k = 1; k <= MAXM; ++k)
        if (k <= M) r[k] += u * ur[k - 1];
      RING_SHIFT(ur, u);
    }
    if (lane_q) {
#pragma unroll
      for (int k = 0; k <= MAXM; ++k)
        if (k <= M) st[SIDX(ST_R + k)] = r[k] / (float)n;
      levinson(st, lane, M);
    }
  }

  // ---- stages 3+4: two normal-equation passes + final innovation pass.
  //   NE pass 0: e = long-AR innovations (q>0) / e = u (q==0)
  //   NE pass 1: e = recursive innovations under pass-0 coefs (q>0 only)
  //   final pass: e under final coefs -> end-of-series rings (q>0 only;
  //               q==0 lanes' rings are complete after NE pass 0 and the
  //               theta terms they feed are zero)
  const bool any_q = __any(lane_q);
  if (__any(lane_K)) {
    const int m = max(max(p, q), 1);
    // q==0 lanes solve only in pass 0, and a pure-AR wave needs neither
    // the refinement pass nor the final innovation pass (its theta
    // terms are zero, so the e-ring is never consumed)
    const int npass = any_q ? 2 : 1;
    for (int pass = 0; pass < npass; ++pass) {
      const bool solve_here = lane_K && (pass == 0 || lane_q);
      // coefs for this pass's innovation recursion
      float alr[MAXM], phr[4], thr[4];
#pragma unroll
      for (int i = 0; i < MAXM; ++i)
        alr[i] = (lane_q && pass == 0 && i < M) ? st[SIDX(ST_AL + i)] : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) { phr[i] = PHI(i); thr[i] = THETA(i); }
      if (solve_here) {
        for (int i = 0; i < 64; ++i) st[SIDX(ST_A + i)] = 0.0f;
        for (int i = 0; i < 8; ++i) st[SIDX(ST_C + i)] = 0.0f;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) { ur[j] = 0.0f; er[j] = 0.0f; }
      y0 = y1 = 0.0f;
      for (int t = 0; t < S; ++t) {
        const float u = u_step(yT, xc, st, lane, g, G, KX, d, t, wm,
                               &y0, &y1);
        if (t < d) continue;
        const int td = t - d;
        // innovation under this pass's recursion
        float e = u;
        if (pass == 0) {
#pragma unroll
          for (int i = 0; i < MAXM; ++i) e -= alr[i] * ur[i];
        } else {
#pragma unroll
          for (int i = 0; i < 4; ++i)
            e -= phr[i] * ur[i] + thr[i] * er[i];
        }
        if (solve_here && td >= m) {
#pragma unroll
          for (int i = 0; i < 4; ++i)
            if (i < p) st[SIDX(ST_Z + i)] = ur[i];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (j < q) st[SIDX(ST_Z + p + j)] = er[j];
          for (int i = 0; i < K; ++i) {
            const float zi = st[SIDX(ST_Z + i)];
            for (int j = i; j < K; ++j)
              st[SIDX(ST_A + i * 8 + j)] += zi * st[SIDX(ST_Z + j)];
            st[SIDX(ST_C + i)] += zi * u;
          }
        }
        RING_SHIFT(ur, u);
        RING_SHIFT(er, e);
      }
      if (solve_here) {
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
#pragma unroll
        for (int i = 0; i < 4; ++i) { PHI(i) = 0.0f; THETA(i) = 0.0f; }
        for (int i = 0; i < p; ++i) PHI(i) = st[SIDX(ST_C + i)];
        for (int j = 0; j < q; ++j) THETA(j) = st[SIDX(ST_C + p + j)];
#pragma unroll
        for (int i = 0; i < 4; ++i) sp += fabsf(PHI(i));
#pragma unroll
        for (int j = 0; j < 4; ++j) sq += fabsf(THETA(j));
        if (sp > SHRINK) {
          const float s = SHRINK / sp;
#pragma unroll
          for (int i = 0; i < 4; ++i) PHI(i) *= s;
        }
        if (sq > SHRINK) {
          const float s = SHRINK / sq;
#pragma unroll
          for (int j = 0; j < 4; ++j) THETA(j) *= s;
        }
      }
    }
    // final innovation pass under the final coefs (leaves the rings at
    // the end of the series for stage 5 / the fitted-value writer)
    if (any_q) {
      float phr[4], thr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) { phr[i] = PHI(i); thr[i] = THETA(i); }
#pragma unroll
      for (int j = 0; j < 8; ++j) { ur[j] = 0.0f; er[j] = 0.0f; }
      y0 = y1 = 0.0f;
      for (int t = 0; t < S; ++t) {
        const float u = u_step(yT, xc, st, lane, g, G, KX, d, t, wm,
                               &y0, &y1);
        if (t < d) continue;
        float e = u;
#pragma unroll
        for (int i = 0; i < 4; ++i)
          e -= phr[i] * ur[i] + thr[i] * er[i];
        RING_SHIFT(ur, u);
        RING_SHIFT(er, e);
      }
    }
  }

  float chk = ur[0] + er[0] + wm;
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
  __shared__ float st[NSTATE * WAVE];
  const int lane = threadIdx.x;
  const long long g = (long long)blockIdx.x * WAVE + lane;
  const bool active = g < G;
  const long long gg = active ? g : (G - 1);
  const int H = T - S;
  float ur[8], er[8];

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
    const float wm = fit_lane(yT, xc, PJ, st, lane, gg, G, S, KX,
                              p, d, q, &ok, ur, er, bpre, wpre);

    // ---- stage 5: validation forecast + MSE (rings hold u/e at n-1-j)
    float u0 = ur[0], u1 = ur[1], u2 = ur[2], u3 = ur[3];
    float e0 = er[0], e1 = er[1], e2 = er[2], e3 = er[3];
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
  }
}

// Final fit: per-group chosen order (may differ per lane — the passes
// are wave-voted, per-lane work predicated). Writes one-step-ahead
// fitted values [T][G] in y-units, per-group coefficient vector
// (wm, beta[KX], phi[4], theta[4]) and status. beta/wm precomputed by
// the MFMA projection over the FULL-series design when given.
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
    const float* __restrict__ b0, const float* __restrict__ b1,
    const float* __restrict__ b2,        // [KX][G] per d, or null
    const float* __restrict__ w0, const float* __restrict__ w1,
    const float* __restrict__ w2,        // [G] per d, or null
    int T, long long G, int KX) {
  __shared__ float st[NSTATE * WAVE];
  const int lane = threadIdx.x;
  const long long g = (long long)blockIdx.x * WAVE + lane;
  const bool active = g < G;
  const long long gg = active ? g : (G - 1);
  const int p = best_order[gg * 3 + 0];
  const int d = best_order[gg * 3 + 1];
  const int q = best_order[gg * 3 + 2];
  const float* xc = (d == 0) ? xc0 : (d == 1) ? xc1 : xc2;
  const float* PJ = (d == 0) ? pj0 : (d == 1) ? pj1 : pj2;
  const float* bpre = (d == 0) ? b0 : (d == 1) ? b1 : b2;
  const float* wpre = (d == 0) ? w0 : (d == 1) ? w1 : w2;
  const int S = T;  // final fit uses the whole series
  float ur[8], er[8];

  bool ok;
  const float wm = fit_lane(yT, xc, PJ, st, lane, gg, G, S, KX,
                            p, d, q, &ok, ur, er, bpre, wpre);

  // Fitted-value pass: stream t, predict from the lag rings, THEN push
  // the current u/e (one-step-ahead: prediction at t uses lags only).
  {
    const float ph0 = PHI(0), ph1 = PHI(1), ph2 = PHI(2), ph3 = PHI(3);
    const float th0 = THETA(0), th1 = THETA(1), th2 = THETA(2),
                th3 = THETA(3);
#pragma unroll
    for (int j = 0; j < 8; ++j) { ur[j] = 0.0f; er[j] = 0.0f; }
    float y0 = 0.f, y1 = 0.f;
    for (int t = 0; t < T; ++t) {
      const float yv = yT[(long long)t * G + gg];
      const float yl1 = y0, yl2 = y1;
      float w;
      if (d == 0) w = yv;
      else if (d == 1) w = yv - y0;
      else w = yv - 2.0f * y0 + y1;
      y1 = y0; y0 = yv;
      float out;
      if (t < d) {
        out = yv;
      } else {
        const int td = t - d;
        float regx = 0.0f;
        for (int k = 0; k < KX; ++k)
          regx += xc[(long long)td * KX + k] * BETA(k);
        const float acc = ph0 * ur[0] + ph1 * ur[1] + ph2 * ur[2] +
                          ph3 * ur[3] + th0 * er[0] + th1 * er[1] +
                          th2 * er[2] + th3 * er[3];
        const float what = wm + regx + acc;
        if (d == 0) out = what;
        else if (d == 1) out = yl1 + what;
        else out = 2.0f * yl1 - yl2 + what;
        const float u = (w - wm) - regx;
        const float e = u - acc;
        RING_SHIFT(ur, u);
        RING_SHIFT(er, e);
      }
      if (active) fitted[(long long)t * G + g] = out;
    }
  }

  if (active) {
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

extern "C" void launch_groupfit_eval(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2, const int* orders,
    float* mse, unsigned char* statusv,
    const float* b0, const float* b1, const float* b2,
    const float* w0, const float* w1, const float* w2,
    int T, long long G, int S, int C,
    int KX, hipStream_t stream) {
  const int blocks = (int)((G + WAVE - 1) / WAVE);
  hipLaunchKernelGGL(groupfit_eval_kernel, dim3(blocks), dim3(WAVE), 0,
                     stream, yT, xc0, xc1, xc2, pj0, pj1, pj2, orders, mse,
                     statusv, b0, b1, b2, w0, w1, w2, T, G, S, C, KX);
}

extern "C" void launch_groupfit_final(
    const float* yT, const float* xc0, const float* xc1, const float* xc2,
    const float* pj0, const float* pj1, const float* pj2,
    const int* best_order, float* fitted, float* params,
    unsigned char* statusv,
    const float* b0, const float* b1, const float* b2,
    const float* w0, const float* w1, const float* w2,
    int T, long long G, int KX, hipStream_t stream) {
  const int blocks = (int)((G + WAVE - 1) / WAVE);
  hipLaunchKernelGGL(groupfit_final_kernel, dim3(blocks), dim3(WAVE), 0,
                     stream, yT, xc0, xc1, xc2, pj0, pj1, pj2, best_order,
                     fitted, params, statusv, b0, b1, b2, w0, w1, w2,
                     T, G, KX);
}
