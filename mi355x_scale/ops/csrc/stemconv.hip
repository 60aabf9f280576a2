// Hand-written MFMA stem convolution for gfx950: 7x7 stride-2 pad-3,
// Cin=3, Cout=64, NHWC bf16 (fwd + weight-grad; the stem input never
// needs a data-grad).
//
// Why: the ResNet stem is the one conv MIOpen leaves on a generic igemm
// tile (C=3 starves the GEMM K: K = 7*7*3 = 147) — measured 298 us fwd
// + ~294 us wrw per step at bs 212, ~7% of the whole flagship step
// (profiles/r01_bench_resnet18_fused_kernstats.md). A shape-special
// kernel treats it as the GEMM it is:
//
//   fwd:  out[px][co]  = sum_k  patch[px][k]   * w[co][k]     (M=N*HO*WO, N=64, K=147)
//   wrw:  dw[co][k]    = sum_px dy[px][co]     * patch[px][k] (M=64, N=147, K=N*HO*WO)
//
// on v_mfma_f32_16x16x32_bf16 (frag maps per cdna_hip_programming.md §3:
// A row = lane&15, B col = lane&15, k-slice = 8*(lane>>4)+j; C/D
// col = lane&15, row = 4*(lane>>4)+reg).
//
// Memory discipline (ablated, v1/v2 lessons):
//   v1: per-element GLOBAL im2col gather — one load+wait per element,
//       waves >80% stalled (SQ_WAIT_ANY), 2.4 ms.
//   v2: contiguous band stage + LDS->LDS im2col expansion — the MFMA
//       phase alone measured 108 us, but stage (637 us) and expand
//       (541 us) still dominated: serialized load->ds_write round trips.
//   v3 (this): each block owns ONE output row (n, ho); the 7-row input
//       band is staged with short4 vector copies into a PIXEL-PADDED
//       layout (edge zeros materialized, plus an all-zero 8th row that
//       absorbs the K-padding taps), so there is no expansion phase and
//       no bounds checks: the MFMA loop gathers its A-fragments straight
//       from the band at koff(k) + px*6.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define KH 7
#define KW 7
#define CI 3
#define CO 64
#define KTAP (KH * KW * CI)     // 147
#define KPAD 160                // 5 x K32 MFMA steps
#define KLDS 168                // padded weight-tile row stride
#define PXPAD 128               // padded output-row length (WO <= 128)
// band row: 4 zero pixels of lead (data starts 8B-aligned at el 12),
// 2*PXPAD+KW pixel taps, zero tail; rounded to short4 granularity
#define XTROW ((4 + 2 * PXPAD + KW + 4) * CI + 3)  // 816 els
#define XTOFF 12                // data starts at pixel 4 -> element 12
#define STRIDE 2
#define PADDING 3

typedef __bf16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// Stage the 7-row band. ``prime``: zero the whole 8-row tile first (the
// data region is fully overwritten every call with the same W, so
// non-prime calls only re-zero row slots whose ih fell outside the
// image — the lead/tail pads and row 7 stay zero from the prime).
// Copies are flattened over (row, chunk) and register-batched 5 deep so
// the global loads pipeline instead of paying one latency each (the v2
// lesson). Callers barrier before the MFMA phase.
__device__ __forceinline__ void stage_band(
    const bf16* __restrict__ x, bf16* xt, int n, int ho, int H, int W,
    int tid, int nthreads, bool prime) {
  const int rowlen = W * CI;
  if (prime) {
    for (int i = tid; i < 8 * XTROW; i += nthreads) xt[i] = (bf16)0.0f;
  } else {
    for (int r = 0; r < KH; ++r) {
      const int ih = ho * STRIDE - PADDING + r;
      if (ih < 0 || ih >= H)
        for (int i = tid; i < rowlen; i += nthreads)
          xt[r * XTROW + XTOFF + i] = (bf16)0.0f;
    }
  }
  __syncthreads();
  if ((rowlen & 3) == 0) {
    const int nch = rowlen >> 2;
    const int total = KH * nch;
    const bf16x4 z = {(bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f};
    for (int base = 0; base < total; base += nthreads * 5) {
      bf16x4 v[5];
      int rr[5], cc[5];
#pragma unroll
      for (int u = 0; u < 5; ++u) {
        const int i = base + tid + u * nthreads;
        v[u] = z;
        rr[u] = -1;
        if (i < total) {
          const int r = i / nch, c = i - r * nch;
          const int ih = ho * STRIDE - PADDING + r;
          rr[u] = r; cc[u] = c;
          if (ih >= 0 && ih < H)
            v[u] = reinterpret_cast<const bf16x4*>(
                x + ((long long)n * H + ih) * rowlen)[c];
        }
      }
#pragma unroll
      for (int u = 0; u < 5; ++u)
        if (rr[u] >= 0)
          reinterpret_cast<bf16x4*>(
              xt + rr[u] * XTROW + XTOFF)[cc[u]] = v[u];
    }
  } else {  // odd row length (test shapes): scalar fallback
    for (int r = 0; r < KH; ++r) {
      const int ih = ho * STRIDE - PADDING + r;
      if (ih < 0 || ih >= H) continue;
      const bf16* src = x + ((long long)n * H + ih) * rowlen;
      bf16* dst = xt + r * XTROW + XTOFF;
      for (int i = tid; i < rowlen; i += nthreads) dst[i] = src[i];
    }
  }
}

// Ring-band staging for the wrw row loop: consecutive output rows (same
// n) share 5 of their 7 input rows, so after a prime only the TWO new
// rows are staged per step. Row slot = ih & 7 (bitwise works for the
// negative top-edge ihs); invalid rows are zeroed in their slot. The
// K-padding taps (kh = 7) would alias a live slot, so the wrw fragment
// builder zero-selects k >= KTAP instead of reading a dedicated row.
__device__ __forceinline__ void stage_band_ring(
    const bf16* __restrict__ x, bf16* xt, int n, int ho, int H, int W,
    int tid, int nthreads, bool prime) {
  const int rowlen = W * CI;
  const int ih0 = ho * STRIDE - PADDING;
  const int rfirst = prime ? 0 : KH - STRIDE;  // prime: all 7; else last 2
  if ((rowlen & 3) == 0) {
    const int nch = rowlen >> 2;
    const int nrows = KH - rfirst;
    const int total = nrows * nch;
    const bf16x4 z = {(bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f};
    for (int base = 0; base < total; base += nthreads * 5) {
      bf16x4 v[5];
      int ss[5], cc[5];
#pragma unroll
      for (int u = 0; u < 5; ++u) {
        const int i = base + tid + u * nthreads;
        v[u] = z;
        ss[u] = -1;
        if (i < total) {
          const int r = rfirst + i / nch, c = i - (i / nch) * nch;
          const int ih = ih0 + r;
          ss[u] = (ih & 7); cc[u] = c;
          if (ih >= 0 && ih < H)
            v[u] = reinterpret_cast<const bf16x4*>(
                x + ((long long)n * H + ih) * rowlen)[c];
        }
      }
#pragma unroll
      for (int u = 0; u < 5; ++u)
        if (ss[u] >= 0)
          reinterpret_cast<bf16x4*>(
              xt + ss[u] * XTROW + XTOFF)[cc[u]] = v[u];
    }
  } else {
    for (int r = rfirst; r < KH; ++r) {
      const int ih = ih0 + r;
      bf16* dst = xt + (ih & 7) * XTROW + XTOFF;
      if (ih >= 0 && ih < H) {
        const bf16* src = x + ((long long)n * H + ih) * rowlen;
        for (int i = tid; i < rowlen; i += nthreads) dst[i] = src[i];
      } else {
        for (int i = tid; i < rowlen; i += nthreads) dst[i] = (bf16)0.0f;
      }
    }
  }
}

// koff(k): element offset of tap k for pixel 0 in the padded band.
// tap iw = px*STRIDE - PADDING + kw -> element
// XTOFF + (px*STRIDE - PADDING + kw)*CI + ci
//   = (XTOFF - PADDING*CI) + px*(STRIDE*CI) + (kw*CI + ci)
// pad taps (k >= 147) resolve to kh = 7, the all-zero row.
__device__ __forceinline__ int koff_of(int k) {
  const int kh = k / (KW * CI);
  const int r21 = k - kh * (KW * CI);
  return kh * XTROW + (XTOFF - PADDING * CI) + r21;
}

// Pad the weights once per forward: [CO][KTAP] -> [CO][KPAD] (zeros in
// the MFMA K-padding taps). 20 KB — L1-resident for every fwd block, so
// the per-block LDS weight-tile refill (40 serialized loads/lane x
// 23744 blocks) disappears entirely.
extern "C" __global__ __launch_bounds__(256) void stem_pad_weights_kernel(
    const bf16* __restrict__ w, bf16* __restrict__ wp) {
  const int i = blockIdx.x * 256 + threadIdx.x;
  if (i < CO * KPAD) {
    const int co = i / KPAD, k = i - co * KPAD;
    wp[i] = (k < KTAP) ? w[co * KTAP + k] : (bf16)0.0f;
  }
}

// ---------------------------------------------------------------- forward
// block = 256 threads (4 waves) = one output row; wave w computes px
// subtiles {2w, 2w+1}. LDS = band only (13 KB) -> many blocks/CU; the
// B-fragments read the padded weight buffer straight from global (the
// whole 20 KB tile lives in L1).
extern "C" __global__ __launch_bounds__(256) void stem_conv_fwd_kernel(
    const bf16* __restrict__ x,   // [N][H][W][CI]
    const bf16* __restrict__ wp,  // [CO][KPAD] padded (stem_pad_weights)
    bf16* __restrict__ out,       // [N][HO][WO][CO]
    int Nb, int H, int W, int HO, int WO, int phase_mask) {
  __shared__ bf16 xt[8 * XTROW];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int n = blockIdx.x / HO;
  const int ho = blockIdx.x - n * HO;

  if (phase_mask & 1) {
    stage_band(x, xt, n, ho, H, W, tid, 256, true);
  }
  __syncthreads();
  if (!(phase_mask & 4)) return;

  f32x4 acc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int c = 0; c < 4; ++c) acc[m][c] = (f32x4)0.0f;

  const int row16 = lane & 15;
  const int kgrp = lane >> 4;
  const int pxb0 = (wave * 32 + row16) * (STRIDE * CI);
  const int pxb1 = (wave * 32 + 16 + row16) * (STRIDE * CI);
#pragma unroll
  for (int kk = 0; kk < KPAD / 32; ++kk) {
    bf16x8 bfrag[4];
#pragma unroll
    for (int c = 0; c < 4; ++c)
      bfrag[c] = *reinterpret_cast<const bf16x8*>(
          wp + (c * 16 + row16) * KPAD + kk * 32 + kgrp * 8);
    bf16x8 f0, f1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int ko = koff_of(kk * 32 + kgrp * 8 + j);
      f0[j] = xt[ko + pxb0];
      f1[j] = xt[ko + pxb1];
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      acc[0][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          f0, bfrag[c], acc[0][c], 0, 0, 0);
      acc[1][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          f1, bfrag[c], acc[1][c], 0, 0, 0);
    }
  }

  // store: D col = lane&15 (co), row = 4*(lane>>4)+r (px within row)
  const long long rowbase = ((long long)n * HO + ho) * WO;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
    const int pxr = wave * 32 + m * 16 + kgrp * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int px = pxr + r;
      if (px >= WO) continue;
#pragma unroll
      for (int c = 0; c < 4; ++c)
        out[(rowbase + px) * CO + c * 16 + row16] = (bf16)acc[m][c][r];
    }
  }
}

// ---------------------------------------------------------------- wrw
// block = 256 threads (4 waves) loops over ``rows_per_block`` output
// rows; per row: band stage + dy-row stage, then each wave reduces ALL
// four 32-px slices into its own (co, k) QUADRANT (2 co-subtiles x 5
// k-subtiles = 40 f32/lane; px-split waves would hold 160 accumulators
// and VGPR-cap occupancy at 1 wave/SIMD). A-frag = dy^T (row = co,
// k-slice = 8 px), B-frag gathered straight from the padded band.
// Partials atomic-add into the L2-resident [CO][KPAD] fp32 tile once
// per block (spread over 2.5k addresses — NOT the one-address-per-block
// serialization the round-1 notes warn about).
extern "C" __global__ __launch_bounds__(256) void stem_conv_wrw_kernel(
    const bf16* __restrict__ x,    // [N][H][W][CI]
    const bf16* __restrict__ dy,   // [N][HO][WO][CO]
    float* __restrict__ dw,        // [CO][KPAD] fp32 (pre-zeroed)
    int Nb, int H, int W, int HO, int WO, int rows_per_block,
    int phase_mask) {
  __shared__ bf16 xt[8 * XTROW];
  __shared__ bf16 Dy[2 * PXPAD * CO];  // double-buffered dy row slices
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int row16 = lane & 15;
  const int kgrp = lane >> 4;
  const int nrows = Nb * HO;
  const int mw = (wave >> 1) * 2;   // this wave's co-subtiles: mw, mw+1
  const int cw = (wave & 1) * 5;    // this wave's k-subtiles: cw .. cw+4
  const int row_limit = min((blockIdx.x + 1) * rows_per_block, nrows);
  const int nch = (WO * CO) >> 3;
  const bf16x8 zv = {(bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f,
                     (bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f};

  f32x4 acc[2][5];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int c = 0; c < 5; ++c) acc[m][c] = (f32x4)0.0f;

  // dy is SOFTWARE-PIPELINED: row r+1's slice is loaded into registers
  // while row r's MFMA phase runs, and written to the other Dy buffer
  // after it — the per-row global-load stall (the largest wrw phase in
  // the ablation) hides under compute. Prologue: stage row 0 directly.
  int cur = 0;
  {
    const int row0 = blockIdx.x * rows_per_block;
    if (row0 < nrows) {
      const bf16x8* src = reinterpret_cast<const bf16x8*>(
          dy + (long long)row0 * WO * CO);
      bf16x8* d8 = reinterpret_cast<bf16x8*>(Dy);
      bf16x8 v[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int i = tid + u * 256;
        v[u] = (i < nch) ? src[i] : zv;
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) d8[tid + u * 256] = v[u];
    }
  }

  for (int rr = 0; rr < rows_per_block; ++rr) {
    const int row = blockIdx.x * rows_per_block + rr;
    if (row >= nrows) break;
    const int n = row / HO;
    const int ho = row - n * HO;
    const bool prime = (rr == 0) || (ho == 0);
    __syncthreads();  // previous iteration's readers + Dy writes done
    if (prime)  // lead/tail pads must be zero before the first data rows
      for (int i = tid; i < 8 * XTROW; i += 256) xt[i] = (bf16)0.0f;
    __syncthreads();
    if (phase_mask & 1)
      stage_band_ring(x, xt, n, ho, H, W, tid, 256, prime);
    // prefetch NEXT row's dy slice (latency overlaps band stage,
    // barrier and the MFMA phase below)
    bf16x8 v[4];
    const bool have_next = (phase_mask & 2) && row + 1 < row_limit;
    if (have_next) {
      const bf16x8* src = reinterpret_cast<const bf16x8*>(
          dy + (long long)(row + 1) * WO * CO);
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int i = tid + u * 256;
        v[u] = (i < nch) ? src[i] : zv;
      }
    }
    __syncthreads();
    if (phase_mask & 4) {
      // (MFMA phase reads Dy[cur], written one iteration ago)
    } else {
      if (have_next) {
        bf16x8* d8 = reinterpret_cast<bf16x8*>(Dy + (cur ^ 1) * PXPAD * CO);
#pragma unroll
        for (int u = 0; u < 4; ++u) d8[tid + u * 256] = v[u];
        cur ^= 1;
      }
      continue;
    }

#pragma unroll
    for (int sl = 0; sl < 4; ++sl) {
      const int pxbase = sl * 32;
      bf16x8 afrag[2];
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        bf16x8 f;
        const bf16* dycur = Dy + cur * PXPAD * CO;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          f[j] = dycur[(pxbase + kgrp * 8 + j) * CO + (mw + m) * 16 + row16];
        afrag[m] = f;
      }
#pragma unroll
      for (int c = 0; c < 5; ++c) {
        const int k = (cw + c) * 16 + row16;
        const int kh = k / (KW * CI);
        const int r21 = k - kh * (KW * CI);
        // ring slot for this tap's input row; pad taps zero-selected
        const int ko = ((ho * STRIDE - PADDING + kh) & 7) * XTROW +
                       (XTOFF - PADDING * CI) + r21;
        const bool kz = k >= KTAP;
        bf16x8 bfr;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          bfr[j] = kz ? (bf16)0.0f
                      : xt[ko + (pxbase + kgrp * 8 + j) * (STRIDE * CI)];
#pragma unroll
        for (int m = 0; m < 2; ++m)
          acc[m][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfr, acc[m][c], 0, 0, 0);
      }
    }
    if (have_next) {
      bf16x8* d8 = reinterpret_cast<bf16x8*>(Dy + (cur ^ 1) * PXPAD * CO);
#pragma unroll
      for (int u = 0; u < 4; ++u) d8[tid + u * 256] = v[u];
      cur ^= 1;
    }
  }

  // D row = co (4*(lane>>4)+r), col = k (c*16 + lane&15)
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int c = 0; c < 5; ++c)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = (mw + m) * 16 + kgrp * 4 + r;
        const int k = (cw + c) * 16 + row16;
        atomicAdd(&dw[co * KPAD + k], acc[m][c][r]);
      }
}

// dw fp32 [CO][KPAD] -> bf16 grad in the conv weight's layout [CO][KTAP]
extern "C" __global__ __launch_bounds__(256) void stem_wrw_cast_kernel(
    const float* __restrict__ dw, bf16* __restrict__ out) {
  const int i = blockIdx.x * 256 + threadIdx.x;
  if (i < CO * KTAP) {
    const int co = i / KTAP, k = i - co * KTAP;
    out[i] = (bf16)dw[co * KPAD + k];
  }
}

extern "C" void launch_stem_conv_fwd(const void* x, const void* w, void* out,
                                     void* wp_scratch,
                                     int Nb, int H, int W, int HO, int WO,
                                     hipStream_t stream, int phase_mask) {
  hipLaunchKernelGGL(stem_pad_weights_kernel,
                     dim3((CO * KPAD + 255) / 256), dim3(256), 0, stream,
                     (const bf16*)w, (bf16*)wp_scratch);
  hipLaunchKernelGGL(stem_conv_fwd_kernel, dim3(Nb * HO), dim3(256), 0,
                     stream, (const bf16*)x, (const bf16*)wp_scratch,
                     (bf16*)out, Nb, H, W, HO, WO, phase_mask);
}

extern "C" void launch_stem_conv_wrw(const void* x, const void* dy,
                                     float* dw_f32, void* dw_bf16,
                                     int Nb, int H, int W, int HO, int WO,
                                     hipStream_t stream,
                                     int phase_mask = 7) {
  const int nrows = Nb * HO;
  const int target_blocks = 768;
  const int rpb = (nrows + target_blocks - 1) / target_blocks;
  const int blocks = (nrows + rpb - 1) / rpb;
  hipMemsetAsync(dw_f32, 0, CO * KPAD * sizeof(float), stream);
  hipLaunchKernelGGL(stem_conv_wrw_kernel, dim3(blocks), dim3(256), 0,
                     stream, (const bf16*)x, (const bf16*)dy, dw_f32,
                     Nb, H, W, HO, WO, rpb, phase_mask);
  hipLaunchKernelGGL(stem_wrw_cast_kernel,
                     dim3((CO * KTAP + 255) / 256), dim3(256), 0, stream,
                     dw_f32, (bf16*)dw_bf16);
}
