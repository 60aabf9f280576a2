// NHWC bf16 max-pool 3x3/stride-2/pad-1 (the ResNet stem pool) for
// gfx950.
//
// torch's at::native max_pool2d NHWC kernels on this shape cost 233 us
// forward + 517 us backward per step on MI355X (bs 212: in
// [212,64,112,112], out [212,64,56,56]) and save int64 argmax indices
// (8 B per output element). Here:
//   * forward records the argmax as a u8 window code (0..8) — 5.3 MB
//     instead of 42 MB of index traffic;
//   * backward is atomics-free: each INPUT element checks the <=4
//     output windows that contain it and pulls dy where the code
//     matches (torch's NHWC backward scatters with atomics).
// Both kernels use the BN-style mapping: consecutive threads cover
// consecutive 8-channel slots, vectorized 16 B loads/stores.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __hip_bfloat16 bf16;

#define VEC 8

union BVec {
  uint4 u;
  bf16 h[VEC];
};

union CVec {
  uint2 u;
  unsigned char c[VEC];
};

// out[n,ho,wo,c] = max over 3x3 window; code = argmax position (ih*3+iw
// relative to window origin 2*ho-1, 2*wo-1).
__global__ __launch_bounds__(256) void maxpool_fwd_kernel(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    unsigned char* __restrict__ code, int N, int H, int W, int Ho, int Wo,
    int C) {
  const int lanes = C / VEC;
  const int lane = threadIdx.x % lanes;
  const int rsub = threadIdx.x / lanes;
  const int rows_per_iter = blockDim.x / lanes;
  const long long out_rows = (long long)N * Ho * Wo;
  const long long rstride = (long long)gridDim.x * rows_per_iter;

  for (long long row = (long long)blockIdx.x * rows_per_iter + rsub;
       row < out_rows; row += rstride) {
    // 32-bit decomposition: the 64-bit / and % pair is a ~300-cycle
    // software routine and was a measurable slice of the kernel
    const int r32 = (int)row;
    const int wo = r32 % Wo;
    const int t32 = r32 / Wo;
    const int ho = t32 % Ho;
    const int n = t32 / Ho;
    int h0 = 2 * ho - 1, w0 = 2 * wo - 1;

    float best[VEC];
    unsigned char bcode[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      best[j] = -1e30f;
      bcode[j] = 0;
    }
    for (int ih = 0; ih < 3; ++ih) {
      int h = h0 + ih;
      if (h < 0 || h >= H) continue;
      for (int iw = 0; iw < 3; ++iw) {
        int w = w0 + iw;
        if (w < 0 || w >= W) continue;
        const long long off =
            (((long long)n * H + h) * W + w) * C + (long long)lane * VEC;
        BVec v;
        v.u = *reinterpret_cast<const uint4*>(x + off);
        unsigned char pc = (unsigned char)(ih * 3 + iw);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = __bfloat162float(v.h[j]);
          if (f > best[j]) {
            best[j] = f;
            bcode[j] = pc;
          }
        }
      }
    }
    const long long ooff = row * C + (long long)lane * VEC;
    BVec o;
    CVec cv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      o.h[j] = __float2bfloat16(best[j]);
      cv.c[j] = bcode[j];
    }
    *reinterpret_cast<uint4*>(y + ooff) = o.u;
    *reinterpret_cast<uint2*>(code + ooff) = cv.u;
  }
}

// Fused stem forward: BatchNorm affine + ReLU applied INLINE to the
// conv output while pooling — the 340 MB normalized activation map
// (bs 212) is never materialized. Legal because the recompute-mask BN
// backward (fused_bn.hip) no longer needs the stored y: the map's only
// consumer was this pool. Window taps re-read x up to 2.25x, but those
// hits land in L1/L2; the eliminated full write+read of y is HBM.
__global__ __launch_bounds__(256) void bn_maxpool_fwd_kernel(
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ weight,
    const float* __restrict__ bias, bf16* __restrict__ y,
    unsigned char* __restrict__ code, int N, int H, int W, int Ho, int Wo,
    int C) {
  const int lanes = C / VEC;
  const int lane = threadIdx.x % lanes;
  const int rsub = threadIdx.x / lanes;
  const int rows_per_iter = blockDim.x / lanes;
  const long long out_rows = (long long)N * Ho * Wo;
  const long long rstride = (long long)gridDim.x * rows_per_iter;

  float sc[VEC], sh[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    const int c = lane * VEC + j;
    sc[j] = weight[c] * invstd[c];            // y = sc*x + sh, then relu
    sh[j] = bias[c] - mean[c] * sc[j];
  }

  for (long long row = (long long)blockIdx.x * rows_per_iter + rsub;
       row < out_rows; row += rstride) {
    const int r32 = (int)row;
    const int wo = r32 % Wo;
    const int t32 = r32 / Wo;
    const int ho = t32 % Ho;
    const int n = t32 / Ho;
    int h0 = 2 * ho - 1, w0 = 2 * wo - 1;

    float best[VEC];
    unsigned char bcode[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      best[j] = -1e30f;
      bcode[j] = 0;
    }
    for (int ih = 0; ih < 3; ++ih) {
      int h = h0 + ih;
      if (h < 0 || h >= H) continue;
      for (int iw = 0; iw < 3; ++iw) {
        int w = w0 + iw;
        if (w < 0 || w >= W) continue;
        const long long off =
            (((long long)n * H + h) * W + w) * C + (long long)lane * VEC;
        BVec v;
        v.u = *reinterpret_cast<const uint4*>(x + off);
        unsigned char pc = (unsigned char)(ih * 3 + iw);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = sc[j] * __bfloat162float(v.h[j]) + sh[j];
          if (f < 0.f) f = 0.f;
          if (f > best[j]) {
            best[j] = f;
            bcode[j] = pc;
          }
        }
      }
    }
    const long long ooff = row * C + (long long)lane * VEC;
    BVec o;
    CVec cv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      o.h[j] = __float2bfloat16(best[j]);
      cv.c[j] = bcode[j];
    }
    *reinterpret_cast<uint4*>(y + ooff) = o.u;
    *reinterpret_cast<uint2*>(code + ooff) = cv.u;
  }
}

// LDS-tiled fused stem forward: one block per OUTPUT ROW stages its
// three (1-px zero-padded) input rows once, and the 9-tap windows read
// LDS instead of issuing ~9x L1 requests per output element (the
// untiled kernel measured ~1.4x its adjusted roofline on L1 request
// bandwidth). Used when the padded band fits LDS (W+2)*C*3 bf16.
__global__ __launch_bounds__(256) void bn_maxpool_fwd_tiled_kernel(
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ weight,
    const float* __restrict__ bias, bf16* __restrict__ y,
    unsigned char* __restrict__ code, int N, int H, int W, int Ho, int Wo,
    int C) {
  extern __shared__ char lds_raw[];
  bf16* xt = reinterpret_cast<bf16*>(lds_raw);  // [3][(W+2)*C]
  const int tid = threadIdx.x;
  const int xrow = (W + 2) * C;
  const int n = blockIdx.x / Ho;
  const int ho = blockIdx.x - n * Ho;
  const int h0 = 2 * ho - 1;

  // stage the three padded input rows (zeros at w = -1, W and for
  // out-of-image rows), bf16x8 chunks where C is a multiple of 8
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
  for (int r = 0; r < 3; ++r) {
    const int ih = h0 + r;
    bf16* dst = xt + r * xrow;
    for (int i = tid; i < C / 8; i += 256) {  // left pad px
      reinterpret_cast<bf16x8v*>(dst)[i] = (bf16x8v)(__bf16)0.0f;
      reinterpret_cast<bf16x8v*>(dst + (W + 1) * C)[i] =
          (bf16x8v)(__bf16)0.0f;
    }
    if (ih >= 0 && ih < H) {
      const bf16x8v* src = reinterpret_cast<const bf16x8v*>(
          x + (((long long)n * H + ih) * W) * C);
      bf16x8v* d8 = reinterpret_cast<bf16x8v*>(dst + C);
      const int nch = (W * C) >> 3;
      for (int base = 0; base < nch; base += 256 * 8) {
        bf16x8v v[8];
        int ii[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int i = base + tid + u * 256;
          ii[u] = (i < nch) ? i : -1;
          if (ii[u] >= 0) v[u] = src[i];
        }
#pragma unroll
        for (int u = 0; u < 8; ++u)
          if (ii[u] >= 0) d8[ii[u]] = v[u];
      }
    } else {
      bf16x8v* d8 = reinterpret_cast<bf16x8v*>(dst + C);
      for (int i = tid; i < (W * C) >> 3; i += 256)
        d8[i] = (bf16x8v)(__bf16)0.0f;
    }
  }
  __syncthreads();

  const int lanes = C / VEC;
  const long long rowbase = ((long long)n * Ho + ho) * Wo;
  for (int item = tid; item < Wo * lanes; item += 256) {
    const int wo = item / lanes;
    const int lane = item - wo * lanes;
    float sc[VEC], sh[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      const int c = lane * VEC + j;
      sc[j] = weight[c] * invstd[c];
      sh[j] = bias[c] - mean[c] * sc[j];
    }
    float best[VEC];
    unsigned char bcode[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      best[j] = -1e30f;
      bcode[j] = 0;
    }
    // padded band: window col (2*wo - 1 + iw) lands at LDS px index +1
    const int base_px = 2 * wo;  // = (2*wo - 1) + 1
#pragma unroll
    for (int ih = 0; ih < 3; ++ih) {
#pragma unroll
      for (int iw = 0; iw < 3; ++iw) {
        const bf16* p =
            xt + ih * xrow + (base_px + iw) * C + lane * VEC;
        BVec v;
        v.u = *reinterpret_cast<const uint4*>(p);
        const unsigned char pc = (unsigned char)(ih * 3 + iw);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = sc[j] * __bfloat162float(v.h[j]) + sh[j];
          if (f < 0.f) f = 0.f;
          if (f > best[j]) {
            best[j] = f;
            bcode[j] = pc;
          }
        }
      }
    }
    const long long ooff = (rowbase + wo) * C + (long long)lane * VEC;
    BVec o;
    CVec cv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      o.h[j] = __float2bfloat16(best[j]);
      cv.c[j] = bcode[j];
    }
    *reinterpret_cast<uint4*>(y + ooff) = o.u;
    *reinterpret_cast<uint2*>(code + ooff) = cv.u;
  }
}

// Backward, one 2x2 INPUT tile per thread-lane. The tile at input
// origin (2i, 2j) is touched only by the four output windows
// (i,j), (i,j+1), (i+1,j), (i+1,j+1), and each of the tile's pixels sits
// at a FIXED position inside those windows (k=3, s=2, p=1):
//     (2i  ,2j  ): (i,j)@4
//     (2i  ,2j+1): (i,j)@5   (i,j+1)@3
//     (2i+1,2j  ): (i,j)@7   (i+1,j)@1
//     (2i+1,2j+1): (i,j)@8   (i,j+1)@6  (i+1,j)@2  (i+1,j+1)@0
// so every dy/code vector is loaded once and reused for the whole tile
// (2.25x fewer loads than the per-pixel form, and all code compares are
// against constants). Tiles == the output grid: i in [0,Ho), j in [0,Wo).
__global__ __launch_bounds__(256) void maxpool_bwd_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ code,
    bf16* __restrict__ dx, int N, int H, int W, int Ho, int Wo, int C) {
  const int lanes = C / VEC;
  const int lane = threadIdx.x % lanes;
  const int rsub = threadIdx.x / lanes;
  const int rows_per_iter = blockDim.x / lanes;
  const long long tiles = (long long)N * Ho * Wo;
  const long long rstride = (long long)gridDim.x * rows_per_iter;

  for (long long row = (long long)blockIdx.x * rows_per_iter + rsub;
       row < tiles; row += rstride) {
    const int r32 = (int)row;
    const int j = r32 % Wo;
    const int t32 = r32 / Wo;
    const int i = t32 % Ho;
    const int n = t32 / Ho;
    const long long lvec = (long long)lane * VEC;

    // load the (up to) four windows' dy + code vectors
    BVec g00{}, g01{}, g10{}, g11{};
    CVec c00{}, c01{}, c10{}, c11{};
    bool v01 = (j + 1 < Wo), v10 = (i + 1 < Ho);
    {
      const long long base =
          (((long long)n * Ho + i) * Wo + j) * C + lvec;
      g00.u = *reinterpret_cast<const uint4*>(dy + base);
      c00.u = *reinterpret_cast<const uint2*>(code + base);
      if (v01) {
        g01.u = *reinterpret_cast<const uint4*>(dy + base + C);
        c01.u = *reinterpret_cast<const uint2*>(code + base + C);
      }
      if (v10) {
        const long long b2 = base + (long long)Wo * C;
        g10.u = *reinterpret_cast<const uint4*>(dy + b2);
        c10.u = *reinterpret_cast<const uint2*>(code + b2);
        if (v01) {
          g11.u = *reinterpret_cast<const uint4*>(dy + b2 + C);
          c11.u = *reinterpret_cast<const uint2*>(code + b2 + C);
        }
      }
    }

    const int h0 = 2 * i, w0 = 2 * j;
    const bool r1 = (h0 + 1 < H), cL1 = (w0 + 1 < W);
    BVec o00, o01, o10, o11;
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float a00 = (c00.c[k] == 4) ? __bfloat162float(g00.h[k]) : 0.f;
      float a01 = (c00.c[k] == 5) ? __bfloat162float(g00.h[k]) : 0.f;
      if (v01 && c01.c[k] == 3) a01 += __bfloat162float(g01.h[k]);
      float a10 = (c00.c[k] == 7) ? __bfloat162float(g00.h[k]) : 0.f;
      if (v10 && c10.c[k] == 1) a10 += __bfloat162float(g10.h[k]);
      float a11 = (c00.c[k] == 8) ? __bfloat162float(g00.h[k]) : 0.f;
      if (v01 && c01.c[k] == 6) a11 += __bfloat162float(g01.h[k]);
      if (v10 && c10.c[k] == 2) a11 += __bfloat162float(g10.h[k]);
      if (v10 && v01 && c11.c[k] == 0) a11 += __bfloat162float(g11.h[k]);
      o00.h[k] = __float2bfloat16(a00);
      o01.h[k] = __float2bfloat16(a01);
      o10.h[k] = __float2bfloat16(a10);
      o11.h[k] = __float2bfloat16(a11);
    }
    const long long ibase = (((long long)n * H + h0) * W + w0) * C + lvec;
    *reinterpret_cast<uint4*>(dx + ibase) = o00.u;
    if (cL1) *reinterpret_cast<uint4*>(dx + ibase + C) = o01.u;
    if (r1) {
      const long long ib2 = ibase + (long long)W * C;
      *reinterpret_cast<uint4*>(dx + ib2) = o10.u;
      if (cL1) *reinterpret_cast<uint4*>(dx + ib2 + C) = o11.u;
    }
  }
}

static int mp_grid(long long rows, int C) {
  int rows_per_iter = 256 / (C / VEC);
  long long blocks = (rows + rows_per_iter - 1) / rows_per_iter;
  if (blocks > 2080) blocks = 2080;
  return (int)(blocks > 0 ? blocks : 1);
}

extern "C" void launch_bn_maxpool_fwd(const void* x, const float* mean,
                                      const float* invstd,
                                      const float* weight,
                                      const float* bias, void* y,
                                      void* code, int N, int H, int W,
                                      int Ho, int Wo, int C,
                                      hipStream_t stream) {
  const size_t band = (size_t)3 * (W + 2) * C * sizeof(bf16);
  if (band <= 100 * 1024 && (W * C) % 8 == 0 && H >= 2) {
    static bool attr_done = false;
    if (!attr_done) {
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(bn_maxpool_fwd_tiled_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
      attr_done = true;
    }
    hipLaunchKernelGGL(bn_maxpool_fwd_tiled_kernel, dim3(N * Ho),
                       dim3(256), band, stream, (const bf16*)x, mean,
                       invstd, weight, bias, (bf16*)y,
                       (unsigned char*)code, N, H, W, Ho, Wo, C);
    return;
  }
  hipLaunchKernelGGL(bn_maxpool_fwd_kernel,
                     dim3(mp_grid((long long)N * Ho * Wo, C)), dim3(256),
                     0, stream, (const bf16*)x, mean, invstd, weight,
                     bias, (bf16*)y, (unsigned char*)code, N, H, W, Ho,
                     Wo, C);
}

extern "C" void launch_maxpool_fwd(const void* x, void* y,
                                   unsigned char* code, int N, int H, int W,
                                   int Ho, int Wo, int C,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(maxpool_fwd_kernel,
                     dim3(mp_grid((long long)N * Ho * Wo, C)), dim3(256), 0,
                     stream, (const bf16*)x, (bf16*)y, code, N, H, W, Ho,
                     Wo, C);
}

extern "C" void launch_maxpool_bwd(const void* dy,
                                   const unsigned char* code, void* dx,
                                   int N, int H, int W, int Ho, int Wo,
                                   int C, hipStream_t stream) {
  hipLaunchKernelGGL(maxpool_bwd_kernel,
                     dim3(mp_grid((long long)N * H * W, C)), dim3(256), 0,
                     stream, (const bf16*)dy, code, (bf16*)dx, N, H, W, Ho,
                     Wo, C);
}
