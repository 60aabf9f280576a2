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
    int wo = (int)(row % Wo);
    int ho = (int)((row / Wo) % Ho);
    int n = (int)(row / ((long long)Ho * Wo));
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

// dx[n,h,w,c] = sum over output windows (ho,wo) containing (h,w) of
// dy[n,ho,wo,c] where code matches this input position.
__global__ __launch_bounds__(256) void maxpool_bwd_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ code,
    bf16* __restrict__ dx, int N, int H, int W, int Ho, int Wo, int C) {
  const int lanes = C / VEC;
  const int lane = threadIdx.x % lanes;
  const int rsub = threadIdx.x / lanes;
  const int rows_per_iter = blockDim.x / lanes;
  const long long in_rows = (long long)N * H * W;
  const long long rstride = (long long)gridDim.x * rows_per_iter;

  for (long long row = (long long)blockIdx.x * rows_per_iter + rsub;
       row < in_rows; row += rstride) {
    int w = (int)(row % W);
    int h = (int)((row / W) % H);
    int n = (int)(row / ((long long)H * W));

    float acc[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] = 0.f;

    // windows: ho with 2*ho-1 <= h <= 2*ho+1  =>  ceil((h-1)/2) <= ho
    // <= floor((h+1)/2); h >= 0 so ceil((h-1)/2) == h/2
    int ho_lo = h >> 1;
    int ho_hi = (h + 1) >> 1;
    int wo_lo = w >> 1;
    int wo_hi = (w + 1) >> 1;
    if (ho_lo < 0) ho_lo = 0;
    if (wo_lo < 0) wo_lo = 0;
    if (ho_hi >= Ho) ho_hi = Ho - 1;
    if (wo_hi >= Wo) wo_hi = Wo - 1;

    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      int ih = h - (2 * ho - 1);
      if (ih < 0 || ih > 2) continue;
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        int iw = w - (2 * wo - 1);
        if (iw < 0 || iw > 2) continue;
        unsigned char want = (unsigned char)(ih * 3 + iw);
        const long long ooff =
            (((long long)n * Ho + ho) * Wo + wo) * C + (long long)lane * VEC;
        CVec cv;
        cv.u = *reinterpret_cast<const uint2*>(code + ooff);
        BVec g;
        g.u = *reinterpret_cast<const uint4*>(dy + ooff);
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          if (cv.c[j] == want) acc[j] += __bfloat162float(g.h[j]);
      }
    }
    const long long ioff = row * C + (long long)lane * VEC;
    BVec o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) o.h[j] = __float2bfloat16(acc[j]);
    *reinterpret_cast<uint4*>(dx + ioff) = o.u;
  }
}

static int mp_grid(long long rows, int C) {
  int rows_per_iter = 256 / (C / VEC);
  long long blocks = (rows + rows_per_iter - 1) / rows_per_iter;
  if (blocks > 2080) blocks = 2080;
  return (int)(blocks > 0 ? blocks : 1);
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
