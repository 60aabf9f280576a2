// Weight-gradient kernel for the ResNet block convs: 3x3 stride-1 pad-1,
// Cin == Cout == C (64..512), NHWC bf16, fp32 accumulation.
//
//   dW[co][kh][kw][ci] = sum_{n,h,w} dy[n,h,w,co] * x[n,h+kh-1,w+kw-1,ci]
//
// v2 blocking (v1 lesson: a 32x32 dW tile duplicated the x/dy staging
// across (C/32)^2 tile combos and never beat MIOpen): block = 512
// threads / 8 waves owning a 64x64 dW tile (dup factor (C/64)^2), each
// wave 2 of the 16 (co16 x ci16) positions x 9 taps = 18 f32x4
// accumulators/lane. Per output row (n, h): a 3-row x band (ci64
// slice, 1-px zero-padded columns) in a 4-slot LDS ring (consecutive
// rows stage ONE new row) + the dy row slice, both staged as bf16x4
// chunks batched 8 deep. Per 32-px K-step a wave builds one dy
// A-fragment and, per (position, band row), TEN consecutive x elements
// that shift into the three (dh, dw) B-fragments
// (v_mfma_f32_16x16x32_bf16; maps per cdna_hip_programming.md §3).
// Chunk partials are written NON-atomically to a per-chunk fp32 buffer
// ([chunk][C][9][C], a few tens of MB); a fused reduce+cast kernel sums
// chunks and writes the bf16 weight-grad view directly (the flat-view
// AccumulateGrad add and MIOpen's workspace zero/cast all disappear).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __bf16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define TC 64                   // dW tile edge (co and ci)
#define MAXW 64
#define XROWE ((MAXW + 2) * TC)    // band slot: (W + 2 pad px) x 64 ci

extern "C" __global__ __launch_bounds__(512) void conv3x3_wrw_kernel(
    const bf16* __restrict__ x,    // [N][H][W][C]
    const bf16* __restrict__ dy,   // [N][H][W][C]
    float* __restrict__ part,      // [chunks][C][9][C] fp32
    int Nb, int H, int W, int C, int rows_per_block, int phase_mask) {
  __shared__ bf16 X[4 * XROWE];          // band ring, slot = ih & 3
  __shared__ bf16 Dy[MAXW * TC];         // current dy row slice
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int row16 = lane & 15;
  const int kgrp = lane >> 4;
  const int ci0 = blockIdx.y * TC;
  const int co0 = blockIdx.z * TC;
  const int m16 = wave & 3;              // wave's co16 position (0..3)
  const int cp = (wave >> 2) * 2;        // wave's ci16 pair base (0 or 2)
  const int nrows = Nb * H;
  const int wpad = (W + 31) & ~31;

  f32x4 acc[2][9];
#pragma unroll
  for (int p = 0; p < 2; ++p)
#pragma unroll
    for (int t = 0; t < 9; ++t) acc[p][t] = (f32x4)0.0f;

  for (int rr = 0; rr < rows_per_block; ++rr) {
    const int row = blockIdx.x * rows_per_block + rr;
    if (row >= nrows) break;
    const int n = row / H;
    const int h = row - n * H;
    const bool prime = (rr == 0) || (h == 0);
    __syncthreads();  // previous row's readers are done
    if (!(phase_mask & 1)) continue;

    if (prime) {
      for (int i = tid; i < 4 * XROWE; i += 512) X[i] = (bf16)0.0f;
      __syncthreads();
    }
    {  // x band: prime = rows h-1..h+1, else only row h+1
      const int rfirst = prime ? -1 : 1;
      for (int dh = rfirst; dh <= 1; ++dh) {
        const int ih = h + dh;
        bf16* slot = X + (ih & 3) * XROWE + TC;  // +TC: left zero pad px
        if (ih >= 0 && ih < H) {
          const bf16* src = x + (((long long)n * H + ih) * W) * C + ci0;
          const int nch = (W * TC) >> 2;
          bf16x4* d4 = reinterpret_cast<bf16x4*>(slot);
          for (int base = 0; base < nch; base += 512 * 8) {
            bf16x4 v[8];
            int ii[8];
#pragma unroll
            for (int u = 0; u < 8; ++u) {
              const int i = base + tid + u * 512;
              ii[u] = (i < nch) ? i : -1;
              if (ii[u] >= 0) {
                const int w = i >> 4, c4 = i & 15;  // TC/4 = 16 chunks/px
                v[u] = *reinterpret_cast<const bf16x4*>(
                    src + (long long)w * C + c4 * 4);
              }
            }
#pragma unroll
            for (int u = 0; u < 8; ++u)
              if (ii[u] >= 0) d4[ii[u]] = v[u];
          }
        } else {
          for (int i = tid; i < W * TC; i += 512) slot[i] = (bf16)0.0f;
        }
      }
    }
    {  // dy row slice [wpad][co64], pad px zeroed
      const bf16* src = dy + (((long long)n * H + h) * W) * C + co0;
      const int nch = (wpad * TC) >> 2;
      bf16x4* d4 = reinterpret_cast<bf16x4*>(Dy);
      const bf16x4 z = {(bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f};
      for (int base = 0; base < nch; base += 512 * 8) {
        bf16x4 v[8];
        int ii[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int i = base + tid + u * 512;
          ii[u] = (i < nch) ? i : -1;
          v[u] = z;
          if (ii[u] >= 0) {
            const int w = i >> 4, c4 = i & 15;
            if (w < W)
              v[u] = *reinterpret_cast<const bf16x4*>(
                  src + (long long)w * C + c4 * 4);
          }
        }
#pragma unroll
        for (int u = 0; u < 8; ++u)
          if (ii[u] >= 0) d4[ii[u]] = v[u];
      }
    }
    __syncthreads();
    if (!(phase_mask & 2)) continue;

    for (int pxb = 0; pxb < wpad; pxb += 32) {
      bf16x8 af;  // dy rows = this wave's co16, k-slice = 8 px
#pragma unroll
      for (int j = 0; j < 8; ++j)
        af[j] = Dy[(pxb + kgrp * 8 + j) * TC + m16 * 16 + row16];
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        const int ci = (cp + p) * 16 + row16;
#pragma unroll
        for (int dh = 0; dh < 3; ++dh) {
          const bf16* slot = X + ((h + dh - 1) & 3) * XROWE;
          bf16 e[10];
#pragma unroll
          for (int j = 0; j < 10; ++j)
            e[j] = slot[(pxb + kgrp * 8 + j) * TC + ci];
          bf16x8 b0, b1, b2;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            b0[j] = e[j];
            b1[j] = e[j + 1];
            b2[j] = e[j + 2];
          }
          acc[p][dh * 3 + 0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, b0, acc[p][dh * 3 + 0], 0, 0, 0);
          acc[p][dh * 3 + 1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, b1, acc[p][dh * 3 + 1], 0, 0, 0);
          acc[p][dh * 3 + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, b2, acc[p][dh * 3 + 2], 0, 0, 0);
        }
      }
    }
  }

  // ---- chunk partial (non-atomic): [chunk][co][t][ci] slice
  if (!(phase_mask & 4)) return;
  float* my = part + ((long long)blockIdx.x * C + 0) * 9 * C;
#pragma unroll
  for (int p = 0; p < 2; ++p)
#pragma unroll
    for (int t = 0; t < 9; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + m16 * 16 + kgrp * 4 + r;
        const int ci = ci0 + (cp + p) * 16 + row16;
        my[((long long)co * 9 + t) * C + ci] = acc[p][t][r];
      }
}

// Sum the chunk partials and write the bf16 weight grad (channels_last
// layout == the fp32 buffer's linear order; may be the flat grad view).
extern "C" __global__ __launch_bounds__(256) void conv3x3_reduce_cast_kernel(
    const float* __restrict__ part, bf16* __restrict__ out,
    long long nel, int chunks) {
  // each thread reduces FOUR elements with the chunk loop batched 8
  // deep — a serial per-element chunk walk pays one memory latency per
  // chunk and alone cost more than MIOpen's whole wrw
  const long long i0 = ((long long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 >= nel) return;
  float s[4] = {0.f, 0.f, 0.f, 0.f};
  int c = 0;
  for (; c + 8 <= chunks; c += 8) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      if (i0 + e >= nel) break;
      float a = 0.f;
#pragma unroll
      for (int u = 0; u < 8; ++u)
        a += part[(long long)(c + u) * nel + i0 + e];
      s[e] += a;
    }
  }
  for (; c < chunks; ++c)
#pragma unroll
    for (int e = 0; e < 4; ++e)
      if (i0 + e < nel) s[e] += part[(long long)c * nel + i0 + e];
#pragma unroll
  for (int e = 0; e < 4; ++e)
    if (i0 + e < nel) out[i0 + e] = (bf16)s[e];
}

extern "C" void launch_conv3x3_wrw(const void* x, const void* dy,
                                   float* part, void* dw_bf16,
                                   int Nb, int H, int W, int C,
                                   int chunks, hipStream_t stream,
                                   int phase_mask) {
  const int tiles = C / TC;
  const int nrows = Nb * H;
  const int rpb = (nrows + chunks - 1) / chunks;
  chunks = (nrows + rpb - 1) / rpb;
  const long long nel = (long long)C * 9 * C;
  hipLaunchKernelGGL(conv3x3_wrw_kernel,
                     dim3(chunks, tiles, tiles), dim3(512), 0, stream,
                     (const bf16*)x, (const bf16*)dy, part,
                     Nb, H, W, C, rpb, phase_mask);
  hipLaunchKernelGGL(conv3x3_reduce_cast_kernel,
                     dim3((int)((nel / 4 + 255) / 256)), dim3(256), 0,
                     stream, part, (bf16*)dw_bf16, nel, chunks);
}

// How many chunks the launcher will actually use for a given request
// (python sizes the partial buffer with this).
extern "C" int conv3x3_wrw_chunks(int Nb, int H, int C, int chunks_req) {
  const int nrows = Nb * H;
  int chunks = chunks_req > 0 ? chunks_req : 768 / ((C / TC) * (C / TC));
  if (chunks < 1) chunks = 1;
  if (chunks > nrows) chunks = nrows;
  const int rpb = (nrows + chunks - 1) / chunks;
  return (nrows + rpb - 1) / rpb;
}
