// Weight-gradient kernel for the ResNet block convs: 3x3 stride-1 pad-1,
// Cin == Cout == C (64..512), NHWC bf16, fp32 accumulation.
//
//   dW[co][kh][kw][ci] = sum_{n,h,w} dy[n,h,w,co] * x[n,h+kh-1,w+kw-1,ci]
//
// These sit in MIOpen igemm wrw kernels at ~19% of the bf16 MFMA peak
// plus per-conv fp32-workspace zero + bf16 cast launches
// (SubTensorOpWithScalar1d/CastTensor1d, ~0.43 ms/step total) plus an
// AccumulateGrad add into the flat grad view. Owning the kernel removes
// all three: one kernel + one cast that writes the flat grad view
// directly.
//
// Decomposition (CDNA4): grid = (row-chunk, ci-tile, co-tile); block =
// 256 threads / 4 waves = the four (co16 x ci16) quadrants of a 32x32
// dW tile; each lane carries NINE f32x4 accumulators (one per tap).
// Per output row (n, h): a 3-row x band (ci32 slice, 1-px zero-padded
// columns) lives in an LDS ring — consecutive rows stage only ONE new
// band row — plus the dy row slice; per 32-px K-step the wave builds
// one dy A-fragment and, per band row dh, TEN consecutive x elements
// that shift into the three (dh, dw) B-fragments, then issues 9 MFMAs
// against the SAME A-fragment (v_mfma_f32_16x16x32_bf16; maps per
// cdna_hip_programming.md §3). Partials atomic-add into the per-conv
// fp32 dW buffer, whose layout [co][kh*3+kw][ci] IS channels_last, so
// the cast kernel writes the weight-grad view linearly.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __bf16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define TCI 32                  // ci tile
#define TCO 32                  // co tile
// LDS X band: 4 ring slots x (W + 2 zero-pad px) x TCI, W <= MAXW
#define MAXW 64
#define XROWE ((MAXW + 2) * TCI)   // elements per band slot

extern "C" __global__ __launch_bounds__(256) void conv3x3_wrw_kernel(
    const bf16* __restrict__ x,    // [N][H][W][C]
    const bf16* __restrict__ dy,   // [N][H][W][C]
    float* __restrict__ dwf,       // [C][9][C] fp32 (pre-zeroed)
    int Nb, int H, int W, int C, int rows_per_block, int phase_mask) {
  __shared__ bf16 X[4 * XROWE];          // band ring, slot = ih & 3
  __shared__ bf16 Dy[MAXW * TCO];        // current dy row slice
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int row16 = lane & 15;
  const int kgrp = lane >> 4;
  const int ci0 = blockIdx.y * TCI;      // this block's ci range
  const int co0 = blockIdx.z * TCO;      // this block's co range
  const int co_off = (wave & 1) * 16;    // wave quadrant
  const int ci_off = (wave >> 1) * 16;
  const int nrows = Nb * H;
  const int wpad = (W + 31) & ~31;       // k-steps cover wpad px

  f32x4 acc[9];
#pragma unroll
  for (int t = 0; t < 9; ++t) acc[t] = (f32x4)0.0f;

  for (int rr = 0; rr < rows_per_block; ++rr) {
    const int row = blockIdx.x * rows_per_block + rr;
    if (row >= nrows) break;
    const int n = row / H;
    const int h = row - n * H;
    const bool prime = (rr == 0) || (h == 0);
    __syncthreads();  // previous row's readers are done
    if (!(phase_mask & 1)) continue;

    // ---- stage the x band (ci32 slice): prime = rows h-1..h+1, else
    // only the new row h+1; out-of-image rows zeroed. Column zero-pads
    // (w = -1, W) are zeroed on prime and never overwritten.
    if (prime) {
      for (int i = tid; i < 4 * XROWE; i += 256) X[i] = (bf16)0.0f;
      __syncthreads();
    }
    {
      const int rfirst = prime ? -1 : 1;
      for (int dh = rfirst; dh <= 1; ++dh) {
        const int ih = h + dh;
        bf16* slot = X + (ih & 3) * XROWE + TCI;  // +TCI: left zero pad
        if (ih >= 0 && ih < H) {
          const bf16* src = x + (((long long)n * H + ih) * W) * C + ci0;
          // bf16x4 chunks (8 B aligned: C and ci0 are multiples of 32),
          // batched 8 deep so the global loads pipeline
          const int nch = (W * TCI) >> 2;
          bf16x4* d4 = reinterpret_cast<bf16x4*>(slot);
          for (int base = 0; base < nch; base += 256 * 8) {
            bf16x4 v[8];
            int ii[8];
#pragma unroll
            for (int u = 0; u < 8; ++u) {
              const int i = base + tid + u * 256;
              ii[u] = (i < nch) ? i : -1;
              if (ii[u] >= 0) {
                const int w = i >> 3, c4 = i & 7;  // TCI/4 = 8 chunks/px
                v[u] = *reinterpret_cast<const bf16x4*>(
                    src + (long long)w * C + c4 * 4);
              }
            }
#pragma unroll
            for (int u = 0; u < 8; ++u)
              if (ii[u] >= 0) d4[ii[u]] = v[u];
          }
        } else {
          for (int i = tid; i < W * TCI; i += 256) slot[i] = (bf16)0.0f;
        }
      }
    }
    // ---- stage the dy row slice [wpad][co32] (pad px zeroed),
    // bf16x4 chunks batched 8 deep
    {
      const bf16* src = dy + (((long long)n * H + h) * W) * C + co0;
      const int nch = (wpad * TCO) >> 2;
      bf16x4* d4 = reinterpret_cast<bf16x4*>(Dy);
      const bf16x4 z = {(bf16)0.0f, (bf16)0.0f, (bf16)0.0f, (bf16)0.0f};
      for (int base = 0; base < nch; base += 256 * 8) {
        bf16x4 v[8];
        int ii[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int i = base + tid + u * 256;
          ii[u] = (i < nch) ? i : -1;
          v[u] = z;
          if (ii[u] >= 0) {
            const int w = i >> 3, c4 = i & 7;
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

    // ---- K-steps over this row's pixels
    for (int pxb = 0; pxb < wpad; pxb += 32) {
      // A-fragment: dy rows = co, k-slice = 8 px
      bf16x8 af;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        af[j] = Dy[(pxb + kgrp * 8 + j) * TCO + co_off + row16];
      // per band row: 10 consecutive x elements shift into the three
      // (dh, dw) B-fragments
#pragma unroll
      for (int dh = 0; dh < 3; ++dh) {
        const bf16* slot = X + ((h + dh - 1) & 3) * XROWE;
        bf16 e[10];
#pragma unroll
        for (int j = 0; j < 10; ++j)
          e[j] = slot[(pxb + kgrp * 8 + j) * TCI + ci_off + row16];
        bf16x8 b0, b1, b2;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          b0[j] = e[j];          // dw = 0 (iw = px - 1 -> pad offset 0)
          b1[j] = e[j + 1];      // dw = 1
          b2[j] = e[j + 2];      // dw = 2
        }
        acc[dh * 3 + 0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, b0, acc[dh * 3 + 0], 0, 0, 0);
        acc[dh * 3 + 1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, b1, acc[dh * 3 + 1], 0, 0, 0);
        acc[dh * 3 + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, b2, acc[dh * 3 + 2], 0, 0, 0);
      }
    }
  }

  // ---- partials: D row = co (4*(lane>>4)+r), col = ci (lane&15)
  if (!(phase_mask & 4)) return;
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int co = co0 + co_off + kgrp * 4 + r;
      const int ci = ci0 + ci_off + row16;
      atomicAdd(&dwf[((long long)co * 9 + t) * C + ci], acc[t][r]);
    }
}

// fp32 [C][9][C] -> bf16 channels_last weight grad (same linear order);
// writes the flat grad view directly (the AccumulateGrad add and
// MIOpen's workspace zero+cast launches all disappear).
extern "C" __global__ __launch_bounds__(256) void conv3x3_wrw_cast_kernel(
    const float* __restrict__ dwf, bf16* __restrict__ out, long long n) {
  const long long i = (long long)blockIdx.x * 256 + threadIdx.x;
  if (i < n) out[i] = (bf16)dwf[i];
}

extern "C" void launch_conv3x3_wrw(const void* x, const void* dy,
                                   float* dwf, void* dw_bf16,
                                   int Nb, int H, int W, int C,
                                   hipStream_t stream, int phase_mask,
                                   int chunks_override) {
  const int tiles = C / TCI;
  const int nrows = Nb * H;
  // enough blocks to fill the chip; fewer chunks for bigger tile grids
  int chunks = chunks_override > 0 ? chunks_override
                                   : 2048 / (tiles * tiles);
  if (chunks < 1) chunks = 1;
  if (chunks > nrows) chunks = nrows;
  const int rpb = (nrows + chunks - 1) / chunks;
  chunks = (nrows + rpb - 1) / rpb;
  const long long nel = (long long)C * 9 * C;
  hipMemsetAsync(dwf, 0, nel * sizeof(float), stream);
  hipLaunchKernelGGL(conv3x3_wrw_kernel,
                     dim3(chunks, tiles, tiles), dim3(256), 0, stream,
                     (const bf16*)x, (const bf16*)dy, dwf,
                     Nb, H, W, C, rpb, phase_mask);
  hipLaunchKernelGGL(conv3x3_wrw_cast_kernel,
                     dim3((int)((nel + 255) / 256)), dim3(256), 0, stream,
                     dwf, (bf16*)dw_bf16, nel);
}
