// Hand-written xGMI p2p primitives (SURVEY §2.4 deliverable c): IPC
// buffer sharing + a multi-source reduce kernel, the building blocks of
// the direct peer-to-peer all-reduce in parallel/p2p_allreduce.py.
//
// Rationale (MI355X topology): each GPU has 7 point-to-point xGMI links
// at ~153 GB/s. RCCL's ring all-reduce is bound by ONE link regardless
// of ring size; a direct reduce-scatter (every rank pulls all peers'
// copies of its own shard — 7 concurrent link reads) + all-gather
// (every rank pulls each peer's reduced shard) uses all 7 links at once.
// For ResNet-18's 45 MB fp32 flat gradient, shard reads are ~5.6 MB per
// link: latency-dominated territory where the two-phase direct pattern
// wins over the ring pipeline.
//
// The reference disables P2P outright (NCCL_P2P_DISABLE=1,
// deep_learning/2.distributed-data-loading-petastorm.py:362-363) as a
// cloud workaround; on MI355X peer access is the whole point.
//
// Cross-process buffer sharing uses hipIpc* (dmabuf mode,
// HSA_ENABLE_IPC_MODE_LEGACY=0). IPC handles cover whole allocations,
// so shareable buffers are allocated HERE via hipMalloc (base pointers),
// not taken from the caching allocator.

#include <hip/hip_runtime.h>

#define CHECK_HIP(expr)                                             \
  do {                                                              \
    hipError_t _e = (expr);                                         \
    if (_e != hipSuccess) return _e;                                \
  } while (0)

extern "C" hipError_t p2p_alloc(void** ptr, size_t bytes) {
  return hipMalloc(ptr, bytes);
}

extern "C" hipError_t p2p_free(void* ptr) { return hipFree(ptr); }

extern "C" hipError_t p2p_get_handle(void* ptr, hipIpcMemHandle_t* h) {
  return hipIpcGetMemHandle(h, ptr);
}

extern "C" hipError_t p2p_open_handle(const hipIpcMemHandle_t* h,
                                      void** ptr) {
  return hipIpcOpenMemHandle(ptr, *h, hipIpcMemLazyEnablePeerAccess);
}

extern "C" hipError_t p2p_close_handle(void* ptr) {
  return hipIpcCloseMemHandle(ptr);
}

// dst[i] += sum_k src_k[i] over up to 8 source pointers (the peers'
// copies of this rank's shard, read across xGMI). float4-vectorized;
// grid-stride so any shard size works.
#define MAX_SRCS 8

struct SrcPtrs {
  const float* p[MAX_SRCS];
};

__global__ __launch_bounds__(256) void reduce_add_kernel(
    float* __restrict__ dst, SrcPtrs srcs, int nsrc, long long n4,
    long long tail_base, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n4; i += stride) {
    float4 acc = reinterpret_cast<float4*>(dst)[i];
    for (int k = 0; k < nsrc; ++k) {
      const float4 v = reinterpret_cast<const float4*>(srcs.p[k])[i];
      acc.x += v.x;
      acc.y += v.y;
      acc.z += v.z;
      acc.w += v.w;
    }
    reinterpret_cast<float4*>(dst)[i] = acc;
  }
  // scalar tail
  for (long long i = tail_base + (long long)blockIdx.x * blockDim.x +
                     threadIdx.x;
       i < n; i += stride) {
    float acc = dst[i];
    for (int k = 0; k < nsrc; ++k) acc += srcs.p[k][i];
    dst[i] = acc;
  }
}

extern "C" void p2p_reduce_add(float* dst, const float** srcs, int nsrc,
                               long long n, hipStream_t stream) {
  SrcPtrs sp{};
  for (int k = 0; k < nsrc && k < MAX_SRCS; ++k) sp.p[k] = srcs[k];
  long long n4 = n / 4;
  long long tail_base = n4 * 4;
  int block = 256;
  long long want = (n4 + block - 1) / block;
  int grid = (int)(want < 1 ? 1 : (want > 2048 ? 2048 : want));
  hipLaunchKernelGGL(reduce_add_kernel, dim3(grid), dim3(block), 0, stream,
                     dst, sp, nsrc, n4, tail_base, n);
}

// Plain device-to-device copy (peer pull over xGMI when src is a mapped
// peer pointer): used by the all-gather phase.
extern "C" hipError_t p2p_copy(void* dst, const void* src, size_t bytes,
                               hipStream_t stream) {
  return hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToDevice, stream);
}

extern "C" hipError_t p2p_can_access_peer(int* can, int dev, int peer) {
  return hipDeviceCanAccessPeer(can, dev, peer);
}
