// Torch-extension bindings for the p2p primitives (module `_p2p`).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <vector>

extern "C" hipError_t p2p_alloc(void** ptr, size_t bytes);
extern "C" hipError_t p2p_free(void* ptr);
extern "C" hipError_t p2p_get_handle(void* ptr, hipIpcMemHandle_t* h);
extern "C" hipError_t p2p_open_handle(const hipIpcMemHandle_t* h, void** p);
extern "C" hipError_t p2p_close_handle(void* ptr);
extern "C" void p2p_reduce_add(float* dst, const float** srcs, int nsrc,
                               long long n, hipStream_t stream);
extern "C" hipError_t p2p_copy(void* dst, const void* src, size_t bytes,
                               hipStream_t stream);
extern "C" hipError_t p2p_can_access_peer(int* can, int dev, int peer);

static void check(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, what, ": ", hipGetErrorString(e));
}

// Allocate an IPC-shareable fp32 buffer (hipMalloc base pointer — the
// caching allocator's sub-block pointers can't be IPC-exported cleanly)
// and wrap it as a torch tensor that frees on destruction.
torch::Tensor alloc_shared_f32(int64_t numel, int64_t device) {
  int prev = 0;
  check(hipGetDevice(&prev), "hipGetDevice");
  check(hipSetDevice((int)device), "hipSetDevice");
  void* ptr = nullptr;
  hipError_t err = p2p_alloc(&ptr, (size_t)numel * 4);
  hipSetDevice(prev);
  check(err, "hipMalloc");
  auto options = torch::TensorOptions()
                     .dtype(torch::kFloat32)
                     .device(torch::kCUDA, (c10::DeviceIndex)device);
  return torch::from_blob(
      ptr, {numel}, [](void* p) { p2p_free(p); }, options);
}

py::bytes get_ipc_handle(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda(), "GPU tensor required");
  hipIpcMemHandle_t h;
  check(p2p_get_handle(t.data_ptr(), &h), "hipIpcGetMemHandle");
  return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
}

int64_t open_ipc_handle(py::bytes handle) {
  std::string s = handle;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
  hipIpcMemHandle_t h;
  std::memcpy(&h, s.data(), sizeof(h));
  void* ptr = nullptr;
  check(p2p_open_handle(&h, &ptr), "hipIpcOpenMemHandle");
  return (int64_t)(uintptr_t)ptr;
}

void close_ipc_handle(int64_t ptr) {
  check(p2p_close_handle((void*)(uintptr_t)ptr), "hipIpcCloseMemHandle");
}

// dst[offset:offset+count] += sum_k src_k[offset:offset+count]
// srcs are raw device pointers (IPC-opened peer buffers).
void reduce_add(torch::Tensor dst, std::vector<int64_t> srcs,
                int64_t offset, int64_t count) {
  TORCH_CHECK(dst.is_cuda() && dst.scalar_type() == torch::kFloat32 &&
                  dst.is_contiguous(),
              "dst must be contiguous f32 on GPU");
  TORCH_CHECK(offset >= 0 && offset + count <= dst.numel(), "range oob");
  TORCH_CHECK(srcs.size() <= 8, "at most 8 sources");
  std::vector<const float*> sp;
  for (int64_t p : srcs)
    sp.push_back(reinterpret_cast<const float*>((uintptr_t)p) + offset);
  p2p_reduce_add(dst.data_ptr<float>() + offset, sp.data(),
                 (int)sp.size(), count,
                 at::cuda::getCurrentHIPStream().stream());
}

// dst[offset:offset+count] = src_ptr[offset:offset+count]  (peer pull)
void copy_from_peer(torch::Tensor dst, int64_t src_ptr, int64_t offset,
                    int64_t count) {
  TORCH_CHECK(dst.is_cuda() && dst.scalar_type() == torch::kFloat32 &&
                  dst.is_contiguous(),
              "dst must be contiguous f32 on GPU");
  TORCH_CHECK(offset >= 0 && offset + count <= dst.numel(), "range oob");
  check(p2p_copy(dst.data_ptr<float>() + offset,
                 reinterpret_cast<const float*>((uintptr_t)src_ptr) + offset,
                 (size_t)count * 4,
                 at::cuda::getCurrentHIPStream().stream()),
        "p2p copy");
}

bool can_access_peer(int64_t dev, int64_t peer) {
  int can = 0;
  check(p2p_can_access_peer(&can, (int)dev, (int)peer),
        "hipDeviceCanAccessPeer");
  return can != 0;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("alloc_shared_f32", &alloc_shared_f32,
        "IPC-shareable fp32 device buffer (hipMalloc base pointer)");
  m.def("get_ipc_handle", &get_ipc_handle);
  m.def("open_ipc_handle", &open_ipc_handle);
  m.def("close_ipc_handle", &close_ipc_handle);
  m.def("reduce_add", &reduce_add,
        "dst[range] += sum of peer buffers over the same range");
  m.def("copy_from_peer", &copy_from_peer);
  m.def("can_access_peer", &can_access_peer);
}
