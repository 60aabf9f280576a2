// Torch extension bindings for the MI355X HIP kernels (module `_C`).
// Compiled under PyTorch-ROCm; kernels live in the .hip translation units.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

extern "C" void launch_normalize_u8_to_bf16(
    const uint8_t* in, void* out, const float* scale_dev,
    const float* shift_dev, long long n, hipStream_t stream);

// in:  uint8 contiguous [N,H,W,3] (NHWC)
// out: bf16 tensor with identical memory order (channels_last NCHW view
//      is constructed Python-side)
// scale/shift: float32 device tensors of 3 elements:
//      scale[c] = 1/(255*std[c]), shift[c] = -mean[c]/std[c]
void normalize_u8_to_bf16(torch::Tensor in, torch::Tensor out,
                          torch::Tensor scale, torch::Tensor shift) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda(), "tensors must be on GPU");
  TORCH_CHECK(in.scalar_type() == torch::kUInt8, "input must be uint8");
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "output must be bf16");
  TORCH_CHECK(in.is_contiguous(), "input must be contiguous (NHWC)");
  TORCH_CHECK(in.numel() == out.numel(), "element count mismatch");
  TORCH_CHECK(scale.is_cuda() && shift.is_cuda() && scale.numel() == 3 &&
                  shift.numel() == 3,
              "scale/shift must be 3-element device float tensors");
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32 &&
              shift.scalar_type() == torch::kFloat32,
              "scale/shift must be float32");
  auto stream = at::cuda::getCurrentHIPStream();
  launch_normalize_u8_to_bf16(
      in.data_ptr<uint8_t>(), out.data_ptr(),
      scale.data_ptr<float>(), shift.data_ptr<float>(),
      (long long)in.numel(), stream.stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("normalize_u8_to_bf16", &normalize_u8_to_bf16,
        "fused uint8 NHWC -> normalized bf16 (same memory order)");
}
