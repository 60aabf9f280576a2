"""MFMA stem convolution (7x7 stride-2 pad-3, 3->64, NHWC bf16).

The ResNet stem is the one conv MIOpen leaves on a generic igemm tile
(Cin=3 starves the GEMM K: K = 7*7*3 = 147) — MIOpen measured ~298 us
forward + ~294 us weight-grad per step at bs 212 on the flagship bench,
~7% of the whole step. ``ops/csrc/stemconv.hip`` computes it as the
GEMM it is on ``v_mfma_f32_16x16x32_bf16`` with a pixel-padded LDS
input band per output row: forward 210 us (1.73x MIOpen, padded
weights read from L1-resident global), weight-grad 267 us (1.12x;
ring-band staging + a software-pipelined dy stage prefetched into
registers under the MFMA phase).

``StemConv2d`` subclasses ``nn.Conv2d`` (state-dict compatible, same
init); the HIP path engages on CUDA bf16 channels-last inputs with the
exact stem geometry, everything else falls back to ``F.conv2d`` (the
CPU numerics baseline). The stem input needs no data-grad (it is the
normalized image batch), so backward produces only the weight grad.
Replaces the reference's torchvision conv1 (``deep_learning/
2.distributed-data-loading-petastorm.py:150``'s resnet50 stem).
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F


class _StemConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor):
        from . import _C
        N, C, H, W = x.shape
        HO = (H + 2 * 3 - 7) // 2 + 1
        WO = (W + 2 * 3 - 7) // 2 + 1
        out = torch.empty((N, 64, HO, WO), dtype=torch.bfloat16,
                          device=x.device,
                          memory_format=torch.channels_last)
        # kernels address raw NHWC storage: hand the physical views over
        wp = torch.empty(64 * 160, dtype=torch.bfloat16, device=x.device)
        _C.stem_conv_fwd(x.permute(0, 2, 3, 1), weight.permute(0, 2, 3, 1),
                         out.permute(0, 2, 3, 1), wp)
        ctx.save_for_backward(x, weight)
        return out

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        from . import _C
        x, weight = ctx.saved_tensors
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dy = dy.contiguous(memory_format=torch.channels_last)
        dw = torch.empty_like(weight)
        scratch = torch.empty(64 * 160, dtype=torch.float32,
                              device=x.device)
        _C.stem_conv_wrw(x.permute(0, 2, 3, 1), dy.permute(0, 2, 3, 1),
                         scratch, dw.permute(0, 2, 3, 1))
        return None, dw


class StemConv2d(nn.Conv2d):
    """7x7/s2/p3 3->64 conv with the MFMA HIP fast path."""

    def __init__(self):
        super().__init__(3, 64, 7, stride=2, padding=3, bias=False)

    def _hip_ok(self, x: torch.Tensor) -> bool:
        from . import HAVE_EXT
        return (HAVE_EXT and x.is_cuda
                and os.environ.get("MI355X_STEM_CONV", "1") == "1"
                and x.shape[-1] <= 262  # kernel's 128-px output-row cap
                and x.dtype == torch.bfloat16
                and self.weight.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and self.weight.is_contiguous(
                    memory_format=torch.channels_last)
                and not x.requires_grad)  # stem input never needs dx

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._hip_ok(x):
            return _StemConvFn.apply(x, self.weight)
        if x.is_cuda and torch.cuda.is_available():
            from . import HAVE_EXT, require_ext
            if HAVE_EXT is False and x.dtype == torch.bfloat16:
                require_ext()  # no silent eager fallback on a GPU box
        return F.conv2d(x, self.weight, None, self.stride, self.padding)
