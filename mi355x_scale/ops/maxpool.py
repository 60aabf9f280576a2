"""Fused NHWC bf16 max-pool 3x3/stride-2/pad-1 — the stem pool of the
reference's ResNet family (``deep_learning/2...py:150``).

Replaces torch's at::native NHWC max-pool pair, which cost 233 us fwd +
517 us bwd per flagship step on MI355X and saves int64 indices (8 B per
output element); this kernel records u8 window codes (see
ops/csrc/maxpool.hip) and runs an atomics-free pull-style backward. The
torch composition is the CPU path and numerics oracle.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _C, require_ext


def _supported(x: torch.Tensor) -> bool:
    c = x.shape[1]
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and c % 8 == 0 and c <= 2048 and (256 % (c // 8)) == 0)


class _MaxPool3x3s2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        n, c, h, w = x.shape
        ho, wo = (h + 1) // 2, (w + 1) // 2
        y = torch.empty((n, c, ho, wo), dtype=x.dtype, device=x.device,
                        memory_format=torch.channels_last)
        code = torch.empty(n * ho * wo * c, dtype=torch.uint8,
                           device=x.device)
        _C.maxpool3x3s2_fwd(x, y, code)
        ctx.save_for_backward(code)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (code,) = ctx.saved_tensors
        if not dy.is_contiguous(memory_format=torch.channels_last):
            dy = dy.contiguous(memory_format=torch.channels_last)
        dx = torch.empty(ctx.in_shape, dtype=dy.dtype, device=dy.device,
                         memory_format=torch.channels_last)
        _C.maxpool3x3s2_bwd(dy, code, dx)
        return dx


class MaxPool3x3s2(nn.Module):
    """Drop-in for ``nn.MaxPool2d(3, stride=2, padding=1)``."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _supported(x):
            require_ext()
            x = x.contiguous(memory_format=torch.channels_last)
            return _MaxPool3x3s2Fn.apply(x)
        return F.max_pool2d(x, 3, stride=2, padding=1)
