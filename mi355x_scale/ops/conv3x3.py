"""3x3/s1/p1 CxC conv with the MFMA weight-grad fast path.

Forward and backward-data stay on MIOpen (its igemm fwd/bwd-data tiles
are competitive); the weight grad — MIOpen's weakest leg on these
shapes, plus its fp32-workspace zero + bf16-cast launches and the
AccumulateGrad add into the flat grad view — runs on
``ops/csrc/conv3x3wrw.hip``. Inside hipGraph capture the cast kernel
writes the weight's flat grad view DIRECTLY (the fused-BN fuse_acc
pattern), so the add disappears from the captured step.

STATUS (round 2, measured on MI355X — benchmarks/bench_conv3x3_wrw.py):
numerics verified against the fp32 reference on every block-conv shape,
but two blocking iterations both lost to MIOpen igemm wrw (v1 32x32
tiles: 0.56x at C=64; v2 64x64 tiles / 8-wave blocks / non-atomic chunk
partials: 0.35x at C=64 down to 0.10x at C=512). Ablations localize the
cost to the thin per-row phase structure: one staging latency + three
barriers per 56-px output row, at 1 block/CU (211 VGPRs -> 2
waves/SIMD), cannot fill the machine, and the K-padding waste at
W<=14 is fatal for layer3/4. Default-OFF (``MI355X_CONV_WRW=1`` opts
in); the round-3 path is software-pipelined multi-row K-phases (stage
row r+1 into registers during row r's MFMAs) plus a VGPR diet — see
NOTES_NEXT.md.

``Conv3x3`` subclasses ``nn.Conv2d`` (state-dict compatible, same
init). When enabled, the HIP path takes CUDA bf16 channels-last with
Cin == Cout ∈ {64..2048 step 64}, 9 <= W <= 64; everything else falls
back to the standard conv autograd.
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor):
        out = torch.ops.aten.convolution(
            x, weight, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1)
        ctx.save_for_backward(x, weight)
        return out

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        from . import _C
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = torch.ops.aten.convolution_backward(
                dy, x, weight, None, [1, 1], [1, 1], [1, 1], False,
                [0, 0], 1, [True, False, False])[0]
        C = weight.shape[0]
        nch = _C.conv3x3_wrw_nchunks(x.shape[0], x.shape[2], C, 0)
        scratch = torch.empty(nch * C * 9 * C, dtype=torch.float32,
                              device=x.device)
        g = weight.grad
        fuse = (torch.cuda.is_current_stream_capturing()
                and g is not None and g.dtype == torch.bfloat16
                and g.is_contiguous(memory_format=torch.channels_last))
        dw = g if fuse else torch.empty_like(weight)
        _C.conv3x3_wrw(x.permute(0, 2, 3, 1), dy.permute(0, 2, 3, 1),
                       scratch, dw.permute(0, 2, 3, 1))
        # fuse: the cast kernel wrote the flat grad view; returning None
        # skips autograd's AccumulateGrad add for this weight
        return dx, (None if fuse else dw)


class Conv3x3(nn.Conv2d):
    """3x3/s1/p1 conv (no bias) with the MFMA weight-grad fast path."""

    def __init__(self, cin: int, cout: int):
        super().__init__(cin, cout, 3, stride=1, padding=1, bias=False)

    def _hip_ok(self, x: torch.Tensor) -> bool:
        from . import HAVE_EXT
        C = self.out_channels
        return (HAVE_EXT and x.is_cuda
                and os.environ.get("MI355X_CONV_WRW", "0") == "1"
                and self.in_channels == C
                and C % 64 == 0 and C <= 2048
                and 9 <= x.shape[-1] <= 64
                and x.dtype == torch.bfloat16
                and self.weight.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and self.weight.is_contiguous(
                    memory_format=torch.channels_last))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._hip_ok(x):
            return _Conv3x3Fn.apply(x, self.weight)
        return F.conv2d(x, self.weight, None, 1, 1)
