"""Device-side image preprocessing (fused uint8→bf16 normalize).

GPU path: one HIP kernel (``csrc/preprocess.hip``) converts a uint8 NHWC
batch to a normalized bf16 channels-last tensor in a single HBM pass.
CPU path (and the numerics reference in tests): plain torch ops in fp32.

Replaces the reference's per-row CPU transform
(``deep_learning/2.distributed-data-loading-petastorm.py:282-296``).
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)

_consts_cache = {}


def _consts(device: torch.device, mean: Sequence[float],
            std: Sequence[float]) -> Tuple[torch.Tensor, torch.Tensor]:
    key = (device, tuple(mean), tuple(std))
    got = _consts_cache.get(key)
    if got is None:
        scale = torch.tensor([1.0 / (255.0 * s) for s in std],
                             dtype=torch.float32, device=device)
        shift = torch.tensor([-m / s for m, s in zip(mean, std)],
                             dtype=torch.float32, device=device)
        got = _consts_cache[key] = (scale, shift)
    return got


def normalize_images(images_u8: torch.Tensor,
                     mean: Sequence[float] = IMAGENET_MEAN,
                     std: Sequence[float] = IMAGENET_STD,
                     out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """uint8 [N,H,W,3] → bf16 channels-last [N,3,H,W], (x/255 - mean)/std.

    On GPU this is the fused HIP kernel (mandatory — raises if the
    extension is missing). On CPU it is the torch reference path.
    """
    if images_u8.dim() != 4 or images_u8.shape[-1] != 3:
        raise ValueError(f"expected [N,H,W,3] uint8, got {tuple(images_u8.shape)}")
    n, h, w, _ = images_u8.shape
    if images_u8.is_cuda:
        from . import _C, require_ext, HAVE_EXT
        require_ext()
        images_u8 = images_u8.contiguous()
        if out is None:
            out = torch.empty((n, 3, h, w), dtype=torch.bfloat16,
                              device=images_u8.device,
                              memory_format=torch.channels_last)
        scale, shift = _consts(images_u8.device, mean, std)
        _C.normalize_u8_to_bf16(images_u8, out, scale, shift)
        return out
    # CPU / reference path (fp32 math, then cast) — also the test oracle.
    x = images_u8.to(torch.float32) / 255.0
    mean_t = torch.tensor(mean, dtype=torch.float32)
    std_t = torch.tensor(std, dtype=torch.float32)
    x = (x - mean_t) / std_t                      # [N,H,W,3]
    x = x.permute(0, 3, 1, 2)                     # [N,3,H,W] (view)
    return x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
