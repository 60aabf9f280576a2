import numpy as np
import pytest
import torch

from mi355x_scale.ops.preprocess import (IMAGENET_MEAN, IMAGENET_STD,
                                         normalize_images)


def _reference_fp32(u8_nhwc: torch.Tensor) -> torch.Tensor:
    x = u8_nhwc.to(torch.float32) / 255.0
    mean = torch.tensor(IMAGENET_MEAN)
    std = torch.tensor(IMAGENET_STD)
    return ((x - mean) / std).permute(0, 3, 1, 2)


def test_normalize_cpu_matches_reference():
    torch.manual_seed(0)
    x = torch.randint(0, 256, (2, 8, 8, 3), dtype=torch.uint8)
    out = normalize_images(x)
    ref = _reference_fp32(x)
    assert out.shape == (2, 3, 8, 8)
    assert torch.allclose(out.float(), ref, atol=1e-2)  # bf16 rounding


@pytest.mark.gpu
def test_normalize_gpu_kernel_vs_cpu_fp32():
    """HIP kernel vs plain fp32 torch reference (SURVEY §4 numerics rule)."""
    torch.manual_seed(0)
    x = torch.randint(0, 256, (4, 224, 224, 3), dtype=torch.uint8)
    ref = _reference_fp32(x)
    out = normalize_images(x.cuda()).float().cpu()
    # bf16 has ~3 decimal digits; normalized range ~[-2.2, 2.7]
    assert (out - ref).abs().max() < 2e-2
    # channel order must not be swapped: check per-channel means
    for c in range(3):
        assert abs(out[:, c].mean() - ref[:, c].mean()) < 1e-2


@pytest.mark.gpu
def test_normalize_gpu_is_channels_last_bf16():
    x = torch.randint(0, 256, (2, 32, 32, 3), dtype=torch.uint8).cuda()
    out = normalize_images(x)
    assert out.dtype == torch.bfloat16
    assert out.is_contiguous(memory_format=torch.channels_last)


@pytest.mark.gpu
def test_normalize_gpu_odd_sizes():
    """Tail path: n not a multiple of 16 lanes*16B."""
    x = torch.randint(0, 256, (1, 7, 9, 3), dtype=torch.uint8)
    ref = _reference_fp32(x)
    out = normalize_images(x.cuda()).float().cpu()
    assert (out - ref).abs().max() < 2e-2
