"""Fused NHWC maxpool vs the torch oracle (fwd values, bwd routing)."""
import pytest
import torch
import torch.nn.functional as F

from mi355x_scale.ops.maxpool import MaxPool3x3s2


def test_cpu_fallback_matches_torch():
    x = torch.randn(2, 8, 14, 14)
    torch.testing.assert_close(MaxPool3x3s2()(x),
                               F.max_pool2d(x, 3, stride=2, padding=1))


@pytest.mark.gpu
@pytest.mark.parametrize("C,hw", [(64, 112), (64, 113), (128, 28)])
def test_gpu_fwd_bwd_vs_fp32(C, hw):
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    x32 = torch.randn(3, C, hw, hw, device=dev)
    xh = (x32.bfloat16().contiguous(memory_format=torch.channels_last)
          .requires_grad_(True))
    y = MaxPool3x3s2()(xh)
    gseed = torch.randn(y.shape, device=dev).bfloat16()
    y.backward(gseed)

    xr = xh.detach().float().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, stride=2, padding=1)
    yr.backward(gseed.float())

    torch.testing.assert_close(y.float(), yr, atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(xh.grad.float(), xr.grad, atol=1e-2,
                               rtol=1e-2)


@pytest.mark.gpu
def test_gpu_tie_routing_matches_torch():
    """Equal values in a window: gradient must go to the FIRST max in
    row-major window order, exactly like torch."""
    dev = torch.device("cuda:0")
    x = torch.ones(1, 8, 8, 8, device=dev)
    xh = (x.bfloat16().contiguous(memory_format=torch.channels_last)
          .requires_grad_(True))
    y = MaxPool3x3s2()(xh)
    y.backward(torch.ones_like(y))
    xr = x.clone().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, stride=2, padding=1)
    yr.backward(torch.ones_like(yr))
    torch.testing.assert_close(xh.grad.float(), xr.grad)
