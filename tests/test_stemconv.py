"""MFMA stem conv (ops/csrc/stemconv.hip) vs the fp32 torch reference.

Numerics contract per the repo convention: the HIP kernel is compared
against plain PyTorch fp32 F.conv2d on the same data.
"""
import os

import pytest
import torch
import torch.nn.functional as F

from mi355x_scale.ops.stemconv import StemConv2d, _StemConvFn


def _mk(n=8, h=64, w=64, seed=0, dev="cuda"):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 3, h, w, generator=g).to(dev)
    wt = (torch.randn(64, 3, 7, 7, generator=g) * 0.05).to(dev)
    return x, wt


def test_stemconv_cpu_fallback_matches_conv2d():
    m = StemConv2d()
    x = torch.randn(2, 3, 32, 32)
    want = F.conv2d(x, m.weight, None, 2, 3)
    assert torch.equal(m(x), want)


@pytest.mark.gpu
def test_stemconv_fwd_matches_fp32_reference():
    x, wt = _mk(n=8, h=96, w=96)
    ref = F.conv2d(x, wt, None, 2, 3)  # fp32 reference
    xb = x.to(torch.bfloat16).to(memory_format=torch.channels_last)
    wb = wt.to(torch.bfloat16).to(memory_format=torch.channels_last)
    out = _StemConvFn.apply(xb, wb).float()
    torch.cuda.synchronize()
    rel = ((out - ref).norm() / ref.norm()).item()
    assert rel < 2e-2, f"fwd rel L2 {rel}"
    # bf16 F.conv2d comparison (same input rounding) must be tighter
    ref_bf = F.conv2d(xb.float(), wb.float(), None, 2, 3)
    rel_bf = ((out - ref_bf).norm() / ref_bf.norm()).item()
    assert rel_bf < 1e-2, f"fwd vs bf16-input fp32 conv rel {rel_bf}"


@pytest.mark.gpu
def test_stemconv_fwd_odd_sizes_and_padding_edges():
    # odd spatial size -> exercises the zero-padding boundary logic and
    # the partial final pixel block
    x, wt = _mk(n=3, h=75, w=53, seed=4)
    ref = F.conv2d(x.to(torch.bfloat16).float(),
                   wt.to(torch.bfloat16).float(), None, 2, 3)
    xb = x.to(torch.bfloat16).to(memory_format=torch.channels_last)
    wb = wt.to(torch.bfloat16).to(memory_format=torch.channels_last)
    out = _StemConvFn.apply(xb, wb).float()
    torch.cuda.synchronize()
    rel = ((out - ref).norm() / ref.norm()).item()
    assert rel < 1e-2, f"odd-size fwd rel L2 {rel}"


@pytest.mark.gpu
def test_stemconv_wrw_matches_fp32_reference():
    x, wt = _mk(n=6, h=96, w=96, seed=2)
    xf = x.clone().requires_grad_(False)
    wf = wt.clone().requires_grad_(True)
    out = F.conv2d(xf, wf, None, 2, 3)
    g = torch.Generator().manual_seed(9)
    dy = torch.randn(out.shape, generator=g).to(x.device)
    out.backward(dy)
    ref_dw = wf.grad.clone()

    xb = x.to(torch.bfloat16).to(memory_format=torch.channels_last)
    wb = wt.to(torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    out2 = _StemConvFn.apply(xb, wb)
    out2.backward(dy.to(torch.bfloat16).to(
        memory_format=torch.channels_last))
    torch.cuda.synchronize()
    got = wb.grad.float()
    rel = ((got - ref_dw).norm() / ref_dw.norm()).item()
    assert rel < 3e-2, f"wrw rel L2 {rel}"
    cos = torch.nn.functional.cosine_similarity(
        got.reshape(-1).double(), ref_dw.reshape(-1).double(), dim=0).item()
    assert cos > 0.999


@pytest.mark.gpu
def test_stemconv_module_graph_capturable():
    """The stem module must replay inside a hipGraph (the flagship step
    captures it) — capture one fwd+bwd, replay, finite outputs."""
    m = StemConv2d().to("cuda").to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    x = torch.randn(4, 3, 64, 64, device="cuda").to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    for _ in range(2):  # warmup
        m(x).sum().backward()
        m.weight.grad = None
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g, capture_error_mode="thread_local"):
        y = m(x)
        loss = y.float().sum()
        loss.backward()
    for _ in range(3):
        g.replay()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    assert torch.isfinite(m.weight.grad.float()).all()
