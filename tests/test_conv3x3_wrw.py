"""MFMA 3x3/s1 weight-grad kernel (ops/csrc/conv3x3wrw.hip) vs the fp32
torch reference, across the ResNet block-conv shape family."""
import pytest
import torch
import torch.nn.functional as F

from mi355x_scale.ops.conv3x3 import Conv3x3, _Conv3x3Fn


def test_conv3x3_cpu_fallback_matches_conv2d():
    m = Conv3x3(32, 32)
    x = torch.randn(2, 32, 16, 16)
    assert torch.equal(m(x), F.conv2d(x, m.weight, None, 1, 1))


@pytest.mark.gpu
@pytest.mark.parametrize("C,HW,n", [(64, 56, 4), (128, 28, 6),
                                    (256, 14, 8), (512, 9, 8),
                                    (64, 33, 3)])
def test_conv3x3_wrw_matches_fp32_reference(C, HW, n):
    g = torch.Generator().manual_seed(C + HW)
    x = torch.randn(n, C, HW, HW, generator=g).cuda()
    wt = (torch.randn(C, C, 3, 3, generator=g) * 0.05).cuda()
    dy = torch.randn(n, C, HW, HW, generator=g).cuda()

    wf = wt.clone().requires_grad_(True)
    out = F.conv2d(x, wf, None, 1, 1)
    out.backward(dy)
    ref = wf.grad.clone()

    xb = x.to(torch.bfloat16).to(memory_format=torch.channels_last)
    wb = wt.to(torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    xb2 = xb.clone().requires_grad_(True)
    out2 = _Conv3x3Fn.apply(xb2, wb)
    out2.backward(dy.to(torch.bfloat16).to(
        memory_format=torch.channels_last))
    torch.cuda.synchronize()
    got = wb.grad.float()
    rel = ((got - ref).norm() / ref.norm()).item()
    assert rel < 3e-2, f"C={C} HW={HW}: wrw rel L2 {rel}"
    cos = torch.nn.functional.cosine_similarity(
        got.reshape(-1).double(), ref.reshape(-1).double(), dim=0).item()
    assert cos > 0.999, f"C={C} HW={HW}: cos {cos}"
    # dx (MIOpen path through our Function) must also track the reference
    xf = x.clone().requires_grad_(True)
    F.conv2d(xf, wt, None, 1, 1).backward(dy)
    relx = ((xb2.grad.float() - xf.grad).norm() / xf.grad.norm()).item()
    assert relx < 3e-2, f"dx rel {relx}"


@pytest.mark.gpu
def test_conv3x3_in_graph_writes_view_directly():
    """Captured step: the cast kernel writes the flat grad view and the
    AccumulateGrad add is skipped — grads must still match eager."""
    C, HW = 64, 28
    torch.manual_seed(0)
    m = Conv3x3(C, C).cuda().to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    x = torch.randn(2, C, HW, HW, device="cuda").to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    # eager pass (fresh grad tensor path)
    m.weight.grad = None
    m(x).float().sum().backward()
    eager = m.weight.grad.float().clone()
    # flat-view-style grad: pre-installed channels_last bf16 buffer
    gview = torch.zeros_like(m.weight)
    m.weight.grad = gview
    for _ in range(2):  # warmup (non-capturing: add path)
        gview.zero_()
        m(x).float().sum().backward()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g, capture_error_mode="thread_local"):
        gview.zero_()
        m(x).float().sum().backward()
    g.replay()
    g.replay()  # direct write: replays must NOT accumulate
    torch.cuda.synchronize()
    assert m.weight.grad.data_ptr() == gview.data_ptr()
    rel = ((gview.float() - eager).norm() / eager.norm()).item()
    assert rel < 1e-2, f"graph grad diverged: {rel}"
