"""Numerics for the fused NHWC bf16 BatchNorm(+residual)(+ReLU) HIP kernels.

GPU tests compare the HIP path against the plain PyTorch fp32 composition
(F.batch_norm + add + relu) run on the same data — the fallback branch of
``FusedBNReLU2d`` itself. CPU tests pin the fallback's semantics against
``nn.BatchNorm2d`` so the oracle is anchored to stock PyTorch.
"""
import pytest
import torch
import torch.nn.functional as F

from mi355x_scale.ops.fused_bn import FusedBNReLU2d


def _ref_forward(x32, res32, w, b, rm, rv, training, momentum, eps, relu):
    y = F.batch_norm(x32, rm, rv, w, b, training, momentum, eps)
    if res32 is not None:
        y = y + res32
    return F.relu(y) if relu else y


def test_cpu_fallback_matches_batchnorm2d():
    torch.manual_seed(0)
    m = FusedBNReLU2d(16, relu=False)
    ref = torch.nn.BatchNorm2d(16)
    ref.load_state_dict(
        {k: v for k, v in m.state_dict().items()}, strict=True)
    x = torch.randn(4, 16, 8, 8)
    torch.testing.assert_close(m(x), ref(x))
    torch.testing.assert_close(m.running_mean, ref.running_mean)
    torch.testing.assert_close(m.running_var, ref.running_var)


def test_cpu_residual_relu_composition():
    torch.manual_seed(1)
    m = FusedBNReLU2d(8, relu=True)
    x = torch.randn(2, 8, 4, 4)
    r = torch.randn(2, 8, 4, 4)
    y = m(x, residual=r)
    ref = torch.nn.BatchNorm2d(8)
    want = F.relu(ref(x) + r)
    torch.testing.assert_close(y, want)


def test_no_plain_batchnorm_in_models():
    """Every BN in the model family must be the fused module — a plain
    nn.BatchNorm2d silently routes to MIOpen on GPU (this caught the
    stem/downsample BNs left unconverted in round 1)."""
    from mi355x_scale.models import resnet18, resnet50
    for model in (resnet18(num_classes=10), resnet50(num_classes=10)):
        leftovers = [n for n, m in model.named_modules()
                     if isinstance(m, torch.nn.BatchNorm2d)]
        assert not leftovers, leftovers
        fused = [m for m in model.modules()
                 if isinstance(m, FusedBNReLU2d)]
        assert len(fused) >= 20


def test_state_dict_interop_with_batchnorm2d():
    m = FusedBNReLU2d(32)
    bn = torch.nn.BatchNorm2d(32)
    m.load_state_dict(bn.state_dict())  # both directions must round-trip
    bn.load_state_dict(m.state_dict())


@pytest.mark.gpu
@pytest.mark.parametrize("C,hw", [(64, 56), (128, 28), (512, 7),
                                  (2048, 4)])
@pytest.mark.parametrize("relu,use_res", [(True, False), (False, False),
                                          (True, True)])
def test_gpu_fused_fwd_bwd_vs_fp32(C, hw, relu, use_res):
    torch.manual_seed(42)
    dev = torch.device("cuda:0")
    N = 9  # deliberately not a power of two
    x32 = torch.randn(N, C, hw, hw, device=dev)
    r32 = torch.randn(N, C, hw, hw, device=dev) if use_res else None

    m = FusedBNReLU2d(C, relu=relu).to(dev)
    with torch.no_grad():
        m.weight.mul_(0).add_(torch.rand(C, device=dev) + 0.5)
        m.bias.add_(torch.randn(C, device=dev) * 0.1)

    # HIP path: bf16 channels_last inputs
    xh = (x32.bfloat16().contiguous(memory_format=torch.channels_last)
          .requires_grad_(True))
    rh = (r32.bfloat16().contiguous(memory_format=torch.channels_last)
          .requires_grad_(True)) if use_res else None
    y = m(xh, residual=rh) if use_res else m(xh)
    assert y.dtype == torch.bfloat16
    gseed = torch.randn(y.shape, device=dev)
    y.backward(gseed.bfloat16())

    # fp32 reference on the SAME bf16-rounded data
    rm = torch.zeros(C, device=dev)
    rv = torch.ones(C, device=dev)
    xr = xh.detach().float().requires_grad_(True)
    rr = rh.detach().float().requires_grad_(True) if use_res else None
    w = m.weight.detach().clone().requires_grad_(True)
    b = m.bias.detach().clone().requires_grad_(True)
    yr = _ref_forward(xr, rr, w, b, rm, rv, True, 0.1, m.eps, relu)
    yr.backward(gseed.bfloat16().float())

    tol = dict(atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(y.float(), yr, **tol)
    torch.testing.assert_close(xh.grad.float(), xr.grad, **tol)
    if use_res:
        torch.testing.assert_close(rh.grad.float(), rr.grad, **tol)
    # channel-reduced param grads: fp32 accumulators, tighter relative tol
    torch.testing.assert_close(m.weight.grad, w.grad, atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(m.bias.grad, b.grad, atol=1e-2, rtol=1e-2)
    # running stats updated on-device by the finalize kernel
    torch.testing.assert_close(m.running_mean, rm, atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(m.running_var, rv, atol=1e-2, rtol=1e-2)


@pytest.mark.gpu
def test_gpu_eval_mode_uses_running_stats():
    torch.manual_seed(3)
    dev = torch.device("cuda:0")
    m = FusedBNReLU2d(64, relu=True).to(dev)
    with torch.no_grad():
        m.running_mean.add_(torch.randn(64, device=dev) * 0.3)
        m.running_var.mul_(torch.rand(64, device=dev) + 0.5)
    m.eval()
    x32 = torch.randn(5, 64, 14, 14, device=dev)
    xh = x32.bfloat16().contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        y = m(xh)
    want = _ref_forward(xh.float(), None, m.weight, m.bias,
                        m.running_mean.clone(), m.running_var.clone(),
                        False, 0.1, m.eps, True)
    torch.testing.assert_close(y.float(), want, atol=3e-2, rtol=3e-2)


@pytest.mark.gpu
def test_gpu_resnet50_bottleneck_step():
    """Bottleneck blocks (reference's actual model family,
    deep_learning/2...py:150) through the fused BN path: one fwd+bwd,
    zero fallbacks, finite loss."""
    from mi355x_scale.models import resnet50
    dev = torch.device("cuda:0")
    m = resnet50(num_classes=100).to(dev).to(
        memory_format=torch.channels_last)
    x = torch.randn(4, 3, 64, 64, device=dev)
    FusedBNReLU2d.gpu_fallbacks.clear()
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss = m(x).float().logsumexp(1).mean()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    assert not FusedBNReLU2d.gpu_fallbacks, \
        FusedBNReLU2d.gpu_fallbacks[:8]


@pytest.mark.gpu
def test_gpu_resnet18_step_runs_fused():
    """One fwd+bwd of the flagship model must route BN through the HIP
    extension (no silent MIOpen fallback): count num_batches_tracked."""
    from mi355x_scale.models import resnet18
    dev = torch.device("cuda:0")
    m = resnet18(num_classes=1000).to(dev).to(
        memory_format=torch.channels_last)
    x = torch.randint(0, 255, (4, 3, 64, 64), device=dev).float()
    FusedBNReLU2d.gpu_fallbacks.clear()
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        y = m(x)
        loss = y.float().logsumexp(1).mean()
    loss.backward()
    assert m.bn1.num_batches_tracked.item() == 1
    assert m.conv1.weight.grad is not None
    assert torch.isfinite(loss)
    assert not FusedBNReLU2d.gpu_fallbacks, (
        f"BN layers silently fell back to MIOpen on GPU: "
        f"{FusedBNReLU2d.gpu_fallbacks[:8]}")


@pytest.mark.gpu
def test_bn_pool_fused_matches_composed():
    """Stem fusion (forward_pooled): maxpool(relu(bn(x))) without
    materializing the normalized map — fwd/bwd parity vs the fp32 torch
    composition."""
    import torch.nn.functional as F
    from mi355x_scale.ops.fused_bn import FusedBNReLU2d

    torch.manual_seed(0)
    dev = "cuda"
    n, c, h, w = 8, 64, 32, 32
    xf = torch.randn(n, c, h, w, device=dev)

    # fp32-math reference on the SAME bf16-rounded inputs/grads (a pure
    # fp32 input would flip argmax codes at near-ties and scatter the
    # pooled grad to different pixels than any bf16 pipeline can)
    xb0 = xf.to(torch.bfloat16)
    bn_ref = torch.nn.BatchNorm2d(c).to(dev)
    xr = xb0.float().requires_grad_(True)
    out_ref = F.max_pool2d(F.relu(bn_ref(xr)), 3, stride=2, padding=1)
    g16 = torch.randn_like(out_ref).to(torch.bfloat16)
    out_ref.backward(g16.float())

    m = FusedBNReLU2d(c).to(dev)
    xb = xb0.to(memory_format=torch.channels_last).requires_grad_(True)
    out = m.forward_pooled(xb)
    out.backward(g16.to(memory_format=torch.channels_last))
    torch.cuda.synchronize()

    rel = ((out.float() - out_ref).norm() / out_ref.norm()).item()
    assert rel < 2e-2, f"fwd rel {rel}"
    relx = ((xb.grad.float() - xr.grad).norm() / xr.grad.norm()).item()
    assert relx < 3e-2, f"dx rel {relx}"
    relw = ((m.weight.grad - bn_ref.weight.grad).norm()
            / bn_ref.weight.grad.norm()).item()
    assert relw < 3e-2, f"dweight rel {relw}"
    relb = ((m.bias.grad - bn_ref.bias.grad).norm()
            / bn_ref.bias.grad.norm()).item()
    assert relb < 3e-2, f"dbias rel {relb}"
    # running stats updated like the reference
    assert torch.allclose(m.running_mean, bn_ref.running_mean, atol=2e-2)
    assert torch.allclose(m.running_var, bn_ref.running_var, atol=2e-2)
    # eval path uses running stats
    m.eval()
    bn_ref.eval()
    with torch.no_grad():
        oe = m.forward_pooled(xb.detach())
        oer = F.max_pool2d(F.relu(bn_ref(xf)), 3, stride=2, padding=1)
    rel_e = ((oe.float() - oer).norm() / oer.norm()).item()
    assert rel_e < 2e-2, f"eval rel {rel_e}"
