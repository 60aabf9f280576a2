import torch

from mi355x_scale.models import resnet18, resnet50


def test_resnet18_shapes_and_params():
    m = resnet18(num_classes=1000)
    n_params = sum(p.numel() for p in m.parameters())
    assert n_params == 11_689_512  # canonical ResNet-18/1000 size
    x = torch.randn(2, 3, 224, 224)
    y = m(x)
    assert y.shape == (2, 1000)
    y.sum().backward()
    assert m.conv1.weight.grad is not None


def test_resnet50_params():
    m = resnet50(num_classes=1000)
    n_params = sum(p.numel() for p in m.parameters())
    assert n_params == 25_557_032  # canonical ResNet-50/1000 size


def test_resnet18_small_input():
    m = resnet18(num_classes=10)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)


def test_resnet34_params():
    from mi355x_scale.models import resnet34
    m = resnet34(num_classes=1000)
    n_params = sum(p.numel() for p in m.parameters())
    assert n_params == 21_797_672  # canonical ResNet-34/1000 size
