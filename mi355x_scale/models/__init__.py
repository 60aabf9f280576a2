"""Model zoo (from-scratch, PyTorch-ROCm)."""

from .resnet import ResNet, resnet18, resnet34, resnet50, build_model  # noqa: F401
