"""ResNet family, implemented from scratch for PyTorch-ROCm.

The reference trains ``torchvision.models.resnet50``
(``deep_learning/2.distributed-data-loading-petastorm.py:150``); the
benchmark model named by BASELINE.json is ResNet-18. torchvision is not a
dependency here — the architectures are defined directly. Convs run
through MIOpen via PyTorch-ROCm in channels-last memory format (MIOpen's
fast conv layout on CDNA); every BatchNorm + residual-add + ReLU group is
a single ``FusedBNReLU2d`` (hand-written bf16 NHWC HIP kernels,
ops/csrc/fused_bn.hip) instead of the MIOpen fp32 4-kernel BN path plus
standalone elementwise passes.
"""
from __future__ import annotations

from typing import List, Optional, Type, Union

import torch
import torch.nn as nn

from ..ops.conv3x3 import Conv3x3
from ..ops.fused_bn import FusedBNReLU2d
from ..ops.maxpool import MaxPool3x3s2
from ..ops.stemconv import StemConv2d


def _conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    if stride == 1 and cin == cout:
        # MFMA weight-grad fast path (ops/csrc/conv3x3wrw.hip)
        return Conv3x3(cin, cout)
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def _conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn1 = FusedBNReLU2d(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.bn2 = FusedBNReLU2d(cout)  # fuses the residual add + relu
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = _conv1x1(cin, cout)
        self.bn1 = FusedBNReLU2d(cout)
        self.conv2 = _conv3x3(cout, cout, stride)
        self.bn2 = FusedBNReLU2d(cout)
        self.conv3 = _conv1x1(cout, cout * self.expansion)
        self.bn3 = FusedBNReLU2d(cout * self.expansion)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=identity)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int = 1000,
                 zero_init_residual: bool = True):
        super().__init__()
        self.inplanes = 64
        self.conv1 = StemConv2d()  # hand-written MFMA 7x7 s2 (ops/csrc/stemconv.hip)
        self.bn1 = FusedBNReLU2d(64)  # stem BN+ReLU in one pass
        self.maxpool = MaxPool3x3s2()  # u8-code NHWC pool kernel on GPU
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)
                elif isinstance(m, BasicBlock):
                    nn.init.zeros_(m.bn2.weight)

    def _make_layer(self, block, planes: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                _conv1x1(self.inplanes, planes * block.expansion, stride),
                FusedBNReLU2d(planes * block.expansion, relu=False),
            )
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        layers += [block(self.inplanes, planes) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # stem: bn1's relu+affine fold INTO the pool window reads (the
        # normalized map is never materialized — ops/fused_bn.py
        # forward_pooled); composed ops off the GPU bf16 path
        x = self.bn1.forward_pooled(self.conv1(x))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(num_classes: int = 1000) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes)


def resnet34(num_classes: int = 1000) -> ResNet:
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes)


MODELS = {"resnet18": resnet18, "resnet34": resnet34, "resnet50": resnet50}


def build_model(name: str, num_classes: int = 1000) -> ResNet:
    try:
        return MODELS[name](num_classes)
    except KeyError:
        raise ValueError(f"unknown model {name!r}; have {sorted(MODELS)}")
