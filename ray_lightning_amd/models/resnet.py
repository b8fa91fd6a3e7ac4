"""ResNet-50 (bottleneck v1) — the benchmark flagship model
(BASELINE.json configs 2-3: samples/sec, synthetic ImageNet).

Own implementation (no torchvision in this environment). Convs go
through MIOpen via PyTorch-ROCm; normalization +
activation + residual are this framework's hand-written fused NHWC
CDNA4 kernel (``ops/fused_bn``) — BN + elementwise were >50% of the
MI355X step time when left to MIOpen/ATen
(profiles/r01_resnet50_1gpu_fixedfind.md).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from ..ops.fused_bn import FusedBatchNorm2d


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = FusedBatchNorm2d(width, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = FusedBatchNorm2d(width, relu=True)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = FusedBatchNorm2d(out_ch, relu=True)  # fused add+relu
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.bn3(self.conv3(out), identity)


class ResNet(nn.Module):
    def __init__(self, layers: List[int], num_classes: int = 1000,
                 zero_init_residual: bool = True):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBatchNorm2d(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):  # FusedBatchNorm2d too
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride,
                          bias=False),
                FusedBatchNorm2d(out_ch))
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(out_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet18_like(num_classes: int = 1000) -> ResNet:
    """Small bottleneck variant for quick tests."""
    return ResNet([1, 1, 1, 1], num_classes)
