"""ResNet family (ResNet-50 default) for the image workloads.

Workload parity: the reference's ResNet-50 example
(examples/resnet50/provider.py: torchvision resnet50(num_classes=200) on
TinyImageNet, SGD+momentum) — re-implemented natively (no torchvision in
the image). Convolutions run on MIOpen through torch (library path, like
the reference's stock-op usage, SURVEY.md section 2.1 note); hand-written
CDNA4 implicit-GEMM conv kernels are the tracked follow-up for the conv
hot path.

fx-traceable; splits cleanly into pipeline stages.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import (AdaptiveAvgPool2d, Conv1x1, FusedBatchNorm2d,
                   MaxPool2d)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1, downsample=None):
        super().__init__()
        self.conv1 = Conv1x1(in_ch, ch)
        self.bn1 = FusedBatchNorm2d(ch)
        self.conv2 = nn.Conv2d(ch, ch, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = FusedBatchNorm2d(ch)
        self.conv3 = Conv1x1(ch, ch * self.expansion)
        self.bn3 = FusedBatchNorm2d(ch * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers=(3, 4, 6, 3), num_classes=200, in_ch=3):
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(in_ch, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = AdaptiveAvgPool2d((1, 1))
        self.flatten = nn.Flatten()
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, FusedBatchNorm2d)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, ch, blocks, stride=1):
        downsample = None
        if stride != 1 or self.in_planes != ch * Bottleneck.expansion:
            downsample = nn.Sequential(
                Conv1x1(self.in_planes, ch * Bottleneck.expansion,
                        stride=stride),
                FusedBatchNorm2d(ch * Bottleneck.expansion))
        layers = [Bottleneck(self.in_planes, ch, stride, downsample)]
        self.in_planes = ch * Bottleneck.expansion
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_planes, ch))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.flatten(self.avgpool(x))
        return self.fc(x)


def resnet50(num_classes=200, in_ch=3):
    return ResNet((3, 4, 6, 3), num_classes, in_ch)


def resnet18_ish(num_classes=10, in_ch=3):
    """Small variant for CPU tests."""
    return ResNet((1, 1, 1, 1), num_classes, in_ch)
