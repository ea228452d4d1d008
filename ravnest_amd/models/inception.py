"""Inception-V3 (CIFAR-10-adapted) for the multi-branch workload.

Workload parity: the reference's Inception example (models.py:96-393, a
CIFAR-adapted Inception3). The multi-branch blocks are what exercises the
planner's multi-consumer routing and the grad pass-through accumulation
(SURVEY.md section 7 hard parts: "test with Inception (multi-branch)").
Own compact implementation; convs on MIOpen through torch (library path).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import Conv1x1, FusedBatchNorm2d
from ..ops import (MaxPool2d as KMaxPool2d, AvgPool2d as KAvgPool2d,
                   AdaptiveAvgPool2d as KAdaptiveAvgPool2d)


class BasicConv2d(nn.Module):
    def __init__(self, in_ch, out_ch, **kw):
        super().__init__()
        if kw.get("kernel_size") == 1 and "stride" not in kw and \
                "padding" not in kw:
            self.conv = Conv1x1(in_ch, out_ch)
        else:
            self.conv = nn.Conv2d(in_ch, out_ch, bias=False, **kw)
        self.bn = FusedBatchNorm2d(out_ch, eps=0.001)

    def forward(self, x):
        return F.relu(self.bn(self.conv(x)), inplace=True)


class InceptionA(nn.Module):
    def __init__(self, in_ch, pool_features):
        super().__init__()
        self.branch1x1 = BasicConv2d(in_ch, 64, kernel_size=1)
        self.branch5x5_1 = BasicConv2d(in_ch, 48, kernel_size=1)
        self.branch5x5_2 = BasicConv2d(48, 64, kernel_size=5, padding=2)
        self.branch3x3dbl_1 = BasicConv2d(in_ch, 64, kernel_size=1)
        self.branch3x3dbl_2 = BasicConv2d(64, 96, kernel_size=3, padding=1)
        self.branch3x3dbl_3 = BasicConv2d(96, 96, kernel_size=3, padding=1)
        self.branch_pool = BasicConv2d(in_ch, pool_features, kernel_size=1)
        self.pool = KAvgPool2d(3, stride=1, padding=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b5 = self.branch5x5_2(self.branch5x5_1(x))
        b3 = self.branch3x3dbl_3(self.branch3x3dbl_2(self.branch3x3dbl_1(x)))
        bp = self.branch_pool(self.pool(x))
        return torch.cat([b1, b5, b3, bp], 1)


class InceptionB(nn.Module):
    def __init__(self, in_ch):
        super().__init__()
        self.branch3x3 = BasicConv2d(in_ch, 384, kernel_size=3, stride=2)
        self.branch3x3dbl_1 = BasicConv2d(in_ch, 64, kernel_size=1)
        self.branch3x3dbl_2 = BasicConv2d(64, 96, kernel_size=3, padding=1)
        self.branch3x3dbl_3 = BasicConv2d(96, 96, kernel_size=3, stride=2)
        self.pool = KMaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([self.branch3x3(x),
                          self.branch3x3dbl_3(self.branch3x3dbl_2(
                              self.branch3x3dbl_1(x))),
                          self.pool(x)], 1)


class InceptionC(nn.Module):
    def __init__(self, in_ch, ch7):
        super().__init__()
        self.branch1x1 = BasicConv2d(in_ch, 192, kernel_size=1)
        self.branch7x7_1 = BasicConv2d(in_ch, ch7, kernel_size=1)
        self.branch7x7_2 = BasicConv2d(ch7, ch7, kernel_size=(1, 7),
                                       padding=(0, 3))
        self.branch7x7_3 = BasicConv2d(ch7, 192, kernel_size=(7, 1),
                                       padding=(3, 0))
        self.branch7x7dbl_1 = BasicConv2d(in_ch, ch7, kernel_size=1)
        self.branch7x7dbl_2 = BasicConv2d(ch7, ch7, kernel_size=(7, 1),
                                          padding=(3, 0))
        self.branch7x7dbl_3 = BasicConv2d(ch7, ch7, kernel_size=(1, 7),
                                          padding=(0, 3))
        self.branch7x7dbl_4 = BasicConv2d(ch7, ch7, kernel_size=(7, 1),
                                          padding=(3, 0))
        self.branch7x7dbl_5 = BasicConv2d(ch7, 192, kernel_size=(1, 7),
                                          padding=(0, 3))
        self.branch_pool = BasicConv2d(in_ch, 192, kernel_size=1)
        self.pool = KAvgPool2d(3, stride=1, padding=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b7 = self.branch7x7_3(self.branch7x7_2(self.branch7x7_1(x)))
        b7d = self.branch7x7dbl_5(self.branch7x7dbl_4(self.branch7x7dbl_3(
            self.branch7x7dbl_2(self.branch7x7dbl_1(x)))))
        bp = self.branch_pool(self.pool(x))
        return torch.cat([b1, b7, b7d, bp], 1)


class InceptionD(nn.Module):
    def __init__(self, in_ch):
        super().__init__()
        self.branch3x3_1 = BasicConv2d(in_ch, 192, kernel_size=1)
        self.branch3x3_2 = BasicConv2d(192, 320, kernel_size=3, stride=2)
        self.branch7x7x3_1 = BasicConv2d(in_ch, 192, kernel_size=1)
        self.branch7x7x3_2 = BasicConv2d(192, 192, kernel_size=(1, 7),
                                         padding=(0, 3))
        self.branch7x7x3_3 = BasicConv2d(192, 192, kernel_size=(7, 1),
                                         padding=(3, 0))
        self.branch7x7x3_4 = BasicConv2d(192, 192, kernel_size=3, stride=2)
        self.pool = KMaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([
            self.branch3x3_2(self.branch3x3_1(x)),
            self.branch7x7x3_4(self.branch7x7x3_3(self.branch7x7x3_2(
                self.branch7x7x3_1(x)))),
            self.pool(x)], 1)


class InceptionE(nn.Module):
    def __init__(self, in_ch):
        super().__init__()
        self.branch1x1 = BasicConv2d(in_ch, 320, kernel_size=1)
        self.branch3x3_1 = BasicConv2d(in_ch, 384, kernel_size=1)
        self.branch3x3_2a = BasicConv2d(384, 384, kernel_size=(1, 3),
                                        padding=(0, 1))
        self.branch3x3_2b = BasicConv2d(384, 384, kernel_size=(3, 1),
                                        padding=(1, 0))
        self.branch3x3dbl_1 = BasicConv2d(in_ch, 448, kernel_size=1)
        self.branch3x3dbl_2 = BasicConv2d(448, 384, kernel_size=3, padding=1)
        self.branch3x3dbl_3a = BasicConv2d(384, 384, kernel_size=(1, 3),
                                           padding=(0, 1))
        self.branch3x3dbl_3b = BasicConv2d(384, 384, kernel_size=(3, 1),
                                           padding=(1, 0))
        self.branch_pool = BasicConv2d(in_ch, 192, kernel_size=1)
        self.pool = KAvgPool2d(3, stride=1, padding=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b3 = self.branch3x3_1(x)
        b3 = torch.cat([self.branch3x3_2a(b3), self.branch3x3_2b(b3)], 1)
        bd = self.branch3x3dbl_2(self.branch3x3dbl_1(x))
        bd = torch.cat([self.branch3x3dbl_3a(bd), self.branch3x3dbl_3b(bd)], 1)
        bp = self.branch_pool(self.pool(x))
        return torch.cat([b1, b3, bd, bp], 1)


class Inception3(nn.Module):
    """CIFAR-sized Inception-V3 (32x32 inputs, stride-1 stem — parity
    with the reference's huyvnphan-derived variant)."""

    def __init__(self, num_classes=10, in_ch=3):
        super().__init__()
        self.stem = nn.Sequential(
            BasicConv2d(in_ch, 32, kernel_size=3, padding=1),
            BasicConv2d(32, 32, kernel_size=3, padding=1),
            BasicConv2d(32, 64, kernel_size=3, padding=1))
        self.mixed5b = InceptionA(64, 32)
        self.mixed5c = InceptionA(256, 64)
        self.mixed5d = InceptionA(288, 64)
        self.mixed6a = InceptionB(288)
        self.mixed6b = InceptionC(768, 128)
        self.mixed6c = InceptionC(768, 160)
        self.mixed6d = InceptionC(768, 160)
        self.mixed6e = InceptionC(768, 192)
        self.mixed7a = InceptionD(768)
        self.mixed7b = InceptionE(1280)
        self.mixed7c = InceptionE(2048)
        self.avgpool = KAdaptiveAvgPool2d((1, 1))
        self.flatten = nn.Flatten()
        self.dropout = nn.Dropout(0.5)
        self.fc = nn.Linear(2048, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.mixed5d(self.mixed5c(self.mixed5b(x)))
        x = self.mixed6e(self.mixed6d(self.mixed6c(self.mixed6b(
            self.mixed6a(x)))))
        x = self.mixed7c(self.mixed7b(self.mixed7a(x)))
        x = self.dropout(self.flatten(self.avgpool(x)))
        return self.fc(x)
