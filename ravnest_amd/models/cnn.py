"""Small 2-conv CNN for the 8x8-digits walkthrough workload.

Workload parity with the reference's CNN_Net (models.py:3-44): two conv
blocks (conv-relu-dropout-bn, conv-relu-pool-dropout-bn) then two linear
layers to 10 classes — the canonical 3-stage pipeline used by the
walkthrough and the CPU plumbing config of BASELINE.json.
"""
import torch
import torch.nn as nn

from ..ops import MaxPool2d as KMaxPool2d
from ..ops import Softmax as KSoftmax


class CNN(nn.Module):
    def __init__(self, in_channels: int = 1, n_classes: int = 10,
                 dropout: float = 0.25):
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 16, 3, stride=1, padding=1)
        self.act1 = nn.ReLU()
        self.drop1 = nn.Dropout2d(dropout)
        self.bn1 = nn.BatchNorm2d(16)
        self.conv2 = nn.Conv2d(16, 32, 3, stride=1, padding=1)
        self.act2 = nn.ReLU()
        self.pool2 = KMaxPool2d(2)
        self.drop2 = nn.Dropout2d(dropout)
        self.bn2 = nn.BatchNorm2d(32)
        self.flatten = nn.Flatten()
        self.fc1 = nn.Linear(32 * 4 * 4, 256)
        self.act3 = nn.ReLU()
        self.drop3 = nn.Dropout(dropout * 2)
        self.bn3 = nn.BatchNorm1d(256)
        self.fc2 = nn.Linear(256, n_classes)
        self.softmax = KSoftmax(-1)

    def forward(self, x):
        x = self.bn1(self.drop1(self.act1(self.conv1(x))))
        x = self.bn2(self.drop2(self.pool2(self.act2(self.conv2(x)))))
        x = self.flatten(x)
        x = self.bn3(self.drop3(self.act3(self.fc1(x))))
        return self.softmax(self.fc2(x))
