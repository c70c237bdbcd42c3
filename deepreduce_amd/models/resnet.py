"""ResNet-20 (CIFAR) and ResNet-50 (ImageNet) — standard architectures,
written directly (torchvision is not available in this environment).

ResNet-20: He et al. CIFAR variant (3 stages x 3 BasicBlocks, 16/32/64
channels, ~0.27M params).  ResNet-50: Bottleneck v1.5 (stride in the 3x3),
~25.6M params.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, out_ch, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, 1, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride, bias=False), nn.BatchNorm2d(out_ch)
            )

    def forward(self, x):
        identity = x if self.down is None else self.down(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride, bias=False), nn.BatchNorm2d(out_ch)
            )

    def forward(self, x):
        identity = x if self.down is None else self.down(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class CifarResNet(nn.Module):
    def __init__(self, layers_per_stage=3, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(16)
        self.relu = nn.ReLU(inplace=True)
        stages = []
        in_ch = 16
        for i, out_ch in enumerate([16, 32, 64]):
            for j in range(layers_per_stage):
                stride = 2 if (i > 0 and j == 0) else 1
                stages.append(BasicBlock(in_ch, out_ch, stride))
                in_ch = out_ch
        self.stages = nn.Sequential(*stages)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(64, num_classes)

    def forward(self, x):
        x = self.relu(self.bn1(self.conv1(x)))
        x = self.stages(x)
        return self.fc(self.pool(x).flatten(1))


class ImageNetResNet(nn.Module):
    def __init__(self, blocks=(3, 4, 6, 3), num_classes=1000):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, 2, 3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        stages = []
        in_ch = 64
        for i, (width, n) in enumerate(zip([64, 128, 256, 512], blocks)):
            for j in range(n):
                stride = 2 if (i > 0 and j == 0) else 1
                stages.append(Bottleneck(in_ch, width, stride))
                in_ch = width * Bottleneck.expansion
        self.stages = nn.Sequential(*stages)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(2048, num_classes)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.stages(x)
        return self.fc(self.pool(x).flatten(1))


def resnet20(num_classes: int = 10) -> nn.Module:
    return CifarResNet(3, num_classes)


def resnet50(num_classes: int = 1000) -> nn.Module:
    return ImageNetResNet((3, 4, 6, 3), num_classes)
