"""MobileNetV2 — the paper's federated CIFAR-10 backbone (deepreduce
.nips21.pdf p.33 Table 5: MobileNet/CIFAR-10 FedAvg experiments).
Standard inverted-residual architecture, CIFAR variant (stride-1 stem)
by default, random-init for synthetic benchmarks.
"""
from __future__ import annotations

import torch.nn as nn


class _InvertedResidual(nn.Module):
    def __init__(self, c_in, c_out, stride, expand):
        super().__init__()
        hidden = c_in * expand
        self.use_res = stride == 1 and c_in == c_out
        layers = []
        if expand != 1:
            layers += [nn.Conv2d(c_in, hidden, 1, bias=False),
                       nn.BatchNorm2d(hidden), nn.ReLU6(inplace=True)]
        layers += [
            nn.Conv2d(hidden, hidden, 3, stride, 1, groups=hidden, bias=False),
            nn.BatchNorm2d(hidden), nn.ReLU6(inplace=True),
            nn.Conv2d(hidden, c_out, 1, bias=False), nn.BatchNorm2d(c_out),
        ]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    # (expand, c_out, repeats, stride)
    cfg = [(1, 16, 1, 1), (6, 24, 2, 1), (6, 32, 3, 2), (6, 64, 4, 2),
           (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self, num_classes: int = 10, in_res: int = 32):
        super().__init__()
        stem_stride = 1 if in_res <= 64 else 2
        c = 32
        features = [nn.Conv2d(3, c, 3, stem_stride, 1, bias=False),
                    nn.BatchNorm2d(c), nn.ReLU6(inplace=True)]
        for expand, c_out, reps, stride in self.cfg:
            for i in range(reps):
                features.append(_InvertedResidual(c, c_out, stride if i == 0 else 1,
                                                  expand))
                c = c_out
        features += [nn.Conv2d(c, 1280, 1, bias=False), nn.BatchNorm2d(1280),
                     nn.ReLU6(inplace=True), nn.AdaptiveAvgPool2d(1)]
        self.features = nn.Sequential(*features)
        self.classifier = nn.Linear(1280, num_classes)

    def forward(self, x):
        return self.classifier(self.features(x).flatten(1))


def mobilenet_v2(num_classes: int = 10, in_res: int = 32) -> MobileNetV2:
    return MobileNetV2(num_classes, in_res)
