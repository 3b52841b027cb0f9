"""DenseNet-BC for CIFAR (reference: model_ops/densenet.py:18-116;
the reference default is depth 190 / growth 40 — configurable here, with a
lighter default depth so CPU tests stay fast)."""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


class _Bottleneck(nn.Module):
    def __init__(self, in_planes, growth_rate):
        super().__init__()
        inter = 4 * growth_rate
        self.bn1 = nn.BatchNorm2d(in_planes)
        self.conv1 = nn.Conv2d(in_planes, inter, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(inter)
        self.conv2 = nn.Conv2d(inter, growth_rate, 3, padding=1, bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.bn1(x)))
        out = self.conv2(F.relu(self.bn2(out)))
        return torch.cat([x, out], 1)


class _Transition(nn.Module):
    def __init__(self, in_planes, out_planes):
        super().__init__()
        self.bn = nn.BatchNorm2d(in_planes)
        self.conv = nn.Conv2d(in_planes, out_planes, 1, bias=False)

    def forward(self, x):
        out = self.conv(F.relu(self.bn(x)))
        return F.avg_pool2d(out, 2)


class DenseNet(nn.Module):
    def __init__(self, depth=40, growth_rate=12, reduction=0.5, num_classes=10,
                 in_channels=3):
        super().__init__()
        n = (depth - 4) // 6  # bottleneck blocks per dense stage
        planes = 2 * growth_rate
        self.conv1 = nn.Conv2d(in_channels, planes, 3, padding=1, bias=False)
        stages = []
        for i in range(3):
            blocks = []
            for _ in range(n):
                blocks.append(_Bottleneck(planes, growth_rate))
                planes += growth_rate
            stages.append(nn.Sequential(*blocks))
            if i < 2:
                out_planes = int(math.floor(planes * reduction))
                stages.append(_Transition(planes, out_planes))
                planes = out_planes
        self.features = nn.Sequential(*stages)
        self.bn = nn.BatchNorm2d(planes)
        self.linear = nn.Linear(planes, num_classes)

    def forward(self, x):
        out = self.conv1(x)
        out = self.features(out)
        out = F.relu(self.bn(out))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.linear(out)
