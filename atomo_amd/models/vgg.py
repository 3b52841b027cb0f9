"""VGG for 32x32 inputs (reference: model_ops/vgg.py:15-108, cfg table)."""

import torch.nn as nn

_CFG = {
    "A": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "B": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "D": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M", 512, 512, 512, "M"],
    "E": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512, 512, "M",
          512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, cfg_key: str, batch_norm: bool, num_classes=10, in_channels=3):
        super().__init__()
        layers, c = [], in_channels
        for v in _CFG[cfg_key]:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers.append(nn.Conv2d(c, v, 3, padding=1, bias=not batch_norm))
                if batch_norm:
                    layers.append(nn.BatchNorm2d(v))
                layers.append(nn.ReLU(inplace=True))
                c = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            nn.Linear(512, 512),
            nn.ReLU(inplace=True),
            nn.Dropout(),
            nn.Linear(512, 512),
            nn.ReLU(inplace=True),
            nn.Dropout(),
            nn.Linear(512, num_classes),
        )

    def forward(self, x):
        x = self.features(x).flatten(1)
        return self.classifier(x)


def vgg11(num_classes=10, in_channels=3):
    return VGG("A", False, num_classes, in_channels)


def vgg11_bn(num_classes=10, in_channels=3):
    return VGG("A", True, num_classes, in_channels)


def vgg13(num_classes=10, in_channels=3):
    return VGG("B", False, num_classes, in_channels)


def vgg13_bn(num_classes=10, in_channels=3):
    return VGG("B", True, num_classes, in_channels)


def vgg16(num_classes=10, in_channels=3):
    return VGG("D", False, num_classes, in_channels)


def vgg16_bn(num_classes=10, in_channels=3):
    return VGG("D", True, num_classes, in_channels)


def vgg19(num_classes=10, in_channels=3):
    return VGG("E", False, num_classes, in_channels)


def vgg19_bn(num_classes=10, in_channels=3):
    return VGG("E", True, num_classes, in_channels)
