"""VGG family for CIFAR-shaped inputs (ref: src/model_ops/vgg.py:15-68).
BN variants use the fused PsBatchNorm2d (BN+ReLU in one NHWC kernel pass)."""
from __future__ import annotations

import torch.nn as nn

from ..ops.conv import PsConv2d
from ..ops.linear import PsLinear
from ..ops.modules import PsBatchNorm2d
from ..ops.pool import max_pool2d as ps_max_pool2d


class PsMaxPool2d(nn.Module):
    """nn.MaxPool2d(2,2) on the in-tree NHWC kernels (torch fallback on
    CPU); a module so make_layers stays an nn.Sequential."""

    def __init__(self, kernel_size: int, stride: int):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride

    def forward(self, x):
        return ps_max_pool2d(x, self.kernel_size, self.stride)

_CFG = {
    'VGG11': [64, 'M', 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512, 'M'],
    'VGG13': [64, 64, 'M', 128, 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512, 'M'],
    'VGG16': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 'M', 512, 512, 512, 'M',
              512, 512, 512, 'M'],
    'VGG19': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 256, 'M', 512, 512, 512, 512,
              'M', 512, 512, 512, 512, 'M'],
}


def _make_layers(cfg, in_channels: int, batch_norm: bool) -> nn.Sequential:
    layers = []
    c = in_channels
    for v in cfg:
        if v == 'M':
            layers.append(PsMaxPool2d(2, 2))
        else:
            layers.append(PsConv2d(c, v, 3, padding=1, bias=not batch_norm))
            if batch_norm:
                layers.append(PsBatchNorm2d(v, relu=True))   # fused BN+ReLU
            else:
                layers.append(nn.ReLU(inplace=True))
            c = v
    layers.append(nn.AdaptiveAvgPool2d(1))
    return nn.Sequential(*layers)


class VGG(nn.Module):
    def __init__(self, name: str, num_classes: int = 10, in_channels: int = 3,
                 batch_norm: bool = False):
        super().__init__()
        self.features = _make_layers(_CFG[name], in_channels, batch_norm)
        self.classifier = PsLinear(512, num_classes)

    def forward(self, x):
        x = self.features(x).flatten(1)
        return self.classifier(x)


def VGG11(num_classes=10, in_channels=3):
    return VGG('VGG11', num_classes, in_channels, False)


def VGG13(num_classes=10, in_channels=3):
    return VGG('VGG13', num_classes, in_channels, False)


def VGG16(num_classes=10, in_channels=3):
    return VGG('VGG16', num_classes, in_channels, False)


def VGG19(num_classes=10, in_channels=3):
    return VGG('VGG19', num_classes, in_channels, False)


def VGG11_BN(num_classes=10, in_channels=3):
    return VGG('VGG11', num_classes, in_channels, True)


def VGG13_BN(num_classes=10, in_channels=3):
    return VGG('VGG13', num_classes, in_channels, True)


def VGG16_BN(num_classes=10, in_channels=3):
    return VGG('VGG16', num_classes, in_channels, True)


def VGG19_BN(num_classes=10, in_channels=3):
    return VGG('VGG19', num_classes, in_channels, True)
