"""ResNet family, CIFAR variant (3x3 stem, no maxpool), as in the reference
model zoo (ref: src/model_ops/resnet.py:14-113 — ResNet-18 = 62 param
tensors / 11,173,962 params on CIFAR-10).

MI355X-native structure: every BN is a PsBatchNorm2d with its ReLU (and the
block's residual add) fused into the BN epilogue — on GPU these run as
hand-written NHWC CDNA4 kernels (ops/kernels/batchnorm.hip); on CPU they
fall back to torch ops with identical math.

For ImageNet-shaped inputs (224x224, num_classes >= 200) the stem switches
to 7x7/stride-2 + maxpool so ResNet-50 on synthetic 3x224x224 (BASELINE
config 5) has the standard compute shape.
"""
from __future__ import annotations

import torch.nn as nn

from ..ops.conv import PsConv2d, conv_with_passthrough
from ..ops.linear import PsLinear
from ..ops.pool import max_pool2d as ps_max_pool2d, global_avg_pool
from ..ops.modules import PsBatchNorm2d


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = PsConv2d(in_planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = PsBatchNorm2d(planes, relu=True)
        self.conv2 = PsConv2d(planes, planes, 3, stride=1, padding=1, bias=False)
        self.bn2 = PsBatchNorm2d(planes, relu=True)   # fused: relu(bn + shortcut)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes * self.expansion:
            self.shortcut = nn.Sequential(
                PsConv2d(in_planes, planes * self.expansion, 1, stride=stride, bias=False),
                PsBatchNorm2d(planes * self.expansion),
            )

    def forward(self, x):
        # conv1 passes x through so the residual branch's gradient fuses
        # into conv1's dgrad epilogue (ops/conv.py _ConvCarryFn) instead of
        # an autograd elementwise add at the fork
        h, x_sc = conv_with_passthrough(self.conv1, x)
        out = self.bn1(h)
        return self.bn2(self.conv2(out), residual=self.shortcut(x_sc))


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = PsConv2d(in_planes, planes, 1, bias=False)
        self.bn1 = PsBatchNorm2d(planes, relu=True)
        self.conv2 = PsConv2d(planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = PsBatchNorm2d(planes, relu=True)
        self.conv3 = PsConv2d(planes, planes * self.expansion, 1, bias=False)
        self.bn3 = PsBatchNorm2d(planes * self.expansion, relu=True)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes * self.expansion:
            self.shortcut = nn.Sequential(
                PsConv2d(in_planes, planes * self.expansion, 1, stride=stride, bias=False),
                PsBatchNorm2d(planes * self.expansion),
            )

    def forward(self, x):
        h, x_sc = conv_with_passthrough(self.conv1, x)
        out = self.bn1(h)
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=self.shortcut(x_sc))


class ResNet(nn.Module):
    def __init__(self, block, num_blocks, num_classes: int = 10,
                 in_channels: int = 3, imagenet_stem: bool = False):
        super().__init__()
        self.in_planes = 64
        self.imagenet_stem = imagenet_stem
        if imagenet_stem:
            self.conv1 = PsConv2d(in_channels, 64, 7, stride=2, padding=3, bias=False)
        else:
            self.conv1 = PsConv2d(in_channels, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = PsBatchNorm2d(64, relu=True)
        self.layer1 = self._make_layer(block, 64, num_blocks[0], stride=1)
        self.layer2 = self._make_layer(block, 128, num_blocks[1], stride=2)
        self.layer3 = self._make_layer(block, 256, num_blocks[2], stride=2)
        self.layer4 = self._make_layer(block, 512, num_blocks[3], stride=2)
        self.linear = PsLinear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, n, stride):
        strides = [stride] + [1] * (n - 1)
        layers = []
        for s in strides:
            layers.append(block(self.in_planes, planes, s))
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        if self.imagenet_stem:
            out = ps_max_pool2d(out, 3, stride=2, padding=1)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = global_avg_pool(out)
        return self.linear(out)


def _imagenet(num_classes: int) -> bool:
    # 1000-way default => ImageNet-shaped stem (BASELINE config 5)
    return num_classes >= 200


def ResNet18(num_classes=10, in_channels=3):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, in_channels, _imagenet(num_classes))


def ResNet34(num_classes=10, in_channels=3):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, in_channels, _imagenet(num_classes))


def ResNet50(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, in_channels, _imagenet(num_classes))


def ResNet101(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, in_channels, _imagenet(num_classes))


def ResNet152(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, in_channels, _imagenet(num_classes))
