"""Model zoo + factory (reference parity: src/util.py:8-19, src/model_ops/)."""
from __future__ import annotations

import torch.nn as nn

from .lenet import LeNet
from .resnet import ResNet18, ResNet34, ResNet50, ResNet101, ResNet152
from .vgg import VGG11, VGG11_BN, VGG13, VGG13_BN, VGG16, VGG16_BN, VGG19, VGG19_BN

_FACTORY = {
    'lenet': LeNet,
    'resnet18': ResNet18,
    'resnet34': ResNet34,
    'resnet50': ResNet50,
    'resnet101': ResNet101,
    'resnet152': ResNet152,
    # ref util.py:19 maps "VGG11" to vgg11_bn — keep that behavior;
    # the plain variants stay reachable as VGG11_plain etc.
    'vgg11': VGG11_BN,
    'vgg11_plain': VGG11,
    'vgg11_bn': VGG11_BN,
    'vgg13_plain': VGG13,
    'vgg13': VGG13_BN,
    'vgg13_bn': VGG13_BN,
    'vgg16_plain': VGG16,
    'vgg16': VGG16_BN,
    'vgg16_bn': VGG16_BN,
    'vgg19_plain': VGG19,
    'vgg19': VGG19_BN,
    'vgg19_bn': VGG19_BN,
}


def build_model(name: str, num_classes: int = 10, in_channels: int = 3) -> nn.Module:
    """Build a model by CLI name (ref factory: src/util.py:8-19).

    The reference's `--network=ResNet` meant ResNet-18; accept that alias.
    LeNet forces in_channels as given (MNIST=1)."""
    key = name.lower()
    if key == 'resnet':
        key = 'resnet18'
    if key == 'vgg':
        key = 'vgg11'
    if key not in _FACTORY:
        raise ValueError(f"unknown network {name!r}; have {sorted(_FACTORY)}")
    if key == 'lenet':
        return LeNet(num_classes=num_classes, in_channels=in_channels)
    return _FACTORY[key](num_classes=num_classes, in_channels=in_channels)


__all__ = ['build_model', 'LeNet', 'ResNet18', 'ResNet34', 'ResNet50',
           'ResNet101', 'ResNet152', 'VGG11', 'VGG13', 'VGG16', 'VGG19',
           'VGG11_BN', 'VGG13_BN', 'VGG16_BN', 'VGG19_BN']
