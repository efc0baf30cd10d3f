"""LeNet for MNIST-shaped inputs.

Same architecture/parameter-count surface as the reference's LeNet
(ref: src/model_ops/lenet.py:16-37 — 8 param tensors, 431,080 params for
1x28x28 inputs): conv(1->20,k5) -> pool -> conv(20->50,k5) -> pool ->
fc(800->500) -> fc(500->nc).

There is no hand-unrolled "split" variant here: the reference's LeNetSplit
(per-layer Isend interleaved with backward, src/model_ops/lenet.py:39-258)
is realized framework-wide by bucketed gradient hooks pushing RCCL ops on a
side HIP stream (ps_pytorch_amd/parallel/worker.py) — every model gets the
comm/compute overlap without a hand-written backward.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.conv import PsConv2d
from ..ops.linear import PsLinear
from ..ops.pool import max_pool2d as ps_max_pool2d
import torch.nn.functional as F


class LeNet(nn.Module):
    def __init__(self, num_classes: int = 10, in_channels: int = 1):
        super().__init__()
        self.conv1 = PsConv2d(in_channels, 20, kernel_size=5)
        # keep the conv1 -> relu -> pool -> conv2 pipeline 8-aligned on the
        # kernel path: conv1 emits 24 channels (zero pad in weight space),
        # conv2 absorbs the pre-padded input (see PsConv2d.out_pad). conv2
        # itself stays at 50 — its output flattens into fc1's 800 features,
        # where pad channels would shift feature positions.
        self.conv1.out_pad = 24
        self.conv2 = PsConv2d(20, 50, kernel_size=5)
        # 28x28 -> conv5 -> 24 -> pool -> 12 -> conv5 -> 8 -> pool -> 4
        self.fc1 = PsLinear(50 * 4 * 4, 500)
        self.fc2 = PsLinear(500, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # conv -> pool -> relu -> conv -> pool -> relu -> fc -> fc, exactly
        # the reference's op order (src/model_ops/lenet.py:24-35 — relu AFTER
        # pool, and no activation between fc1 and fc2). pool-then-relu is
        # also 4x cheaper on the relu: max commutes with monotonic relu.
        x = F.relu(ps_max_pool2d(self.conv1(x), 2))
        x = F.relu(ps_max_pool2d(self.conv2(x), 2))
        x = x.flatten(1)
        x = self.fc1(x)
        return self.fc2(x)
