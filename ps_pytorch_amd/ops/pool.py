"""NHWC pooling ops on the in-tree CDNA4 kernels (ops/kernels/pool.hip):
max_pool2d with a deterministic argmax-gather backward, and global average
pool (the adaptive_avg_pool2d(x, 1) every ResNet/VGG head uses).

Replaces at::native pooling on the GPU hot path (ref reaches these through
F.max_pool2d, src/model_ops/lenet.py:31-32, and avg_pool2d, resnet.py:97);
CPU and unsupported shapes fall back to torch (the numerics reference)."""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import require_lib, current_stream_ptr

_CL = torch.channels_last
_ENABLED = os.environ.get('PS_POOL', '1') != '0'


def _supported(x: torch.Tensor) -> bool:
    return (_ENABLED and x.is_cuda and x.dim() == 4
            and x.dtype == torch.bfloat16)


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, stride, pad):
        lib = require_lib()
        x = x.contiguous(memory_format=_CL)
        Nb, C, H, W = x.shape
        P = (H + 2 * pad - k) // stride + 1
        Q = (W + 2 * pad - k) // stride + 1
        y = torch.empty((Nb, C, P, Q), dtype=x.dtype, device=x.device,
                        memory_format=_CL)
        arg = torch.empty(Nb * P * Q * C, dtype=torch.uint8, device=x.device)
        lib.ps_maxpool_fwd(y.data_ptr(), arg.data_ptr(), x.data_ptr(),
                           Nb, H, W, C, P, Q, k, k, stride, pad,
                           current_stream_ptr())
        ctx.save_for_backward(arg)
        ctx.conf = (Nb, C, H, W, P, Q, k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = require_lib()
        (arg,) = ctx.saved_tensors
        Nb, C, H, W, P, Q, k, stride, pad = ctx.conf
        dy = dy.contiguous(memory_format=_CL)
        dx = torch.empty((Nb, C, H, W), dtype=dy.dtype, device=dy.device,
                         memory_format=_CL)
        lib.ps_maxpool_bwd(dx.data_ptr(), dy.data_ptr(), arg.data_ptr(),
                           Nb, H, W, C, P, Q, k, k, stride, pad,
                           current_stream_ptr())
        return dx, None, None, None


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        lib = require_lib()
        x = x.contiguous(memory_format=_CL)
        Nb, C, H, W = x.shape
        y = torch.empty((Nb, C), dtype=x.dtype, device=x.device)
        lib.ps_gavgpool_fwd(y.data_ptr(), x.data_ptr(), Nb, H * W, C,
                            current_stream_ptr())
        ctx.conf = (Nb, C, H, W)
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = require_lib()
        Nb, C, H, W = ctx.conf
        dy = dy.contiguous()
        dx = torch.empty((Nb, C, H, W), dtype=dy.dtype, device=dy.device,
                         memory_format=_CL)
        lib.ps_gavgpool_bwd(dx.data_ptr(), dy.data_ptr(), Nb, H * W, C,
                            current_stream_ptr())
        return dx


def max_pool2d(x: torch.Tensor, kernel_size: int, stride: int = None,
               padding: int = 0) -> torch.Tensor:
    stride = stride if stride is not None else kernel_size
    if _supported(x):
        return _MaxPoolFn.apply(x, kernel_size, stride, padding)
    return F.max_pool2d(x, kernel_size, stride, padding)


def global_avg_pool(x: torch.Tensor) -> torch.Tensor:
    """adaptive_avg_pool2d(x, 1).flatten(1): [N,C,H,W] -> [N,C]."""
    if _supported(x):
        return _GlobalAvgPoolFn.apply(x)
    return F.adaptive_avg_pool2d(x, 1).flatten(1)
