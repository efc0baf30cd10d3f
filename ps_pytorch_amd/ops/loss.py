"""Fused softmax cross-entropy on the in-tree CDNA4 kernels
(ops/kernels/softmaxce.hip). Mean reduction, f32 loss from bf16 logits.

Replaces the at::native softmax/nll pair the reference reaches through
nn.CrossEntropyLoss (ref: src/distributed_worker.py:98, nn_ops.py:60) on
the GPU hot path; CPU path is F.cross_entropy on f32 (the reference)."""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import require_lib, current_stream_ptr

_ENABLED = os.environ.get('PS_CE', '1') != '0'


class _SoftmaxCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        lib = require_lib()
        x = logits.contiguous()
        t = target.contiguous()
        if t.dtype != torch.int64:
            t = t.to(torch.int64)
        M, C = x.shape
        loss = torch.empty(1, dtype=torch.float32, device=x.device)
        lse = torch.empty(M, dtype=torch.float32, device=x.device)
        row_ws = torch.empty(M, dtype=torch.float32, device=x.device)
        lib.ps_softmax_ce_fwd(loss.data_ptr(), lse.data_ptr(),
                              row_ws.data_ptr(), x.data_ptr(), t.data_ptr(),
                              M, C, current_stream_ptr())
        ctx.save_for_backward(x, lse, t)
        return loss[0]

    @staticmethod
    def backward(ctx, dloss):
        lib = require_lib()
        x, lse, t = ctx.saved_tensors
        M, C = x.shape
        dl = dloss.reshape(1).to(device=x.device, dtype=torch.float32) \
                  .contiguous()
        dx = torch.empty_like(x)
        lib.ps_softmax_ce_bwd(dx.data_ptr(), x.data_ptr(), lse.data_ptr(),
                              t.data_ptr(), dl.data_ptr(), M, C,
                              current_stream_ptr())
        return dx, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean-reduced CE; fused GPU kernel for bf16 [M, C] logits."""
    if (_ENABLED and logits.is_cuda and logits.dim() == 2
            and logits.dtype == torch.bfloat16):
        return _SoftmaxCEFn.apply(logits, target)
    return F.cross_entropy(logits.float(), target)
