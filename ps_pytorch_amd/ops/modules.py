"""Fused NHWC BatchNorm(+residual)(+ReLU) module on the hand-written CDNA4
kernels (ops/kernels/batchnorm.hip), with a torch fallback that doubles as
the numerics reference (tests/test_bn_*).

State-dict surface matches nn.BatchNorm2d (weight, bias, running_mean,
running_var, num_batches_tracked) so checkpoints stay evaluator-compatible.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import require_lib, dtype_tag, current_stream_ptr

_CL = torch.channels_last


def _kernel_ok(x: torch.Tensor) -> bool:
    if not x.is_cuda or x.dim() != 4:
        return False
    C = x.shape[1]
    # stats kernels need C/8 lane-groups dividing the 256-thread block
    return (C % 8 == 0 and C <= 2048 and (256 % (C // 8)) == 0 and
            x.dtype in (torch.bfloat16, torch.float32))


class _FusedBNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, residual, running_mean, running_var,
                momentum, eps, relu):
        lib = require_lib()
        x = x.contiguous(memory_format=_CL)
        if residual is not None:
            residual = residual.contiguous(memory_format=_CL)
        N, C, H, W = x.shape
        M = N * H * W
        y = torch.empty_like(x)
        save_mean = torch.empty(C, dtype=torch.float32, device=x.device)
        save_invstd = torch.empty(C, dtype=torch.float32, device=x.device)
        ws = torch.empty(5 * C, dtype=torch.float32, device=x.device)
        partial = torch.empty(1024 * 2 * C, dtype=torch.float32, device=x.device)
        # bit-packed relu mask (1 byte per 8 channels): the backward reads
        # this instead of re-reading the full y tensor for the mask
        mask = (torch.empty(M * (C // 8), dtype=torch.uint8, device=x.device)
                if relu else None)
        # stats handshake: the producing conv's epilogue may have already
        # written this activation's per-channel sum/sumsq partials
        ext = getattr(x, '_ps_bn_stats', None)
        lib.ps_bn_fwd(
            x.data_ptr(), y.data_ptr(), gamma.data_ptr(), beta.data_ptr(),
            running_mean.data_ptr(), running_var.data_ptr(),
            save_mean.data_ptr(), save_invstd.data_ptr(), ws.data_ptr(),
            partial.data_ptr(),
            residual.data_ptr() if residual is not None else 0,
            mask.data_ptr() if mask is not None else 0,
            ext[0].data_ptr() if ext is not None else 0,
            ext[1] if ext is not None else 0,
            M, C, float(momentum), float(eps), 1, int(relu),
            dtype_tag(x.dtype), current_stream_ptr())
        ctx.save_for_backward(x, gamma, save_mean, save_invstd)
        ctx.mask = mask
        ctx.relu = relu
        ctx.has_res = residual is not None
        # steal-mode flat_g targets (see parallel/flat.py attach_grads)
        ctx.gtgt = (getattr(gamma, '_ps_flat_grad_fn', None),
                    getattr(beta, '_ps_flat_grad_fn', None))
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = require_lib()
        x, gamma, save_mean, save_invstd = ctx.saved_tensors
        mask = ctx.mask
        dy = dy.contiguous(memory_format=_CL)
        N, C, H, W = x.shape
        M = N * H * W
        dx = torch.empty_like(x)
        dres = torch.empty_like(x) if ctx.has_res else None
        g_tgt = ctx.gtgt[0]() if ctx.gtgt[0] is not None else None
        b_tgt = ctx.gtgt[1]() if ctx.gtgt[1] is not None else None
        dgamma = (g_tgt if g_tgt is not None and g_tgt.dtype == gamma.dtype
                  and g_tgt.is_cuda else torch.empty_like(gamma))
        dbeta = (b_tgt if b_tgt is not None and b_tgt.dtype == gamma.dtype
                 and b_tgt.is_cuda else torch.empty_like(gamma))
        ws = torch.empty(5 * C, dtype=torch.float32, device=x.device)
        partial = torch.empty(1024 * 2 * C, dtype=torch.float32, device=x.device)
        lib.ps_bn_bwd(
            x.data_ptr(), mask.data_ptr() if mask is not None else 0,
            dy.data_ptr(), gamma.data_ptr(),
            save_mean.data_ptr(), save_invstd.data_ptr(), dx.data_ptr(),
            dgamma.data_ptr(), dbeta.data_ptr(),
            dres.data_ptr() if dres is not None else 0, ws.data_ptr(),
            partial.data_ptr(),
            M, C, int(ctx.relu), dtype_tag(x.dtype), current_stream_ptr())
        return (dx, dgamma, dbeta, dres, None, None, None, None, None)


def _bn_eval_fused(x, gamma, beta, residual, rmean, rvar, eps, relu):
    lib = require_lib()
    x = x.contiguous(memory_format=_CL)
    if residual is not None:
        residual = residual.contiguous(memory_format=_CL)
    N, C, H, W = x.shape
    y = torch.empty_like(x)
    ws = torch.empty(5 * C, dtype=torch.float32, device=x.device)
    lib.ps_bn_fwd(x.data_ptr(), y.data_ptr(), gamma.data_ptr(),
                  beta.data_ptr(), rmean.data_ptr(), rvar.data_ptr(),
                  0, 0, ws.data_ptr(), 0,
                  residual.data_ptr() if residual is not None else 0, 0,
                  0, 0,
                  N * H * W, C, 0.0, float(eps), 0, int(relu),
                  dtype_tag(x.dtype), current_stream_ptr())
    return y


class PsBatchNorm2d(nn.Module):
    """BatchNorm2d with optional fused residual-add + ReLU epilogue.

    forward(x, residual=None) == relu?(bn(x) + residual).
    GPU training path: hand-written NHWC kernels; otherwise torch ops.
    """

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer('running_mean', torch.zeros(num_features))
        self.register_buffer('running_var', torch.ones(num_features))
        self.register_buffer('num_batches_tracked',
                             torch.tensor(0, dtype=torch.long))

    def _ensure_f32_stats(self) -> None:
        # running stats stay f32 even when the module is cast to bf16
        if self.running_mean.dtype != torch.float32:
            self.running_mean.data = self.running_mean.data.float()
            self.running_var.data = self.running_var.data.float()

    # num_batches_tracked: the canonical buffer update is a device-side
    # long add — 20 kernel launches per ResNet-18 step for a counter nobody
    # reads during training. Count on the host and materialize into the
    # buffer only when a state_dict is taken (or synced in from a load).
    def _sync_nbt_out(self, *a):
        host = getattr(self, '_nbt_host', 0)
        if host:
            self.num_batches_tracked.fill_(int(self.num_batches_tracked)
                                           + host)
            self._nbt_host = 0

    def _load_from_state_dict(self, *args, **kw):
        self._nbt_host = 0
        super()._load_from_state_dict(*args, **kw)

    def _save_to_state_dict(self, *args, **kw):
        # (root state_dict() recurses here, not into child .state_dict())
        self._sync_nbt_out()
        super()._save_to_state_dict(*args, **kw)

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.training:
            self._nbt_host = getattr(self, '_nbt_host', 0) + 1
        if self.training and _kernel_ok(x):
            self._ensure_f32_stats()
            return _FusedBNFn.apply(x, self.weight, self.bias, residual,
                                    self.running_mean, self.running_var,
                                    self.momentum, self.eps, self.relu)
        if (not self.training and _kernel_ok(x)
                and not torch.is_grad_enabled()):
            # eval-mode fused path (running-stats normalize + res + relu)
            self._ensure_f32_stats()
            return _bn_eval_fused(x, self.weight, self.bias, residual,
                                  self.running_mean, self.running_var,
                                  self.eps, self.relu)
        # torch fallback (CPU, eval mode, unsupported shapes)
        rm = self.running_mean
        rv = self.running_var
        if rm.dtype != x.dtype and not x.is_cuda:
            rm = rm.to(x.dtype)
            rv = rv.to(x.dtype)
            y = F.batch_norm(x, rm, rv, self.weight, self.bias,
                             self.training, self.momentum, self.eps)
            if self.training:   # write back updated stats
                self.running_mean.data.copy_(rm.float())
                self.running_var.data.copy_(rv.float())
        else:
            self._ensure_f32_stats()
            y = F.batch_norm(x, self.running_mean, self.running_var,
                             self.weight, self.bias, self.training,
                             self.momentum, self.eps)
        if residual is not None:
            y = y + residual
        if self.relu:
            y = F.relu(y)
        return y

    def extra_repr(self) -> str:
        return (f"{self.num_features}, eps={self.eps}, "
                f"momentum={self.momentum}, relu={self.relu}")
