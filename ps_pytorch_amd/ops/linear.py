"""PsLinear — bf16 MFMA GEMM Linear on the in-tree CDNA4 kernels.

Replaces the Tensile/rocBLAS GEMMs the reference reaches through nn.Linear
(ref: src/model_ops/lenet.py:29-33 fc layers, resnet.py:100-103 classifier).

A Linear IS a 1x1 convolution over an [M, 1, 1, K] NHWC image, and torch's
weight layout [N, K] is exactly the conv kernel's [K_out, R*S*C] operand —
so all three GEMMs ride the tuned implicit-GEMM kernel set in
ops/kernels/conv.hip with zero new device code:

  fwd   : y[M,N] = x[M,K] @ w[N,K]^T + b     -> ps_conv_fwd   (R=S=1)
  dgrad : dx[M,K] = dy[M,N] @ w[N,K]          -> ps_conv_dgrad (host
          transposes w once per backward, as the conv path does)
  wgrad : dw[N,K] = dy^T @ x  (contraction M) -> ps_conv_wgrad split-K with
          the same deterministic slab reduce
  db    : column sum of dy                    -> ps_conv_bias_grad

PsLinear subclasses nn.Linear: parameter shapes, init and state_dict
surface are identical (evaluator/checkpoint compatible).
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import require_lib, current_stream_ptr
from .conv import _PADK, _pad_rows, _pad_target, _wgrad_split

# A/B kill-switch: PS_LINEAR=0 routes through torch (Tensile) GEMMs.
_ENABLED = os.environ.get('PS_LINEAR', '1') != '0'


def _supported(x: torch.Tensor, w: torch.Tensor) -> bool:
    return (_ENABLED and x.is_cuda and x.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16)


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        lib = require_lib()
        shape = x.shape
        x2 = x.reshape(-1, shape[-1]).contiguous()
        wc = w.contiguous()
        M, K = x2.shape
        N = wc.shape[0]
        y = torch.empty((M, N), dtype=x.dtype, device=x.device)
        lib.ps_conv_fwd(x2.data_ptr(), wc.data_ptr(),
                        b.data_ptr() if b is not None else 0, y.data_ptr(), 0,
                        M, 1, 1, K, N, 1, 1, 1, 1, 1, 0,
                        current_stream_ptr())
        ctx.save_for_backward(x2, wc)
        ctx.conf = (shape, b is not None)
        # steal-mode flat_g targets (see parallel/flat.py attach_grads)
        ctx.gtgt = (getattr(w, '_ps_flat_grad_fn', None),
                    getattr(b, '_ps_flat_grad_fn', None) if b is not None else None)
        return y.reshape(*shape[:-1], N)

    @staticmethod
    def backward(ctx, dout):
        lib = require_lib()
        x, w = ctx.saved_tensors
        shape, has_bias = ctx.conf
        M, K = x.shape
        N = w.shape[0]
        dy = dout.reshape(M, N).contiguous()
        dx = dw = db = None
        # ragged-N (fc1's 500, classifier's 10): the dgrad/wgrad contraction
        # runs over N — pad dy and wT rows to a 64-multiple so the b128 load
        # path applies (see ops/conv.py _pad_rows)
        padn = (_PADK and N % 8 != 0
                and (ctx.needs_input_grad[0] or ctx.needs_input_grad[1]))
        if padn:
            Np = _pad_target(N)
            dyp = _pad_rows(lib, dy, M, N, Np)
        else:
            Np, dyp = N, dy
        if ctx.needs_input_grad[0]:
            wt = torch.empty(K * N, dtype=w.dtype, device=w.device)
            lib.ps_wt_transpose(wt.data_ptr(), w.data_ptr(), N, K,
                                current_stream_ptr())
            if padn:
                wt = _pad_rows(lib, wt, K, N, Np)
            dx = torch.empty_like(x)
            lib.ps_conv_dgrad(dyp.data_ptr(), wt.data_ptr(), dx.data_ptr(), 0,
                              M, 1, 1, K, Np, 1, 1, 1, 1, 1, 0,
                              current_stream_ptr())
            dx = dx.reshape(shape)
        if ctx.needs_input_grad[1]:
            split = _wgrad_split(M, Np, K, 1, 1, stride=1, pad=0, P=1, Q=1)
            partial = torch.empty(split * Np * K, dtype=torch.float32,
                                  device=x.device)
            wt_tgt = ctx.gtgt[0]() if ctx.gtgt[0] is not None else None
            if padn:
                dwp = torch.empty((Np, K), dtype=w.dtype, device=w.device)
                lib.ps_conv_wgrad(dyp.data_ptr(), x.data_ptr(),
                                  partial.data_ptr(), dwp.data_ptr(),
                                  M, 1, 1, K, Np, 1, 1, 1, 1, 1, 0,
                                  split, current_stream_ptr())
                dw = (wt_tgt if wt_tgt is not None and wt_tgt.is_cuda
                      else torch.empty_like(w))
                dw.copy_(dwp[:N])
            else:
                dw = (wt_tgt if wt_tgt is not None and wt_tgt.dtype == w.dtype
                      and wt_tgt.is_cuda else torch.empty_like(w))
                lib.ps_conv_wgrad(dy.data_ptr(), x.data_ptr(),
                                  partial.data_ptr(), dw.data_ptr(),
                                  M, 1, 1, K, N, 1, 1, 1, 1, 1, 0,
                                  split, current_stream_ptr())
        if has_bias and ctx.needs_input_grad[2]:
            b_tgt = ctx.gtgt[1]() if ctx.gtgt[1] is not None else None
            db = (b_tgt if b_tgt is not None and b_tgt.dtype == dy.dtype
                  and b_tgt.is_cuda
                  else torch.empty(N, dtype=dy.dtype, device=dy.device))
            bpart = torch.empty(512 * Np, dtype=torch.float32,
                                device=dy.device)
            if padn:   # aligned rows: colsum's ushort8 path (see conv.py)
                dbp = torch.empty(Np, dtype=dy.dtype, device=dy.device)
                lib.ps_conv_bias_grad(dbp.data_ptr(), dyp.data_ptr(),
                                      bpart.data_ptr(), M, Np,
                                      current_stream_ptr())
                db.copy_(dbp[:N])
            else:
                lib.ps_conv_bias_grad(db.data_ptr(), dy.data_ptr(),
                                      bpart.data_ptr(), M, N,
                                      current_stream_ptr())
        return dx, dw, db


class PsLinear(nn.Linear):
    """nn.Linear whose GPU bf16 path runs the in-tree CDNA4 MFMA GEMMs."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _supported(x, self.weight):
            return _LinearFn.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)
