"""PS-side SGD on the flat master buffer.

Reference parity: src/optim/sgd.py:59-92 — an SGD fork whose step() consumes
gradients arriving from the wire (there: a list of numpy arrays, one per
layer; here: ONE flat f32/bf16 tensor already summed across workers by RCCL).
The momentum/nesterov algebra matches torch.optim.SGD. On GPU the whole step
is one fused HIP kernel (ops/kernels/fused_sgd.hip) that also re-packs the
wire payload for the next weight broadcast.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..ops.functional import fused_sgd_step


class FlatSGD:
    def __init__(self, flat_w: torch.Tensor, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, nesterov: bool = False):
        if flat_w.dtype != torch.float32:
            raise TypeError("master weights must be f32")
        self.w = flat_w
        self.lr = float(lr)
        self.momentum = float(momentum)
        self.weight_decay = float(weight_decay)
        self.nesterov = bool(nesterov)
        self.m = torch.zeros_like(flat_w)

    @torch.no_grad()
    def step(self, grad_sum: torch.Tensor, grad_scale: float = 1.0,
             wire_out: Optional[torch.Tensor] = None,
             region=None, advance: bool = True) -> None:   # advance: parity with FlatAdam
        """region=(start, end): update only that flat slice — the per-bucket
        pipelined update/broadcast path (elementwise math: slicing exact)."""
        if region is not None:
            st, en = region
            fused_sgd_step(self.w[st:en], grad_sum[st:en], self.m[st:en],
                           self.lr, self.momentum, self.weight_decay,
                           grad_scale, self.nesterov,
                           wire_out[st:en] if wire_out is not None else None)
            return
        fused_sgd_step(self.w, grad_sum, self.m, self.lr, self.momentum,
                       self.weight_decay, grad_scale, self.nesterov, wire_out)

    def state_dict(self) -> dict:
        return {'momentum_buffer': self.m, 'lr': self.lr,
                'momentum': self.momentum, 'weight_decay': self.weight_decay,
                'nesterov': self.nesterov}

    def load_state_dict(self, sd: dict) -> None:
        self.m.copy_(sd['momentum_buffer'])
        self.lr = sd['lr']
        self.momentum = sd['momentum']
        self.weight_decay = sd['weight_decay']
        self.nesterov = sd['nesterov']
