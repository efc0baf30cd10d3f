"""PS-side Adam on the flat master buffer.

Reference parity: src/optim/adam.py:38-95 (full Adam incl. amsgrad, consuming
wire gradients). GPU path: ONE fused HIP kernel over the flat buffer
(ops/kernels/fused_sgd.hip: ps_fused_adam) — m/v update, bias-corrected
step, optional amsgrad and wire re-pack in a single pass; CPU path: the
equivalent torch ops (the numerics reference).
"""
from __future__ import annotations

from typing import Optional

import torch


class FlatAdam:
    def __init__(self, flat_w: torch.Tensor, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, amsgrad: bool = False):
        if flat_w.dtype != torch.float32:
            raise TypeError("master weights must be f32")
        self.w = flat_w
        self.lr = float(lr)
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(eps)
        self.weight_decay = float(weight_decay)
        self.amsgrad = bool(amsgrad)
        self.t = 0
        self.exp_avg = torch.zeros_like(flat_w)
        self.exp_avg_sq = torch.zeros_like(flat_w)
        self.max_exp_avg_sq = torch.zeros_like(flat_w) if amsgrad else None

    @torch.no_grad()
    def step(self, grad_sum: torch.Tensor, grad_scale: float = 1.0,
             wire_out: Optional[torch.Tensor] = None,
             region=None, advance: bool = True) -> None:
        """region=(start, end): slice update for the pipelined per-bucket
        path; pass advance=True exactly once per logical step (first
        bucket) so the bias-correction t stays per-step."""
        from ..ops.functional import fused_adam_step
        if advance:
            self.t += 1
        if region is not None:
            st, en = region
            fused_adam_step(self.w[st:en], grad_sum[st:en],
                            self.exp_avg[st:en], self.exp_avg_sq[st:en],
                            self.t, self.lr, self.beta1, self.beta2, self.eps,
                            self.weight_decay, grad_scale,
                            self.max_exp_avg_sq[st:en] if self.max_exp_avg_sq
                            is not None else None,
                            wire_out[st:en] if wire_out is not None else None)
            return
        fused_adam_step(self.w, grad_sum, self.exp_avg, self.exp_avg_sq,
                        self.t, self.lr, self.beta1, self.beta2, self.eps,
                        self.weight_decay, grad_scale,
                        self.max_exp_avg_sq, wire_out)

    def state_dict(self) -> dict:
        return {'t': self.t, 'exp_avg': self.exp_avg,
                'exp_avg_sq': self.exp_avg_sq,
                'max_exp_avg_sq': self.max_exp_avg_sq,
                'lr': self.lr, 'betas': (self.beta1, self.beta2),
                'eps': self.eps, 'weight_decay': self.weight_decay,
                'amsgrad': self.amsgrad}

    def load_state_dict(self, sd: dict) -> None:
        self.t = sd['t']
        self.exp_avg.copy_(sd['exp_avg'])
        self.exp_avg_sq.copy_(sd['exp_avg_sq'])
        if self.amsgrad and sd['max_exp_avg_sq'] is not None:
            self.max_exp_avg_sq.copy_(sd['max_exp_avg_sq'])
        self.lr = sd['lr']
        self.beta1, self.beta2 = sd['betas']
        self.eps = sd['eps']
        self.weight_decay = sd['weight_decay']
