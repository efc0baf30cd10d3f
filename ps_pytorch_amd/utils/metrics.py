"""prec@k accuracy (ref: src/nn_ops.py:14-27)."""
from __future__ import annotations

import torch


def accuracy(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    maxk = max(topk)
    _, pred = output.topk(maxk, dim=1)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    batch = target.size(0)
    for k in topk:
        ck = correct[:k].reshape(-1).float().sum(0)
        res.append(ck.mul_(100.0 / batch))
    return res
