"""Single-machine trainer — the scalability yardstick and correctness oracle.

Reference parity: src/nn_ops.py:29-106 (NN_Trainer.build_model /
train_and_validate / validate) + src/single_machine.py. Uses the SAME engine
pieces as the distributed path (FlatSpace views + FlatSGD fused update) so
the 1-GPU bench measures the framework, not a different code path:
master weights f32, compute dtype bf16 on GPU (f32 on CPU), one fused
update kernel per step.
"""
from __future__ import annotations

import time
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .ops.loss import cross_entropy as ps_cross_entropy

from .config import JobConfig, num_classes_of, input_shape_of
from .models import build_model
from .optim import FlatSGD, FlatAdam
from .parallel.flat import FlatSpace, prep_model
from .utils.metrics import accuracy
from .utils.logging import get_logger

logger = get_logger('ps_pytorch_amd.trainer')


class NNTrainer:
    def __init__(self, cfg: JobConfig, device: Optional[torch.device] = None):
        self.cfg = cfg
        if device is None:
            device = torch.device('cuda' if (cfg.enable_gpu and torch.cuda.is_available())
                                  else 'cpu')
        self.device = device
        self.compute_dtype = (torch.bfloat16 if (device.type == 'cuda' and
                                                 cfg.compute_dtype == 'bf16')
                              else torch.float32)
        self.network: Optional[nn.Module] = None
        self.flat: Optional[FlatSpace] = None
        self.optimizer = None
        self.cur_step = 0

    def build_model(self, num_classes: Optional[int] = None) -> None:
        cfg = self.cfg
        nc = num_classes if num_classes is not None else num_classes_of(cfg.dataset)
        in_ch = input_shape_of(cfg.dataset)[0]
        torch.manual_seed(cfg.seed)
        net = build_model(cfg.network, num_classes=nc, in_channels=in_ch)
        net = prep_model(net, self.device, self.compute_dtype)
        self.network = net
        self.flat = FlatSpace(net, bucket_bytes=int(cfg.bucket_mb * 2 ** 20))
        if getattr(cfg, 'resume_step', 0):
            from .utils.checkpoint import load_model_step
            load_model_step(net, cfg.train_dir, cfg.resume_step, strict=False)
        # f32 master copy + fused update; flat_g is the "wire" (local = trivially summed)
        self.master_w = self.flat.flat_w.detach().to(torch.float32).clone()
        if getattr(cfg, 'optimizer', 'sgd') == 'adam':
            self.optimizer = FlatAdam(self.master_w, lr=cfg.lr)
        else:
            self.optimizer = FlatSGD(self.master_w, lr=cfg.lr,
                                     momentum=cfg.momentum)
        # steal mode on GPU: Ps ops write grads into flat_g directly (no
        # per-param AccumulateGrad add kernels); view mode on CPU
        self.flat.attach_grads(steal=(self.device.type == 'cuda'))

    def _loss(self, data, target):
        out = self.network(data)
        return ps_cross_entropy(out, target), out

    def _step_body(self, data, target) -> torch.Tensor:
        self.flat.zero_grads()
        loss, _ = self._loss(data, target)
        loss.backward()
        self.flat.harvest_grads()   # steal mode: catch torch-fallback grads
        # fused: master update + re-pack into the live (possibly bf16) params
        self.optimizer.step(self.flat.flat_g, grad_scale=1.0,
                            wire_out=self.flat.flat_w)
        return loss.detach()

    def train_step(self, data, target) -> float:
        """forward/backward/update; returns loss."""
        if data.dim() == 4 and data.is_cuda:
            data = data.contiguous(memory_format=torch.channels_last)
        loss = self._step_body(data, target)
        self.cur_step += 1
        return float(loss)

    # ---- hipGraph-captured step (single-GPU hot path) ----
    # The whole step (zero -> forward -> backward -> fused update) replays as
    # ONE hipGraph: ~580 kernel dispatches/step collapse to one launch,
    # removing host launch overhead and inter-kernel gaps (MI355X guide:
    # 'graph-replay-floor'). Static input buffers; loss read lazily.

    def enable_graph(self, example_x: torch.Tensor,
                     example_y: torch.Tensor) -> bool:
        if self.device.type != 'cuda':
            return False
        example_x = example_x.contiguous(memory_format=torch.channels_last) \
            if example_x.dim() == 4 else example_x
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):     # warmup allocs/algos outside capture
                    self._step_body(example_x, example_y)
            torch.cuda.current_stream().wait_stream(side)
            self._gx = example_x.clone()
            self._gy = example_y.clone()
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._gloss = self._step_body(self._gx, self._gy)
            return True
        except Exception as e:   # pragma: no cover - depends on runtime
            import warnings
            warnings.warn(f"hipGraph capture failed, staying eager: {e}")
            self._graph = None
            return False

    def graph_step(self, data: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        """Replay the captured step on new data; returns the (device) loss
        tensor — read it only when needed to avoid a sync per step."""
        self._gx.copy_(data, non_blocking=True)
        self._gy.copy_(target, non_blocking=True)
        self._graph.replay()
        # host-side BN batch counters don't tick during replay (host code
        # doesn't run); keep them canonical here
        from .ops.modules import PsBatchNorm2d
        for m in self.network.modules():
            if isinstance(m, PsBatchNorm2d):
                m._nbt_host = getattr(m, '_nbt_host', 0) + 1
        self.cur_step += 1
        return self._gloss

    def train_and_validate(self, train_loader, test_loader,
                           max_steps: Optional[int] = None) -> None:
        cfg = self.cfg
        max_steps = max_steps or cfg.max_steps
        self.network.train()
        for epoch in range(cfg.epochs):
            for batch_idx, (data, target) in enumerate(train_loader):
                t0 = time.time()
                data = data.to(self.device, self.compute_dtype)
                target = target.to(self.device)
                loss = self.train_step(data, target)
                if self.cur_step % cfg.log_interval == 0:
                    logger.info('Step: %d, Epoch: %d, Loss: %.4f, Time: %.4f',
                                self.cur_step, epoch, loss, time.time() - t0)
                if self.cur_step >= max_steps:
                    return
            self.validate(test_loader)

    @torch.no_grad()
    def validate(self, test_loader) -> float:
        self.network.eval()
        tot, p1_sum, p5_sum, loss_sum = 0, 0.0, 0.0, 0.0
        for data, target in test_loader:
            data = data.to(self.device, self.compute_dtype)
            target = target.to(self.device)
            out = self.network(data)
            loss_sum += float(F.cross_entropy(out.float(), target,
                                              reduction='sum'))
            k = min(5, out.shape[1])
            p1, pk = accuracy(out.float(), target, topk=(1, k))
            bs = target.size(0)
            p1_sum += float(p1) * bs
            p5_sum += float(pk) * bs
            tot += bs
        self.network.train()
        logger.info('Validation: loss %.4f, prec@1 %.2f, prec@5 %.2f',
                    loss_sum / max(tot, 1), p1_sum / max(tot, 1), p5_sum / max(tot, 1))
        return p1_sum / max(tot, 1)
