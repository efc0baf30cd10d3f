"""Parameter-server role (rank 0): global model owner + fused update.

Reference parity: src/sync_replicas_master_nn.py (SyncReplicasMaster_NN) —
per step: announce + broadcast weights, fan-in gradients, average, update,
checkpoint. Differences by design (MI355X-first):

  * The PS is a GPU rank (the reference pinned it to CPU,
    sync_replicas_master_nn.py:131): master weights/momentum live in HBM and
    the whole update is ONE fused HIP kernel (ops/kernels/fused_sgd.hip)
    that also re-packs the next broadcast payload.
  * The Waitany-drain + per-layer += aggregation loop (:157-186, :239-241)
    is replaced by RCCL reduce(sum)-to-root per bucket: the xGMI fabric does
    the summation work in flight; the PS contributes zeros.
  * Partial aggregation (--num-aggregate, :179-207): in 'collective' mode
    every worker contributes and the average divides by W-1 (documented
    deviation, exact when there are no stragglers to drop); 'gather' mode
    (per-worker P2P buffers) restores arrival-order selection.
"""
from __future__ import annotations

import time
from typing import Optional

import torch
import torch.distributed as dist

from ..config import JobConfig, input_shape_of, num_classes_of
from ..models import build_model
from ..optim import FlatAdam, FlatSGD
from ..parallel.flat import FlatSpace, prep_model
from ..parallel.transport import ControlPlane, PSTransport
from ..utils.checkpoint import save_model_step
from ..utils.logging import get_logger, MASTER_LINE

logger = get_logger('ps_pytorch_amd.ps')


class ParameterServer:
    def __init__(self, cfg: JobConfig, rank: int, world: int,
                 device: torch.device, optimizer: Optional[str] = None):
        assert rank == 0, "the PS is rank 0"
        self.cfg = cfg
        self.rank = rank
        self.world = world
        self.device = device
        # --optimizer (default sgd; the reference ships Adam but hardwires
        # SGD, sync_replicas_master_nn.py:126) — explicit arg still wins
        self.optimizer_name = optimizer or getattr(cfg, 'optimizer', 'sgd')
        self.compute_dtype = (torch.bfloat16
                              if (device.type == 'cuda' and cfg.compute_dtype == 'bf16')
                              else torch.float32)
        self.wire_dtype = (torch.bfloat16
                           if (cfg.compress or cfg.wire_dtype == 'bf16') and device.type == 'cuda'
                           else torch.float32)
        self.cur_step = 0
        self.network = None
        self.flat: Optional[FlatSpace] = None
        self.transport: Optional[PSTransport] = None

    def build_model(self, num_classes: Optional[int] = None) -> None:
        cfg = self.cfg
        nc = num_classes if num_classes is not None else num_classes_of(cfg.dataset)
        in_ch = input_shape_of(cfg.dataset)[0]
        torch.manual_seed(cfg.seed)   # identical init across ranks
        net = build_model(cfg.network, num_classes=nc, in_channels=in_ch)
        net = prep_model(net, self.device, self.compute_dtype)
        self.network = net
        self.flat = FlatSpace(net, bucket_bytes=int(cfg.bucket_mb * 2 ** 20))
        # --compress-grad encodes per mode: gather -> block-scaled int8
        # payloads (4x, quant.hip); collective -> bf16 wire (2x) since a
        # summed int8 payload can't ride an in-flight ncclReduce.
        self.transport = PSTransport(self.flat, self.wire_dtype, self.device,
                                     self.rank, self.world,
                                     mode=cfg.aggregation,
                                     compress=cfg.compress,
                                     comm_type=cfg.comm_type,
                                     track_killed=(cfg.mode == 'timeout'
                                                   and cfg.aggregation == 'gather'))
        self.ctrl = (ControlPlane(self.rank, self.world)
                     if cfg.mode == 'kill' else None)
        if cfg.resume_step:
            # warm-start (beyond-reference: the reference never resumes):
            # load model_step_<K> params into the live flat views, then the
            # master copy below snapshots them; optimizer state starts fresh
            from ..utils.checkpoint import load_model_step
            load_model_step(net, cfg.train_dir, cfg.resume_step, strict=False)
            logger.info('PS resumed parameters from step %d', cfg.resume_step)
        # f32 master copy + optimizer state in HBM
        self.master_w = self.flat.flat_w.detach().to(torch.float32).clone()
        if self.optimizer_name == 'adam':
            self.optimizer = FlatAdam(self.master_w, lr=cfg.lr)
        else:
            self.optimizer = FlatSGD(self.master_w, lr=cfg.lr,
                                     momentum=cfg.momentum)
        # first broadcast payload
        self.transport.pack_weights_from(self.master_w)
        # pipelined-broadcast mode: step k's weights go out at the tail of
        # step k-1; step 0's go out here (matching the workers' first fetch)
        if self.transport.bcast_bucketed and dist.is_initialized():
            self.transport.bcast_all_buckets()

    @property
    def grad_scale(self) -> float:
        if self.cfg.aggregation == 'collective':
            return 1.0 / max(self.world - 1, 1)
        return 1.0 / max(min(self.cfg.num_aggregate, self.world - 1), 1)

    def step(self) -> None:
        """One synchronous step (mirrors DistributedWorker.train_step order)."""
        t = self.transport
        if t.bcast_bucketed:
            return self._step_pipelined()
        t.broadcast_weights()
        if t.mode == 'gather':
            t.post_gather_recvs()
            on_quota = self.ctrl.signal if self.ctrl is not None else None
            counts = t.drain_arrivals(self.cfg.num_aggregate, on_quota=on_quota)
            grad = t.acc_g
            # under-quota buckets (timeout-killed workers excluded from the
            # count): pre-scale so the optimizer's global 1/k yields the
            # exact average over the real arrivals
            k = max(1, min(self.cfg.num_aggregate, self.world - 1))
            for b, cnt in zip(self.flat.buckets, counts):
                if 0 < cnt < k:
                    grad[b.start:b.end].mul_(k / cnt)
        else:
            t.recv_buckets(self.flat.buckets)
            t.wait_all()
            grad = t.wire_g
        # fused: average-scale + momentum + update + re-pack next payload
        self.optimizer.step(grad, grad_scale=self.grad_scale,
                            wire_out=t.wire_w)
        self.cur_step += 1

    def _step_pipelined(self) -> None:
        """Collective-mode pipeline: per bucket, wait fan-in -> fused slice
        update -> broadcast that slice immediately. The fan-in tail of later
        buckets and the whole optimizer step hide under wire time; the
        per-rank collective order matches the workers' (transport.py
        bcast_bucketed note). The step's own weights went out at the tail of
        the previous step (step 0: at build_model)."""
        t = self.transport
        t.wait_bcasts()                      # previous tail fully on the wire
        t.recv_buckets(self.flat.buckets)    # post async per-bucket reduces
        last = self.cur_step + 1 >= self.cfg.max_steps
        for i, b in enumerate(self.flat.buckets):
            t.wait_reduce(i)
            self.optimizer.step(t.wire_g, grad_scale=self.grad_scale,
                                wire_out=t.wire_w, region=(b.start, b.end),
                                advance=(i == 0))
            if not last:
                # workers exit after their final step: no dangling
                # collective for them to match
                t.bcast_bucket(b)
        t._works.clear()
        self.cur_step += 1

    def start(self) -> None:
        """Run the PS loop for max_steps (ref: sync_replicas_master_nn.py:133)."""
        cfg = self.cfg
        while self.cur_step < cfg.max_steps:
            t0 = time.time()
            self.step()
            if self.cur_step % cfg.log_interval == 0:
                logger.info(MASTER_LINE.format(self.cur_step, time.time() - t0))
            # the reference's master checkpoints non-BN nets
            # (sync_replicas_master_nn.py:194-196); BN nets are saved by
            # worker rank 1 (see worker.py).
            if (self.cur_step % cfg.eval_freq == 0 and
                    not any(k in cfg.network.lower() for k in ('resnet', 'vgg'))):
                self._save_checkpoint()

    def _save_checkpoint(self) -> None:
        sd = self.flat.state_dict_from_flat(
            self.master_w[:self.flat.total].to(self.flat.dtype))
        save_model_step(sd, self.cfg.train_dir, self.cur_step)
