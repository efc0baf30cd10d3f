"""Worker role: local replica + overlapped bucketed gradient push.

Reference parity: src/distributed_worker.py (per step: fetch step ->
bcast-recv weights -> forward/backward -> per-layer grad push with send
pipelining). The reference's signature optimization — per-layer Isend
interleaved with a hand-unrolled backward (src/model_ops/resnet_split.py:
365-501) — is realized here for EVERY model via post-accumulate-grad hooks:
as backward fills a bucket (buckets are laid out in backward order,
parallel/flat.py), the bucket is packed to wire dtype and reduced to the PS
asynchronously while backward continues. RCCL runs the reduction on its own
HIP stream; program-order bucket launch on every rank replaces MPI tags.
"""
from __future__ import annotations

import os
import time
from typing import Optional

import torch

from ..ops.loss import cross_entropy as ps_cross_entropy

from ..config import JobConfig, input_shape_of, num_classes_of
from ..models import build_model
from ..parallel.flat import FlatSpace, prep_model
from ..parallel.transport import ControlPlane, PSTransport, StepKilled
from ..utils.checkpoint import save_model_step
from ..utils.logging import get_logger, worker_log_line

logger = get_logger('ps_pytorch_amd.worker')

# PS_HIP_TIMING=1: per-phase spans measured with HIP events (GPU execution
# time) instead of host timers (enqueue time) — see _forward_backward.
_HIP_TIMING = os.environ.get('PS_HIP_TIMING', '0') == '1'


class DistributedWorker:
    def __init__(self, cfg: JobConfig, rank: int, world: int,
                 device: torch.device):
        self.cfg = cfg
        self.rank = rank
        self.world = world
        self.device = device
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

    # ---- setup ----

    def build_model(self, num_classes: Optional[int] = None) -> None:
        cfg = self.cfg
        nc = num_classes if num_classes is not None else num_classes_of(cfg.dataset)
        in_ch = input_shape_of(cfg.dataset)[0]
        torch.manual_seed(cfg.seed)   # same init on every rank
        net = build_model(cfg.network, num_classes=nc, in_channels=in_ch)
        net = prep_model(net, self.device, self.compute_dtype)
        self.network = net
        self.flat = FlatSpace(net, bucket_bytes=int(cfg.bucket_mb * 2 ** 20))
        self.flat.attach_grads(steal=(self.device.type == 'cuda'))
        self.transport = PSTransport(self.flat, self.wire_dtype, self.device,
                                     self.rank, self.world,
                                     mode=cfg.aggregation,
                                     compress=cfg.compress,
                                     comm_type=cfg.comm_type,
                                     track_killed=(cfg.mode == 'timeout'
                                                   and cfg.aggregation == 'gather'))
        # straggler handling (ref resnet_split.py:503-728): 'kill' polls the
        # PS's gloo signal from backward hooks; 'timeout' aborts locally past
        # --kill-threshold seconds. Both raise StepKilled mid-backward and
        # send zero payloads for the un-pushed buckets (matching intact).
        self.ctrl = ControlPlane(self.rank, self.world) if cfg.mode == 'kill' else None
        self._step_start = 0.0
        if cfg.overlap:
            self._install_hooks()

    def _install_hooks(self) -> None:
        """Bucket bookkeeping driven by autograd (the overlap engine)."""
        self._param_bucket = {}
        self._bucket_nparams = [0] * len(self.flat.buckets)
        for b in self.flat.buckets:
            for pid in b.param_ids:
                self._param_bucket[pid] = b.index
                if self.flat.params[pid].requires_grad:
                    self._bucket_nparams[b.index] += 1
        self._pending = list(self._bucket_nparams)
        self._ready = [False] * len(self.flat.buckets)
        self._next_launch = 0
        for pid, p in enumerate(self.flat.params):
            if not p.requires_grad:
                continue
            p.register_post_accumulate_grad_hook(self._make_hook(pid))

    def _make_hook(self, pid: int):
        def hook(_param):
            self._check_abort()
            self.flat.ensure_grad_in_flat(pid)
            bi = self._param_bucket[pid]
            self._pending[bi] -= 1
            if self._pending[bi] == 0:
                self._ready[bi] = True
                self._flush_ready()
        return hook

    def _check_abort(self) -> None:
        """Raise StepKilled if the PS said stop (kill mode) or the step ran
        past --kill-threshold (timeout mode)."""
        if self.ctrl is not None and self.ctrl.killed():
            raise StepKilled(f"rank {self.rank}: PS kill signal")
        if (self.cfg.mode == 'timeout'
                and time.time() - self._step_start > self.cfg.kill_threshold):
            raise StepKilled(f"rank {self.rank}: step exceeded "
                             f"{self.cfg.kill_threshold}s")

    def _flush_ready(self) -> None:
        while (self._next_launch < len(self._ready)
               and self._ready[self._next_launch]):
            self.transport.push_bucket(self.flat.buckets[self._next_launch])
            self._next_launch += 1

    def _reset_bucket_state(self) -> None:
        self._next_launch = 0
        if self.cfg.overlap:
            self._pending = list(self._bucket_nparams)
            self._ready = [False] * len(self.flat.buckets)

    # ---- per-step protocol (must mirror ParameterServer.step order) ----

    def fetch_weights(self) -> None:
        self.transport.broadcast_weights()
        self.transport.unpack_weights_into(self.flat.flat_w)

    def push_gradients(self, killed: bool = False) -> None:
        # timeout-mode marking: tell the PS whether this step's payloads are
        # real before it counts them toward the --num-aggregate quota
        self.transport.send_killed_flag(killed)
        if not self.cfg.overlap and not killed:
            self.flat.harvest_grads()   # steal mode: catch fallback grads
        if self.cfg.overlap:
            if not killed:
                self._flush_ready()
            if self._next_launch < len(self.flat.buckets):
                # remaining buckets: zero payloads after an abort, else
                # params that never got grads (unused in graph)
                for b in self.flat.buckets[self._next_launch:]:
                    self.transport.push_bucket(b, killed=killed)
                self._next_launch = len(self.flat.buckets)
        else:
            for b in self.flat.buckets[self._next_launch:]:
                self.transport.push_bucket(b, killed=killed)
            self._next_launch = len(self.flat.buckets)
        self.transport.wait_all()

    def _forward_backward(self, data, target):
        """forward + backward with the straggler-abort protocol around it.
        Returns (loss_or_None, killed).

        PS_HIP_TIMING=1: bracket the phases with HIP events instead of host
        timers — host time.time() around async GPU work measures ENQUEUE,
        not execution (SURVEY.md §5 tracing obligation). Events resolve in
        _resolve_phase_timing() at log time (one sync in timing mode only).
        """
        if self.ctrl is not None:
            self.ctrl.post()
        self._step_start = time.time()
        self.f_dur = self.b_dur = 0.0
        hip_ev = None
        if _HIP_TIMING and data.is_cuda:
            hip_ev = [torch.cuda.Event(enable_timing=True) for _ in range(3)]
            hip_ev[0].record()
        try:
            t0 = time.time()
            out = self.network(data)
            loss = ps_cross_entropy(out, target)
            if hip_ev is not None:
                hip_ev[1].record()
            self.f_dur = time.time() - t0
            t0 = time.time()
            loss.backward()
            if hip_ev is not None:
                hip_ev[2].record()
            self.b_dur = time.time() - t0
            self._hip_ev = hip_ev
            if not self.cfg.overlap:
                self._check_abort()
            return loss.detach(), False
        except StepKilled as e:
            logger.info('%s — aborting rest of backward', e)
            self._hip_ev = None
            return None, True

    def _resolve_phase_timing(self):
        """In PS_HIP_TIMING mode, replace the host-measured forward/backward
        spans with HIP-event-measured GPU execution times (seconds)."""
        ev = getattr(self, '_hip_ev', None)
        if ev is not None:
            ev[2].synchronize()
            self.f_dur = ev[0].elapsed_time(ev[1]) / 1e3
            self.b_dur = ev[1].elapsed_time(ev[2]) / 1e3
            self._hip_ev = None

    def train_step(self, data: torch.Tensor, target: torch.Tensor):
        """One synchronous PS step; returns detached loss (None if killed)."""
        if data.dim() == 4 and data.is_cuda:
            data = data.contiguous(memory_format=torch.channels_last)
        self.fetch_weights()
        self.flat.zero_grads()
        self._reset_bucket_state()
        loss, killed = self._forward_backward(data, target)
        self.push_gradients(killed=killed)
        if self.ctrl is not None:
            self.ctrl.finish()
        self.cur_step += 1
        return loss

    # ---- training loop ----

    def train(self, train_loader, test_loader=None) -> None:
        cfg = self.cfg
        self.network.train()
        epoch = 0
        n_total = len(train_loader.dataset) if hasattr(train_loader, 'dataset') else 0
        while self.cur_step < cfg.max_steps:
            for batch_idx, (data, target) in enumerate(train_loader):
                if self.cur_step >= cfg.max_steps:
                    return
                iter_start = time.time()
                data = data.to(self.device, self.compute_dtype)
                target = target.to(self.device)
                t0 = time.time()
                self.fetch_weights()
                fetch_dur = time.time() - t0
                self.flat.zero_grads()
                self._reset_bucket_state()
                loss, killed = self._forward_backward(data, target)
                t0 = time.time()
                self.push_gradients(killed=killed)
                if self.ctrl is not None:
                    self.ctrl.finish()
                comm_dur = time.time() - t0
                self.cur_step += 1
                if self.cur_step % cfg.log_interval == 0:
                    self._resolve_phase_timing()
                    logger.info(worker_log_line(
                        self.rank, self.cur_step, epoch,
                        batch_idx * cfg.batch_size, n_total,
                        float('nan') if loss is None else float(loss),
                        time.time() - iter_start, fetch_dur, self.f_dur,
                        self.b_dur, comm_dur))
                # checkpoint division of labor mirrors the reference
                # (distributed_worker.py:175-177): rank 1 saves nets with BN
                # buffers (ResNet/VGG) so running stats come from a worker.
                if (self.cur_step % cfg.eval_freq == 0 and self.rank == 1 and
                        any(k in cfg.network.lower() for k in ('resnet', 'vgg'))):
                    save_model_step(self.network, cfg.train_dir, self.cur_step)
            epoch += 1
