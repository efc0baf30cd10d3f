"""RCCL transport for the synchronous PS protocol.

MI355X-native re-expression of the reference's mpi4py wire protocol
(SURVEY.md §2.4): the per-layer tagged Isend/Irecv/bcast calls become
bucketed collectives on ONE flat buffer over `torch.distributed`
(backend "nccl" IS RCCL on ROCm; "gloo" for CPU tests):

  step announce (tag 10)         -> implicit in the weight broadcast
  weight bcast  (tag 11+l, x L)  -> ONE ncclBroadcast of the flat wire buffer
  grad push     (tag 88+l, x LW) -> per-bucket ncclReduce(sum) to rank 0,
                                    issued in bucket order (matching by
                                    program order replaces tag matching)
  kill (tag 77) / control        -> small gloo side-channel (RCCL collectives
                                    cannot be probed or cancelled)

Weight broadcast from rank 0 can use all 7 outgoing xGMI links (every peer
is one hop); gradient fan-in as reduce-to-root is bound by the PS's 7
incoming links (~1 TB/s aggregate) — see SURVEY.md §5.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

from .flat import Bucket, FlatSpace
from ..ops import functional as ops_f

PS_RANK = 0


def init_distributed(backend: Optional[str] = None,
                     device: Optional[torch.device] = None) -> dict:
    """Init from torchrun env (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT)."""
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local_rank = int(os.environ.get('LOCAL_RANK', str(rank)))
    if backend is None:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    if device is None:
        if backend == 'nccl':
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
            device = torch.device('cuda', local_rank % torch.cuda.device_count())
        else:
            device = torch.device('cpu')
    if world > 1 and not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=300))
    return {'rank': rank, 'world': world, 'local_rank': local_rank,
            'backend': backend, 'device': device}


class PSTransport:
    """Bucketed flat-buffer transport between 1 PS (rank 0) and W-1 workers.

    Owns the wire buffers:
      wire_w : flat weights as broadcast payload (wire dtype)
      wire_g : flat gradient accumulator in wire dtype (the reduce target on
               the PS; the reduce source on workers)
    """

    def __init__(self, flat: FlatSpace, wire_dtype: torch.dtype,
                 device: torch.device, rank: int, world: int,
                 group: Optional[dist.ProcessGroup] = None):
        self.flat = flat
        self.rank = rank
        self.world = world
        self.group = group
        self.device = device
        self.wire_dtype = wire_dtype
        n = flat.padded
        self.wire_w = torch.zeros(n, dtype=wire_dtype, device=device)
        self.wire_g = torch.zeros(n, dtype=wire_dtype, device=device)
        self._works: List[dist.Work] = []

    @property
    def num_workers(self) -> int:
        return self.world - 1

    # ---- weights: PS -> all ----

    def broadcast_weights(self) -> None:
        dist.broadcast(self.wire_w, src=PS_RANK, group=self.group)

    def unpack_weights_into(self, dst: torch.Tensor) -> None:
        """Worker side: wire payload -> live flat params (cast if needed)."""
        ops_f.pack_wire(dst, self.wire_w)

    def pack_weights_from(self, src: torch.Tensor) -> None:
        """PS side: master f32 -> wire payload (usually fused into the
        update kernel via wire_out; this explicit path covers step 0)."""
        ops_f.pack_wire(self.wire_w, src)

    # ---- gradients: workers -> PS (sum), bucketed, async ----

    def grad_wire_slice(self, b: Bucket) -> torch.Tensor:
        return self.wire_g[b.start:b.end]

    def push_bucket(self, b: Bucket) -> dist.Work:
        """Worker: pack this bucket's grads to wire dtype and reduce to PS.
        Must be called in bucket-index order on every rank (program-order
        matching)."""
        src = self.flat.grad_slice(b)
        wire = self.grad_wire_slice(b)
        if wire.dtype != src.dtype:
            ops_f.pack_wire(wire, src)
        else:
            wire.copy_(src)
        w = dist.reduce(wire, dst=PS_RANK, op=dist.ReduceOp.SUM,
                        group=self.group, async_op=True)
        self._works.append(w)
        return w

    def recv_buckets(self, buckets: List[Bucket]) -> None:
        """PS: contribute zeros and post all reduces (async, in order)."""
        self.wire_g.zero_()
        for b in buckets:
            w = dist.reduce(self.grad_wire_slice(b), dst=PS_RANK,
                            op=dist.ReduceOp.SUM, group=self.group,
                            async_op=True)
            self._works.append(w)

    def wait_all(self) -> None:
        for w in self._works:
            w.wait()
        self._works.clear()

    def barrier(self) -> None:
        if dist.is_initialized():
            dist.barrier(group=self.group)
