"""RCCL transport for the synchronous PS protocol.

MI355X-native re-expression of the reference's mpi4py wire protocol
(SURVEY.md §2.4): the per-layer tagged Isend/Irecv/bcast calls become
bucketed collectives on ONE flat buffer over `torch.distributed`
(backend "nccl" IS RCCL on ROCm; "gloo" for CPU tests):

  step announce (tag 10)         -> implicit in the weight broadcast
  weight bcast  (tag 11+l, x L)  -> ONE ncclBroadcast of the flat wire buffer
  grad push     (tag 88+l, x LW) -> two aggregation modes:
     'collective': per-bucket ncclReduce(sum) to rank 0, issued in bucket
        order (program-order matching replaces tags); xGMI sums in flight.
     'gather': per-bucket ncclSend to rank 0 into per-worker staging
        buffers; the PS accumulates IN ARRIVAL ORDER with HIP kernels
        (ps_acc / ps_unpack_q8 acc) — this restores the reference's
        Waitany-drain semantics (sync_replicas_master_nn.py:157-186) and
        enables --num-aggregate first-k selection and the compressed
        payload (summed payloads can't ride an in-flight reduction).
  compression                    -> gather mode sends block-scaled int8
        payloads (ops/kernels/quant.hip, 4x smaller than f32) — the GPU
        re-expression of blosc g_compress (ref compression.py:18-46).
  kill (tag 77) / control        -> gloo side-channel (ControlPlane below):
        RCCL collectives cannot be probed or cancelled, so the straggler
        kill signal travels host-side while payload matching stays intact
        (killed workers send zero payloads for their remaining buckets).

Weight broadcast from rank 0 can use all 7 outgoing xGMI links (every peer
is one hop); gradient fan-in (either mode) is bound by the PS's 7
incoming links (~1 TB/s aggregate) — see SURVEY.md §5.
"""
from __future__ import annotations

import datetime
import os
import time
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
    Gather mode adds:
      payload   (worker) / stages[w] (PS) : per-bucket P2P gradient payloads
               (q8-compressed bytes, or wire-dtype values)
      acc_g (PS) : f32 arrival-order accumulator consumed by the optimizer
    """

    def __init__(self, flat: FlatSpace, wire_dtype: torch.dtype,
                 device: torch.device, rank: int, world: int,
                 group: Optional[dist.ProcessGroup] = None,
                 mode: str = 'collective', compress: bool = False,
                 comm_type: str = 'Bcast', track_killed: bool = False):
        if mode not in ('collective', 'gather'):
            raise ValueError(f"unknown aggregation mode {mode!r}")
        self.comm_type = comm_type
        self.flat = flat
        self.rank = rank
        self.world = world
        self.group = group
        self.device = device
        self.wire_dtype = wire_dtype
        self.mode = mode
        self.compress = bool(compress) and mode == 'gather'
        # liveness bound for the PS fan-in (a crashed worker otherwise hangs
        # the drain forever — the reference shares this property, but an
        # 8-GPU production run deserves a loud error over a silent hang)
        self.drain_timeout = float(os.environ.get('PS_DRAIN_TIMEOUT', '600'))
        n = flat.padded
        self.wire_w = torch.zeros(n, dtype=wire_dtype, device=device)
        self.wire_g = torch.zeros(n, dtype=wire_dtype, device=device)
        self._works: List[dist.Work] = []
        self._bcast_works: List[dist.Work] = []
        # program-order matching guard: NCCL IGNORES P2P tags — matching of
        # the gather-mode isend/irecv pairs rests entirely on per-pair FIFO
        # order, so every rank MUST touch buckets in index order. This
        # counter turns a violation into an assert instead of a silent
        # payload mismatch. (The tag= arguments below still matter for the
        # gloo test backend, which does honor them.)
        self._next_push = 0
        # timeout-mode arrival marking: a locally-aborted worker's zero
        # payloads must not occupy --num-aggregate quota slots (the
        # reference's first-k only counts real arrivals). Workers send a
        # 1-byte killed flag per step on a gloo side-group; the PS excludes
        # flagged workers from the quota count.
        self.track_killed = bool(track_killed) and mode == 'gather'
        if self.track_killed and dist.is_initialized():
            self._flag_group = dist.new_group(backend='gloo')
            self._flag_buf = torch.zeros(1, dtype=torch.uint8)
            if rank == PS_RANK:
                self._flag_bufs = {w: torch.zeros(1, dtype=torch.uint8)
                                   for w in range(1, world)}
        else:
            self.track_killed = False

        if mode == 'gather':
            if self.compress:
                # per-bucket q8 payload framing inside ONE byte buffer
                self._pl_off: List[tuple] = []
                cur = 0
                for b in flat.buckets:
                    _, tot = ops_f.q8_layout(b.numel)
                    self._pl_off.append((cur, tot))
                    cur += tot
                self.payload_bytes = cur
                if rank == PS_RANK:
                    self.stages = [torch.zeros(cur, dtype=torch.uint8, device=device)
                                   for _ in range(world - 1)]
                else:
                    self.payload = torch.zeros(cur, dtype=torch.uint8, device=device)
                    self._zero_payload = torch.zeros(
                        max(t for _, t in self._pl_off), dtype=torch.uint8,
                        device=device)
            else:
                if rank == PS_RANK:
                    self.stages = [torch.zeros(n, dtype=wire_dtype, device=device)
                                   for _ in range(world - 1)]
                else:
                    self._zero_payload = torch.zeros(
                        max(b.numel for b in flat.buckets), dtype=wire_dtype,
                        device=device)
            if rank == PS_RANK:
                self.acc_g = torch.zeros(n, dtype=torch.float32, device=device)
            self._pending: dict = {}     # (bucket_idx, worker) -> Work
            if dist.is_initialized() and dist.get_backend(group) == 'nccl':
                self._warmup_p2p()
        if dist.is_initialized() and world > 1:
            self._verify_layout()

    def _verify_layout(self) -> None:
        """All ranks must agree on the flat layout — bucket framing IS the
        wire protocol (no per-message tags to catch a mismatch). Rank 0
        broadcasts a fingerprint of FlatSpace.layout_signature(); a
        divergent rank raises instead of silently mis-framing payloads."""
        import zlib
        sig = zlib.crc32(repr(self.flat.layout_signature()).encode())
        dev = (self.device if dist.get_backend(self.group) == 'nccl'
               else torch.device('cpu'))
        t = torch.tensor([sig], dtype=torch.int64, device=dev)
        mine = int(t)
        dist.broadcast(t, src=PS_RANK, group=self.group)
        if int(t) != mine:
            raise RuntimeError(
                f"rank {self.rank}: flat-layout fingerprint {mine:#x} does "
                f"not match rank 0's {int(t):#x} — model/config divergence; "
                f"bucket framing would corrupt payloads")

    def _warmup_p2p(self) -> None:
        """Open every PS<->worker NCCL P2P channel in one deterministic
        global order before the first real step (lazy per-pair communicator
        creation in mismatched orders can deadlock)."""
        t = torch.zeros(1, device=self.device)
        for w in range(1, self.world):
            if self.rank == PS_RANK:
                dist.recv(t, src=w, group=self.group)
            elif self.rank == w:
                dist.send(t, dst=PS_RANK, group=self.group)
        dist.barrier(group=self.group)

    @property
    def num_workers(self) -> int:
        return self.world - 1

    # ---- weights: PS -> all ----

    # Pipelined per-bucket update/broadcast (collective mode, Bcast
    # comm_type): the PS waits each bucket's fan-in, runs the fused update
    # on that slice, and broadcasts it immediately — the fan-in tail of
    # later buckets and the whole optimizer step hide under wire time.
    # Per-rank collective order stays identical on every rank:
    #   worker: [bcast(k) x B, reduce(k) x B, bcast(k+1) x B, ...]
    #   PS:     [bcast(0,init) x B, reduce(0) x B, bcast(0,tail) x B, ...]
    # (the PS skips the tail broadcast of the final step — no dangling
    # collective for workers that have exited their loop).
    @property
    def bcast_bucketed(self) -> bool:
        return (self.mode == 'collective' and self.comm_type == 'Bcast'
                and os.environ.get('PS_BCAST_PIPE', '1') != '0')

    def bcast_bucket(self, b: Bucket) -> None:
        self._bcast_works.append(
            dist.broadcast(self.wire_w[b.start:b.end], src=PS_RANK,
                           group=self.group, async_op=True))

    def bcast_all_buckets(self) -> None:
        for b in self.flat.buckets:
            self.bcast_bucket(b)

    def wait_bcasts(self) -> None:
        for w in self._bcast_works:
            w.wait()
        self._bcast_works.clear()

    def wait_reduce(self, i: int) -> None:
        """Wait the i-th posted reduce (collective-mode PS pipeline)."""
        self._works[i].wait()

    def broadcast_weights(self) -> None:
        if self.rank != PS_RANK and self.bcast_bucketed:
            self.bcast_all_buckets()
            self.wait_bcasts()
            return
        if self.comm_type == 'Async':
            # reference's deprecated P2P weight distribution
            # (distributed_worker.py:201-219): explicit PS->worker sends
            # instead of the one-collective broadcast.
            if self.rank == PS_RANK:
                works = [dist.isend(self.wire_w, dst=w, group=self.group)
                         for w in range(1, self.world)]
                for wk in works:
                    wk.wait()
            else:
                dist.recv(self.wire_w, src=PS_RANK, group=self.group)
            return
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

    def push_bucket(self, b: Bucket, killed: bool = False) -> dist.Work:
        """Worker: move this bucket's grads toward the PS. Must be called in
        bucket-index order on every rank (program-order matching).

        collective: pack to wire dtype, ncclReduce(sum) to rank 0.
        gather:     pack (q8 or wire dtype), ncclSend to rank 0.
        killed=True sends a zero payload without touching the grads (the
        straggler-abort path: matching stays intact, compute is skipped)."""
        # NCCL ignores tags: matching is per-pair FIFO, so bucket order is
        # the protocol. Enforce it (reset in wait_all at step end).
        assert b.index == self._next_push, (
            f"bucket push out of order: got {b.index}, expected "
            f"{self._next_push} (per-pair FIFO matching would corrupt)")
        self._next_push += 1
        if self.mode == 'collective':
            src = self.flat.grad_slice(b)
            wire = self.grad_wire_slice(b)
            if killed:
                wire.zero_()
            elif wire.dtype != src.dtype:
                ops_f.pack_wire(wire, src)
            else:
                wire.copy_(src)
            w = dist.reduce(wire, dst=PS_RANK, op=dist.ReduceOp.SUM,
                            group=self.group, async_op=True)
        elif self.compress:
            if killed:
                payload = self._zero_payload[:self._pl_off[b.index][1]]
            else:
                payload = self._worker_payload_slice(b)
                ops_f.pack_q8(payload, self.flat.grad_slice(b))
            w = dist.isend(payload, dst=PS_RANK, group=self.group, tag=b.index)
        else:
            if killed:
                wire = self._zero_payload[:b.numel]
            else:
                src = self.flat.grad_slice(b)
                wire = self.grad_wire_slice(b)
                if wire.dtype != src.dtype:
                    ops_f.pack_wire(wire, src)
                else:
                    wire.copy_(src)
            w = dist.isend(wire, dst=PS_RANK, group=self.group, tag=b.index)
        self._works.append(w)
        return w

    def _worker_payload_slice(self, b: Bucket) -> torch.Tensor:
        off, nbytes = self._pl_off[b.index]
        return self.payload[off:off + nbytes]

    def _stage_slice(self, worker: int, b: Bucket) -> torch.Tensor:
        """PS: worker w's staging region for bucket b (w is 1-based rank)."""
        st = self.stages[worker - 1]
        if self.compress:
            off, nbytes = self._pl_off[b.index]
            return st[off:off + nbytes]
        return st[b.start:b.end]

    # -- collective mode, PS side --

    def recv_buckets(self, buckets: List[Bucket]) -> None:
        """PS: contribute zeros and post all reduces (async, in order)."""
        self.wire_g.zero_()
        for b in buckets:
            w = dist.reduce(self.grad_wire_slice(b), dst=PS_RANK,
                            op=dist.ReduceOp.SUM, group=self.group,
                            async_op=True)
            self._works.append(w)

    # -- gather mode, PS side --

    def post_gather_recvs(self) -> None:
        """PS: post one irecv per (bucket, worker). Posting order is
        bucket-major to mirror workers' send order per peer."""
        assert self.rank == PS_RANK and self.mode == 'gather'
        for b in self.flat.buckets:
            for w in range(1, self.world):
                self._pending[(b.index, w)] = dist.irecv(
                    self._stage_slice(w, b), src=w, group=self.group,
                    tag=b.index)

    def drain_arrivals(self, num_aggregate: int, on_quota=None) -> List[int]:
        """PS: poll posted recvs; accumulate the FIRST `num_aggregate`
        arrivals per bucket into acc_g (f32) in arrival order — the
        reference's Waitany-drain (sync_replicas_master_nn.py:157-186) with
        its --num-aggregate first-k selection (:179-207). Later arrivals are
        drained (matching) but discarded. `on_quota()` fires once, the
        moment every bucket has its quota while some recvs are still in
        flight (the straggler-kill trigger). Returns per-bucket counts.

        With track_killed (timeout mode), a worker that aborted locally sent
        a killed=1 flag on the gloo side-group; its zero payloads are drained
        for matching but excluded from the quota count — the reference's
        first-k only ever counts real gradients."""
        assert self.rank == PS_RANK and self.mode == 'gather'
        buckets = self.flat.buckets
        k = max(1, min(num_aggregate, self.world - 1))
        self.acc_g.zero_()
        counts = [0] * len(buckets)
        left = {w: len(buckets) for w in range(1, self.world)}
        killed_of: dict = {}
        quota_fired = False
        for bi, w in self._arrivals():
            left[w] -= 1
            if self.track_killed and w not in killed_of:
                killed_of[w] = self._recv_killed_flag(w)
            if counts[bi] < k and not killed_of.get(w, False):
                b = buckets[bi]
                if self.compress:
                    ops_f.unpack_q8(self.acc_g[b.start:b.end],
                                    self._stage_slice(w, b), accumulate=True)
                else:
                    ops_f.acc_into(self.acc_g[b.start:b.end],
                                   self._stage_slice(w, b))
                counts[bi] += 1
            if (on_quota is not None and not quota_fired
                    and all(c >= k for c in counts)):
                quota_fired = True
                on_quota(sorted(w for w, n in left.items() if n > 0))
        if on_quota is not None and not quota_fired:
            on_quota([])
        return counts

    # -- killed-flag side channel (timeout mode, gather) --

    def send_killed_flag(self, killed: bool) -> None:
        """Worker: one flag byte per step (non-blocking — the PS only recvs
        it lazily at this worker's first arrival, so a blocking send here
        before the bucket pushes would deadlock; completion joins wait_all)."""
        if not self.track_killed or self.rank == PS_RANK:
            return
        self._flag_buf.fill_(1 if killed else 0)
        self._works.append(dist.isend(self._flag_buf, dst=PS_RANK,
                                      group=self._flag_group, tag=78))

    def _recv_killed_flag(self, w: int) -> bool:
        """PS: consume worker w's flag for this step (exactly one per step;
        blocks at most until w finishes its pushes — it sends the flag right
        after launching them, or immediately upon aborting)."""
        buf = self._flag_bufs[w]
        dist.recv(buf, src=w, group=self._flag_group, tag=78)
        return bool(int(buf))

    def _arrivals(self):
        """Yield (bucket_idx, worker) in completion order, draining
        self._pending. NCCL work objects expose real event-backed
        is_completed() — poll them. Gloo only materializes completion
        inside wait(), so one waiter thread per worker blocks through that
        worker's buckets in FIFO send order and feeds a completion queue
        (cross-worker arrival order is what the queue observes)."""
        deadline = time.monotonic() + self.drain_timeout
        if dist.get_backend(self.group) == 'nccl':
            idle_spins = 0
            while self._pending:
                done = [key for key, wk in self._pending.items()
                        if wk.is_completed()]
                if not done:
                    # back off after a few hot spins: pinning a host core for
                    # the whole fan-in window buys nothing once the first
                    # poll came back empty; 50 us keeps drain
                    # latency noise well under a bucket's wire time.
                    idle_spins += 1
                    time.sleep(0 if idle_spins < 64 else 50e-6)
                    if time.monotonic() > deadline:
                        self._drain_timeout_error()
                    continue
                idle_spins = 0
                for key in done:
                    self._pending.pop(key)
                    yield key
            return
        import queue as _queue
        import threading
        q: _queue.Queue = _queue.Queue()
        by_worker: dict = {}
        for (bi, w), wk in sorted(self._pending.items()):
            by_worker.setdefault(w, []).append((bi, wk))
        def waiter(w, items):
            for bi, wk in items:
                wk.wait()
                q.put((bi, w))
        threads = [threading.Thread(target=waiter, args=(w, items), daemon=True)
                   for w, items in by_worker.items()]
        for t in threads:
            t.start()
        total = len(self._pending)
        self._pending.clear()
        for i in range(total):
            try:
                yield q.get(timeout=max(0.001, deadline - time.monotonic()))
            except _queue.Empty:
                self._pending = {(i, -1): None}   # non-empty for the error msg
                self._drain_timeout_error(total - i)
        for t in threads:
            t.join()

    def _drain_timeout_error(self, missing: Optional[int] = None):
        who = (sorted(self._pending.keys()) if missing is None
               else f"{missing} recvs")
        raise RuntimeError(
            f"PS gradient fan-in stalled > {self.drain_timeout:.0f}s "
            f"(PS_DRAIN_TIMEOUT); outstanding (bucket, worker) recvs: {who}. "
            f"A worker likely crashed — check its log; partial aggregation "
            f"(--num-aggregate) still requires every worker to SEND (zeros "
            f"when aborted), matching the reference's Waitany drain.")

    def wait_all(self) -> None:
        for w in self._works:
            w.wait()
        self._works.clear()
        self._next_push = 0

    def barrier(self) -> None:
        """Device-collective barrier on the MAIN group. Do NOT call this
        between a pipelined tail broadcast (PS) and the workers' next fetch
        — per-rank collective order would diverge and deadlock NCCL. Use a
        gloo side-group for host-side synchronization (bench.py does)."""
        if dist.is_initialized():
            dist.barrier(group=self.group)


class StepKilled(Exception):
    """Raised inside a worker's backward to abort the rest of the step
    (the reference's mid-backward straggler abort,
    resnet_split.py:503-615 backward_signal_kill)."""


class ControlPlane:
    """Host-side gloo channel for the straggler-kill protocol.

    The reference polls MPI tag 77 between layers of its hand-unrolled
    backward (resnet_split.py:513-523 Iprobe). RCCL kernels can't be probed
    or cancelled, so the signal travels on a gloo side-group: the PS sends
    EXACTLY ONE byte to every worker per step — 1 as soon as every bucket
    hit its --num-aggregate quota while that worker still has sends in
    flight ("abort your remaining backward"), else 0 at step end. Workers
    post the matching irecv at step start and poll it from autograd hooks;
    the one-send/one-recv-per-step pairing keeps matching stall-free and
    stale-signal-proof (no step-stamped tags needed: FIFO matching + the
    end-of-step wait ensure a message can only belong to the current step,
    the property resnet_split.py:25-42 generate_tag emulated)."""

    KILL_TAG = 77

    def __init__(self, rank: int, world: int):
        self.rank = rank
        self.world = world
        self.group = dist.new_group(backend='gloo')
        if rank == PS_RANK:
            self._bufs = [torch.zeros(1, dtype=torch.uint8)
                          for _ in range(world)]
        else:
            self._buf = torch.zeros(1, dtype=torch.uint8)
            self._work: Optional[dist.Work] = None

    # -- PS side --

    def signal(self, kill_ranks) -> None:
        """Send this step's verdict to every worker (1 = abort rest)."""
        kill = set(kill_ranks)
        works = []
        for w in range(1, self.world):
            self._bufs[w].fill_(1 if w in kill else 0)
            works.append(dist.isend(self._bufs[w], dst=w, group=self.group,
                                    tag=self.KILL_TAG))
        for wk in works:
            wk.wait()

    # -- worker side --
    # gloo only materializes completion inside wait() (is_completed() stays
    # false), so a waiter thread blocks on the verdict and flips an Event
    # the hooks can poll without blocking.

    def post(self) -> None:
        import threading
        self._buf.zero_()
        work = dist.irecv(self._buf, src=PS_RANK, group=self.group,
                          tag=self.KILL_TAG)
        self._evt = threading.Event()

        def waiter(wk, evt):
            wk.wait()
            evt.set()

        self._thread = threading.Thread(target=waiter, args=(work, self._evt),
                                        daemon=True)
        self._thread.start()

    def killed(self) -> bool:
        """Non-blocking poll (called from backward hooks)."""
        return (getattr(self, '_evt', None) is not None
                and self._evt.is_set() and int(self._buf) == 1)

    def finish(self) -> None:
        """End of step: consume this step's verdict (always arrives while
        the PS lives). Bounded join — a dead PS otherwise hangs every
        worker here forever (same liveness policy as PS_DRAIN_TIMEOUT)."""
        if getattr(self, '_thread', None) is not None:
            limit = float(os.environ.get('PS_CTRL_TIMEOUT', '600'))
            self._thread.join(limit)
            if self._thread.is_alive():
                raise RuntimeError(
                    f"rank {self.rank}: no straggler verdict from the PS in "
                    f"{limit:.0f}s (PS_CTRL_TIMEOUT) — PS likely crashed")
            self._thread = None
            self._evt = None
