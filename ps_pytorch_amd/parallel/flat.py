"""Flat contiguous parameter/gradient storage with bucket framing.

This is the MI355X-native replacement for the reference's per-(layer, worker)
tagged-message protocol (ref: sync_replicas_master_nn.py:199-237 posts
L x (W-1) Irecvs with tag 88+layer; distributed_worker.py:254-272 sends one
message per layer). Instead, every rank lays out all parameters in ONE flat
buffer in REVERSE parameter order (the approximate order backward produces
gradients), partitions it into contiguous buckets, and moves whole buckets
with RCCL collectives — the bucket id <-> offset table is agreed at init
(deterministic bucket framing; SURVEY.md §7 "Hard parts").

Reverse-order layout means bucket 0 holds the LAST layers' params, i.e. the
first gradients ready during backward — so overlapped per-bucket reduction
streams in backward order with zero gather/scatter copies (grads are views
into the flat buffer; collectives operate on contiguous slices).
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional, Sequence

import torch
import torch.nn as nn


def prep_model(net: nn.Module, device: torch.device,
               dtype: torch.dtype) -> nn.Module:
    """Move a model to its compute placement, MI355X-style: on GPU, conv nets
    go channels_last end-to-end (NHWC is the native layout for both the
    MIOpen MFMA conv kernels and our fused BN kernels — the eager NCHW path
    spent 10% of step time in batched_transpose, profiles/r01_step01)."""
    net = net.to(device=device, dtype=dtype)
    if device.type == 'cuda' and any(isinstance(m, nn.Conv2d)
                                     for m in net.modules()):
        net = net.to(memory_format=torch.channels_last)
        # fixed shapes every step: let MIOpen search once for the best
        # MFMA conv kernel per shape
        torch.backends.cudnn.benchmark = True
    return net


@dataclasses.dataclass
class Bucket:
    index: int
    start: int          # element offset into the flat buffer
    end: int            # exclusive
    param_ids: List[int]  # indices into FlatSpace.params covered by this bucket

    @property
    def numel(self) -> int:
        return self.end - self.start


class FlatSpace:
    """Owns flat weight/grad storage for a model; params become views.

    - `flat_w`: one contiguous tensor holding every parameter (reverse order).
    - `flat_g`: matching gradient buffer; `attach_grads()` points every
      `p.grad` at its slice so autograd accumulates in place — the "pack"
      step of the reference (GPU->CPU copy + blosc, distributed_worker.py:259)
      becomes a no-op.
    - Buckets partition [0, total) contiguously, each <= bucket_bytes except
      when a single parameter exceeds it (then the bucket grows to hold it:
      parameters never straddle buckets).
    """

    def __init__(self, model: nn.Module, bucket_bytes: int = 25 * 1024 * 1024,
                 dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None):
        named = list(model.named_parameters())
        self.names: List[str] = [n for n, _ in named][::-1]     # reverse order
        self.params: List[nn.Parameter] = [p for _, p in named][::-1]
        if not self.params:
            raise ValueError("model has no parameters")
        p0 = self.params[0]
        self.dtype = dtype or p0.dtype
        self.device = device or p0.device
        # Parameter offsets are aligned up to ALIGN elements so every bucket
        # boundary (always a param boundary, see _partition) is 16-B aligned
        # even for bf16 storage: the gather-mode HIP kernels (ps_acc,
        # ps_pack_q8, ps_unpack_q8) issue 8/16-byte vector accesses relative
        # to the bucket start. Pad gaps live inside buckets; their grads stay
        # zero and their weights are broadcast as-is (dead bytes on the wire,
        # < ALIGN elems per param).
        ALIGN = 8
        self.offsets: List[int] = []
        total = 0
        for p in self.params:
            total = (total + ALIGN - 1) & ~(ALIGN - 1)
            self.offsets.append(total)
            total += p.numel()
        self.total = total
        self.num_params = sum(p.numel() for p in self.params)
        # pad storage to a multiple of ALIGN: full-buffer HIP kernels
        # (fused_sgd) run purely 16B-vectorized; pad grads stay zero.
        self.padded = (total + ALIGN - 1) & ~(ALIGN - 1)

        # Per-param storage layout: 4D channels_last params are stored in
        # their (O,H,W,I) storage order and re-exposed as strided views, so
        # the flat buffer IS the wire/kernel layout with zero copies. A param
        # is treated channels_last iff its incoming data is (and is not plain
        # contiguous, e.g. 1x1 kernels which are both).
        self._cl: List[bool] = [
            (p.data.dim() == 4
             and p.data.is_contiguous(memory_format=torch.channels_last)
             and not p.data.is_contiguous())
            for p in self.params]

        self.flat_w = torch.zeros(self.padded, dtype=self.dtype, device=self.device)
        # Re-home every parameter as a view of flat_w (keeps autograd/optimizer
        # identity: p is still the same nn.Parameter object).
        for pid, (p, off) in enumerate(zip(self.params, self.offsets)):
            self._view(self.flat_w, p, off, pid).copy_(p.data.detach().to(self.dtype))
            p.data = self._view(self.flat_w, p, off, pid)
        self.flat_g = torch.zeros(self.padded, dtype=self.dtype, device=self.device)

        self.buckets = self._partition(bucket_bytes)

    def _partition(self, bucket_bytes: int) -> List[Bucket]:
        """Contiguous buckets at ALIGNED boundaries: a bucket ends at the
        next param's aligned offset (the alignment gap travels inside the
        bucket), and the last bucket ends at `padded` — so every bucket
        slice start/size is 16-B aligned for the vectorized HIP kernels."""
        elem = self.flat_w.element_size()
        max_elems = max(1, bucket_bytes // elem)
        nparams = len(self.params)
        buckets: List[Bucket] = []
        cur_ids: List[int] = []
        cur_start = 0
        for i, (p, off) in enumerate(zip(self.params, self.offsets)):
            end = self.offsets[i + 1] if i + 1 < nparams else self.padded
            cur_ids.append(i)
            if end - cur_start >= max_elems:
                buckets.append(Bucket(len(buckets), cur_start, end, cur_ids))
                cur_ids = []
                cur_start = end
        if cur_ids:
            buckets.append(Bucket(len(buckets), cur_start, self.padded, cur_ids))
        return buckets

    def _view(self, flat: torch.Tensor, p: nn.Parameter, off: int,
              pid: int) -> torch.Tensor:
        """Slice of `flat` shaped/strided like p (honoring channels_last)."""
        sl = flat[off:off + p.numel()]
        if self._cl[pid]:
            O, I, H, W = p.shape
            return sl.view(O, H, W, I).permute(0, 3, 1, 2)
        return sl.view(p.shape)

    # ---- gradient plumbing ----

    def attach_grads(self, steal: bool = False) -> None:
        """Wire autograd output into flat_g.

        view mode (default, CPU/torch ops): p.grad = flat_g view; autograd
        ACCUMULATES into it (one at::add kernel per param per step).

        steal mode (GPU kernel path): p.grad starts None and every Ps op's
        backward WRITES its weight/bias grad directly into the param's
        flat_g slice (published here as p._ps_flat_grad) and returns that
        view — AccumulateGrad then steals instead of adding, removing ~60
        elementwise add kernels per ResNet-18 step. Ops that fell back to
        torch produce a foreign tensor; ensure_grad_in_flat() detects that
        by pointer and copies, so kill-switches (PS_CONV=0 …) stay correct.
        """
        self._steal = steal
        self._gptrs = []
        elem = self.flat_g.element_size()
        for pid, (p, off) in enumerate(zip(self.params, self.offsets)):
            self._gptrs.append(self.flat_g.data_ptr() + off * elem)
            # a FACTORY, not a held view: AccumulateGrad only steals a grad
            # whose use_count is 1 at accumulation time — any live Python
            # reference to the view forces a clone (one copyBuffer per param
            # per step, measured). Ops call this per backward and let the
            # fresh view die with the autograd edge.
            if steal:
                p._ps_flat_grad_fn = self._view_factory(pid)
                p.grad = None
            else:
                # view mode: autograd accumulates INTO the view, so ops must
                # NOT also write the slice (p.grad += view-of-itself = 2x)
                if hasattr(p, '_ps_flat_grad_fn'):
                    del p._ps_flat_grad_fn
                p.grad = self._view(self.flat_g, p, off, pid)

    def zero_grads(self) -> None:
        self.flat_g.zero_()
        if getattr(self, '_steal', False):
            for p in self.params:
                p.grad = None

    def _view_factory(self, pid: int):
        flat_g, off, p = self.flat_g, self.offsets[pid], self.params[pid]
        shape, cl = tuple(p.shape), self._cl[pid]
        numel = p.numel()

        def make():
            sl = flat_g[off:off + numel]
            if cl:
                O, I, H, W = shape
                return sl.view(O, H, W, I).permute(0, 3, 1, 2)
            return sl.view(shape)
        return make

    def ensure_grad_in_flat(self, pid: int) -> None:
        """Steal mode: if this param's grad landed outside flat_g (a torch
        fallback op), copy it into the slice. No-op when the Ps kernels
        produced it in place (pointer match)."""
        if not getattr(self, '_steal', False):
            return
        p = self.params[pid]
        g = p.grad
        if g is None or g.data_ptr() == self._gptrs[pid]:
            return
        if not getattr(self, '_warned_foreign', False):
            self._warned_foreign = True
            import warnings
            warnings.warn("grad for a parameter landed outside flat_g "
                          "(torch-fallback op?); copying per step")
        self._view(self.flat_g, p, self.offsets[pid], pid).copy_(g)

    def harvest_grads(self) -> None:
        """Steal mode: make sure every param's grad is in flat_g."""
        for pid in range(len(self.params)):
            self.ensure_grad_in_flat(pid)

    def grad_slice(self, b: Bucket) -> torch.Tensor:
        return self.flat_g[b.start:b.end]

    def weight_slice(self, b: Bucket) -> torch.Tensor:
        return self.flat_w[b.start:b.end]

    # ---- (de)serialization helpers ----

    def load_flat(self, src: torch.Tensor) -> None:
        """Copy a flat f32/bf16 vector (same layout) into the live params."""
        self.flat_w.copy_(src.to(self.dtype))

    def state_dict_from_flat(self, flat: Optional[torch.Tensor] = None) -> dict:
        """Reconstruct a {name: tensor} mapping (parameters only) from a flat
        vector in THIS space's layout — used for model_step_<k> checkpoints."""
        flat = self.flat_w if flat is None else flat
        out = {}
        for pid, (name, p, off) in enumerate(zip(self.names, self.params,
                                                 self.offsets)):
            out[name] = self._view(flat, p, off, pid).detach().clone()
        return out

    def layout_signature(self) -> Sequence:
        """Deterministic layout descriptor; ranks compare it at init to agree
        on bucket framing (replaces the reference's per-layer tag contract)."""
        return [(n, tuple(p.shape), off, cl) for n, p, off, cl in
                zip(self.names, self.params, self.offsets, self._cl)]
