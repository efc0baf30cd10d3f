"""PsConv2d — NHWC implicit-GEMM MFMA convolution (ops/kernels/conv.hip).

Replaces the MIOpen path the reference reaches through nn.Conv2d
(ref: src/model_ops/resnet.py:19-97, lenet.py:19-33). GPU training path is
the hand-written CDNA4 kernel set (fwd / dgrad / split-K wgrad, fp32
accumulate, deterministic reduce); CPU and unsupported shapes fall back to
torch's conv, which doubles as the numerics reference in tests.

PsConv2d subclasses nn.Conv2d, so parameter shapes, init and the
state_dict surface are identical (evaluator/checkpoint compatible).
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import require_lib, current_stream_ptr

_CL = torch.channels_last

# A/B kill-switch: PS_CONV=0 routes every conv through torch/MIOpen.
_ENABLED = os.environ.get('PS_CONV', '1') != '0'

# BN-stats handshake sidechannel: _ConvFn.forward's epilogue may produce
# per-channel sum/sumsq partials; apply() re-wraps the output tensor, so
# the Function stashes them here and the module wrapper attaches them to
# the wrapped output (sequential per caller thread — no races).
_PENDING_STATS = None
_BN_FUSE = os.environ.get('PS_BN_FUSE', '0') == '1'


def _pop_stats():
    global _PENDING_STATS
    st, _PENDING_STATS = _PENDING_STATS, None
    return st


def _supported(x: torch.Tensor, w: torch.Tensor, stride, padding,
               dilation, groups) -> bool:
    if not _ENABLED:
        return False
    if not (x.is_cuda and x.dim() == 4 and x.dtype == torch.bfloat16):
        return False
    if groups != 1 or dilation[0] != 1 or dilation[1] != 1:
        return False
    if stride[0] != stride[1] or padding[0] != padding[1]:
        return False
    if stride[0] not in (1, 2):
        return False
    R, S = w.shape[2], w.shape[3]
    return R <= 7 and S <= 7


def _wgrad_split(M: int, K: int, C: int, R: int, S: int,
                 stride: int = 0, pad: int = -1, P: int = 0, Q: int = 0) -> int:
    """Pick the split-K factor: enough blocks to fill 256 CUs (~2 blocks/CU)
    without exploding the f32 partial buffer. Mirrors the kernel dispatch in
    conv.hip (generic / small-RSC / row-halo)."""
    tk = 128 if K >= 128 else 64
    if R * S > 1 and ((C == 4 and R * S * 4 <= 256)
                      or (C % 8 == 0 and C % 64 and R * S * C <= 768)
                      or R * S * C <= 192):
        # flattened small-RSC kernel (padded-channel stems incl. C%8) —
        # MUST mirror conv.hip's dispatch or the split target misjudges
        tiles = ((K + 63) // 64) * ((R * S * C + 63) // 64)
    elif (stride == 1 and R == 3 and S == 3 and pad == 1 and Q == P
          and 0 < Q <= 32 and (Q & (Q - 1)) == 0 and (P & (P - 1)) == 0
          and not os.environ.get('PS_WG_ROW_OFF')):
        tiles = ((K + 63) // 64) * 3 * ((C + 63) // 64)       # row-halo (TK=64)
    else:
        tiles = ((K + tk - 1) // tk) * R * S * ((C + 63) // 64)
    # target ~2 blocks/CU of fill; more split also means more partial-slab
    # traffic in the reduce, so keep a floor of 2 rather than over-splitting
    target = int(os.environ.get('PS_WG_WANT', '512'))
    want = max(2, target // max(tiles, 1))
    max_split = max(1, M // 64)
    return max(1, min(want, max_split, 256))


# Ragged-K backward padding (LeNet K=20/50, fc N=500/10): the dgrad/wgrad
# contraction runs over K, and K % 8 != 0 rows defeat load16's b128 path
# (mostly 4-B misaligned loads plus a masked-scalar tail octet per row —
# conv.hip:70-101). Padding dout rows and wT rows to a 64-multiple with
# zeros is bitwise-neutral (fp32 accum of +0.0 terms) and makes the whole
# reduction take the AL=true path. PS_PADK=0 disables (A/B).
# PS_PADK8=1: pad to an 8-multiple instead of 64 — rows stay 16-B aligned
# (load16<false>'s uint4 branch) at much less padding traffic, but the
# kernels run their AL=false instantiations.
_PADK = os.environ.get('PS_PADK', '1') != '0'
_PADK8 = os.environ.get('PS_PADK8', '0') != '0'


def _pad_target(K: int) -> int:
    return ((K + 7) & ~7) if _PADK8 else ((K + 63) & ~63)


def _pad_rows(lib, src: torch.Tensor, nrows: int, K: int, Kp: int):
    """[nrows, K] -> [nrows, Kp] with zero-filled tails (flat alloc)."""
    dst = torch.empty(nrows * Kp, dtype=src.dtype, device=src.device)
    lib.ps_padc(dst.data_ptr(), src.data_ptr(), nrows, K, Kp,
                current_stream_ptr())
    return dst


class _ConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride, pad):
        lib = require_lib()
        x = x.contiguous(memory_format=_CL)
        wc = w.contiguous(memory_format=_CL)
        Nb, C, H, W = x.shape
        K, _, R, S = w.shape
        P = (H + 2 * pad - R) // stride + 1
        Q = (W + 2 * pad - S) // stride + 1
        # Padded-channel fast paths (SMALL gather with aligned tap loads):
        #  * C<=3 stems -> 4 (two 8-B halves per quantum; dataset inputs,
        #    no dx needed)
        #  * ragged C%8 (LeNet conv2's C=20 -> 24) -> next multiple of 8
        #    (quantum within one tap: 16-B loads; dx computed padded and
        #    sliced back)
        cp = 0
        if R * S > 1:
            if C <= 3 and R * S * 4 <= 256 and not x.requires_grad:
                cp = 4
            elif C % 8 and C < 64:
                c8 = (C + 7) & ~7
                if c8 % 64 and R * S * c8 <= 768:
                    cp = c8
        ctx.cp = cp
        ctx.c_orig = C
        if cp:
            xp = torch.empty((Nb, cp, H, W), dtype=x.dtype, device=x.device,
                             memory_format=_CL)
            lib.ps_padc(xp.data_ptr(), x.data_ptr(), Nb * H * W, C, cp,
                        current_stream_ptr())
            wp = torch.empty((K, cp, R, S), dtype=w.dtype, device=w.device,
                             memory_format=_CL)
            lib.ps_padc(wp.data_ptr(), wc.data_ptr(), K * R * S, C, cp,
                        current_stream_ptr())
            x, wc, C = xp, wp, cp
        # NB: allocate channels_last DIRECTLY — empty().contiguous(CL) runs a
        # full transposing copy of uninitialized memory (~190us at l1 size)
        out = torch.empty((Nb, K, P, Q), dtype=x.dtype, device=x.device,
                          memory_format=_CL)
        # generic (NBUF=2) fwd paths also emit per-channel sum/sumsq
        # partials for a following BatchNorm (attribute handshake); MUST
        # mirror conv.hip's FWD_BODY dispatch
        rsc = R * S * C
        generic_fwd = not ((R * S > 1 and (rsc <= 192 or
                                           (C == 4 and R * S * 4 <= 256) or
                                           (C % 8 == 0 and C % 64 and rsc <= 768)))
                           or (R * S == 1 and C <= 64))
        stats = None
        # Measured NET NEGATIVE at -2.4% e2e (the epilogue's per-block
        # serial stats reduce costs more than the bn_stats pass it saves:
        # fwd 128x64 138 -> 245 us/call while bn_stats dropped 272 -> 38
        # us/step) — default off; PS_BN_FUSE=1 re-enables the experiment.
        # NB: grad mode is force-disabled inside Function.forward, so
        # torch.is_grad_enabled() is useless here; the param's
        # requires_grad is the training signal.
        if (_BN_FUSE and generic_fwd and K % 8 == 0 and w.requires_grad):
            tiles_m = (Nb * P * Q + 127) // 128
            stats = torch.empty(tiles_m * 2 * K, dtype=torch.float32,
                                device=x.device)
        lib.ps_conv_fwd(x.data_ptr(), wc.data_ptr(),
                        b.data_ptr() if b is not None else 0, out.data_ptr(),
                        stats.data_ptr() if stats is not None else 0,
                        Nb, H, W, C, K, P, Q, R, S, stride, pad,
                        current_stream_ptr())
        if stats is not None:
            # NOTE: apply() re-wraps forward's return value, so a python
            # attribute set HERE would be lost — stash it for the module
            # wrapper to attach to the wrapped output
            global _PENDING_STATS
            _PENDING_STATS = (stats, (Nb * P * Q + 127) // 128)
        ctx.save_for_backward(x, wc)
        ctx.conf = (stride, pad, b is not None)
        # steal-mode targets: write dw/db straight into the flat_g slices
        # (parallel/flat.py attach_grads) — AccumulateGrad then steals the
        # returned view instead of launching an add per param.
        ctx.gtgt = (getattr(w, '_ps_flat_grad_fn', None),
                    getattr(b, '_ps_flat_grad_fn', None) if b is not None else None)
        return out

    @staticmethod
    def backward(ctx, dout, dcarry=None):
        """Shared by _ConvFn (dcarry always None) and _ConvCarryFn (dcarry =
        the residual path's gradient w.r.t. x, fused into the dx epilogue)."""
        lib = require_lib()
        x, w = ctx.saved_tensors
        stride, pad, has_bias = ctx.conf
        dout = dout.contiguous(memory_format=_CL)
        Nb, C, H, W = x.shape
        K, _, R, S = w.shape
        P, Q = dout.shape[2], dout.shape[3]
        dx = dw = db = None
        # ragged-K: pad dout rows once, shared by dgrad and wgrad below.
        # K % 8 only — K%64-but-8-aligned douts (the out_pad pipeline's
        # K=24) measured FASTER unpadded (3.59M vs 3.38M img/s): branchy
        # 16-B-aligned loads beat eating 64/K x padc traffic.
        padk = (_PADK and K % 8 != 0
                and (ctx.needs_input_grad[0] or ctx.needs_input_grad[1]))
        if padk:
            Kp = _pad_target(K)
            doutp = _pad_rows(lib, dout, Nb * P * Q, K, Kp)
        else:
            Kp, doutp = K, dout
        if ctx.needs_input_grad[0]:
            # dgrad wants wT[R,S,C,K] in memory so its B stage is the same
            # contiguous-in-contraction load as fwd (no in-kernel transpose);
            # one tiled-transpose kernel launch per backward (~us).
            wt = torch.empty(R * S * C * K, dtype=w.dtype, device=w.device)
            lib.ps_wt_transpose(wt.data_ptr(), w.data_ptr(), K, R * S * C,
                                current_stream_ptr())
            if padk:   # pad the wT rows to match the padded contraction
                wt = _pad_rows(lib, wt, R * S * C, K, Kp)
            dx = torch.empty_like(x).contiguous(memory_format=_CL)
            if dcarry is not None and ctx.cp:
                # padded path: fuse the carry AFTER the slice instead
                dxp_carry = dcarry
                dcarry = None
            else:
                dxp_carry = None
            if dcarry is not None:
                dcarry = dcarry.contiguous(memory_format=_CL)
            lib.ps_conv_dgrad(doutp.data_ptr(), wt.data_ptr(), dx.data_ptr(),
                              dcarry.data_ptr() if dcarry is not None else 0,
                              Nb, H, W, C, Kp, P, Q, R, S, stride, pad,
                              current_stream_ptr())
            if ctx.cp:
                # computed at padded C: slice the real channels back
                dxs = torch.empty((Nb, ctx.c_orig, H, W), dtype=dx.dtype,
                                  device=dx.device, memory_format=_CL)
                dxs.copy_(dx[:, :ctx.c_orig])
                if dxp_carry is not None:
                    dxs.add_(dxp_carry)
                dx = dxs
        if ctx.needs_input_grad[1]:
            M = Nb * P * Q
            split = _wgrad_split(M, Kp, C, R, S, stride, pad, P, Q)
            partial = torch.empty(split * Kp * R * S * C,
                                  dtype=torch.float32, device=x.device)
            wt_tgt = ctx.gtgt[0]() if ctx.gtgt[0] is not None else None
            if ctx.cp or padk:
                # computed at padded C (stem) and/or padded K (ragged K):
                # slice the real rows/channels back into the (steal) dw.
                # NB: C here is already the padded value (x saved padded).
                dwp = torch.empty((Kp, C, R, S), dtype=w.dtype,
                                  device=w.device, memory_format=_CL)
                lib.ps_conv_wgrad(doutp.data_ptr(), x.data_ptr(),
                                  partial.data_ptr(), dwp.data_ptr(),
                                  Nb, H, W, C, Kp, P, Q, R, S, stride,
                                  pad, split, current_stream_ptr())
                c_out = ctx.c_orig if ctx.cp else C
                dw = (wt_tgt if wt_tgt is not None and wt_tgt.is_cuda
                      else torch.empty((K, c_out, R, S), dtype=w.dtype,
                                       device=w.device, memory_format=_CL))
                dw.copy_(dwp[:K, :c_out])
            else:
                dw = (wt_tgt if wt_tgt is not None and wt_tgt.dtype == w.dtype
                      and wt_tgt.is_cuda
                      else torch.empty_like(w).contiguous(memory_format=_CL))
                lib.ps_conv_wgrad(dout.data_ptr(), x.data_ptr(),
                                  partial.data_ptr(), dw.data_ptr(),
                                  Nb, H, W, C, K, P, Q, R, S, stride, pad,
                                  split, current_stream_ptr())
        if has_bias and ctx.needs_input_grad[2]:
            b_tgt = ctx.gtgt[1]() if ctx.gtgt[1] is not None else None
            db = (b_tgt if b_tgt is not None and b_tgt.dtype == dout.dtype
                  and b_tgt.is_cuda
                  else torch.empty(K, dtype=dout.dtype, device=dout.device))
            bpart = torch.empty(512 * Kp, dtype=torch.float32,
                                device=dout.device)
            if padk:
                # padded rows let colsum take its ushort8 path (ragged K
                # rows fall back to per-element loads — conv.hip colsum)
                dbp = torch.empty(Kp, dtype=dout.dtype, device=dout.device)
                lib.ps_conv_bias_grad(dbp.data_ptr(), doutp.data_ptr(),
                                      bpart.data_ptr(), Nb * P * Q, Kp,
                                      current_stream_ptr())
                db.copy_(dbp[:K])
            else:
                lib.ps_conv_bias_grad(db.data_ptr(), dout.data_ptr(),
                                      bpart.data_ptr(), Nb * P * Q, K,
                                      current_stream_ptr())
        return dx, dw, db, None, None


class _ConvCarryFn(torch.autograd.Function):
    """Conv that also passes x through as a second output for the residual
    branch. At a ResNet block input, x feeds BOTH conv1 and the shortcut;
    autograd would sum the two x-gradients with an elementwise add over the
    full activation (at::CUDAFunctor_add, ~0.25 ms/step on ResNet-18 b1024).
    Routing the shortcut through this node delivers the shortcut's gradient
    as dcarry to the SAME backward call, where the dgrad epilogue adds it
    in-register (ACCF kernels). Ref role: the implicit grad accumulation in
    the reference's BasicBlock.forward (src/model_ops/resnet.py:31-37)."""

    @staticmethod
    def forward(ctx, x, w, b, stride, pad):
        x = x.contiguous(memory_format=_CL)
        out = _ConvFn.forward(ctx, x, w, b, stride, pad)
        return out, x

    @staticmethod
    def backward(ctx, dout, dcarry):
        dx, dw, db, _, _ = _ConvFn.backward(ctx, dout, dcarry)
        return dx, dw, db, None, None


# Default ON: with the vectorized (LDS-staged) dgrad epilogue the carry add
# is a 16-B vector load and the fusion wins +2.0% e2e same-box (105.3k vs
# 103.2k img/s); the earlier scalar-epilogue version lost 2.6%.
_CARRY = os.environ.get('PS_CARRY', '1') != '0'


def conv_with_passthrough(mod: nn.Conv2d, x: torch.Tensor):
    """(conv(x), x) — on the kernel path the pass-through output carries the
    residual branch so its gradient fuses into dgrad; otherwise plain."""
    if _CARRY and _supported(x, mod.weight, mod.stride, mod.padding,
                             mod.dilation, mod.groups) and x.requires_grad:
        out, xp = _ConvCarryFn.apply(x, mod.weight, mod.bias,
                                     mod.stride[0], mod.padding[0])
        st = _pop_stats()
        if st is not None:
            out._ps_bn_stats = st
        return out, xp
    return mod(x), x


class PsConv2d(nn.Conv2d):
    """nn.Conv2d whose GPU bf16 path runs the in-tree CDNA4 kernels.

    Ragged-channel pipelines (LeNet's 20 -> 50) can stay 8-aligned end to
    end via `out_pad` (plain attr, default 0, not in state_dict): on the
    kernel path the conv emits out_pad channels by zero-padding its WEIGHT
    and bias (tiny tensors, differentiable F.pad — grads slice back), so
    the following relu/pool run on aligned rows and the next PsConv2d sees
    a pre-padded input, which it absorbs by zero-padding its own weight's C
    dim. The pad channels are exactly zero forward and get exactly-zero
    grads; the activation-space pads (padc over multi-MB tensors each
    step) disappear. Torch-fallback paths ignore out_pad and slice off any
    pre-pad, so CPU/eval flows stay at the nominal channel counts.
    """
    out_pad = 0

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w, b = self.weight, self.bias
        if _supported(x, w, self.stride, self.padding,
                      self.dilation, self.groups):
            cin = x.shape[1]
            if cin > self.in_channels:
                # pre-padded input from an out_pad producer upstream
                w = F.pad(w, (0, 0, 0, 0, 0, cin - self.in_channels))
            if self.out_pad > self.out_channels:
                kp = self.out_pad - self.out_channels
                w = F.pad(w, (0, 0, 0, 0, 0, 0, 0, kp))
                if b is not None:
                    b = F.pad(b, (0, kp))
            out = _ConvFn.apply(x, w, b, self.stride[0], self.padding[0])
            st = _pop_stats()
            if st is not None:
                out._ps_bn_stats = st
            return out
        if x.shape[1] != self.in_channels:   # fallback: drop pad channels
            x = x[:, :self.in_channels]
        return super().forward(x)
