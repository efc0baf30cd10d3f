"""Python-facing ops: HIP kernels on GPU, torch reference path on CPU.

Every GPU entry point calls require_lib() — a GPU host without the in-tree
libps_hip.so fails loudly (no silent eager fallback). The CPU paths double
as the numerics references for the kernel tests (tests/test_ops_*.py).
"""
from __future__ import annotations

from typing import Optional

import torch

from . import require_lib, dtype_tag, current_stream_ptr


def _check_flat(t: torch.Tensor, name: str, n: int) -> None:
    if not t.is_contiguous():
        raise ValueError(f"{name} must be contiguous")
    if t.numel() != n:
        raise ValueError(f"{name} numel {t.numel()} != {n}")


def fused_sgd_step(
    w: torch.Tensor,            # f32 master weights (flat)
    grad_sum: torch.Tensor,     # f32 or bf16 summed grads (flat)
    momentum_buf: torch.Tensor, # f32 (flat)
    lr: float,
    momentum: float = 0.0,
    weight_decay: float = 0.0,
    grad_scale: float = 1.0,    # usually 1/num_aggregate
    nesterov: bool = False,
    wire_out: Optional[torch.Tensor] = None,  # optional f32/bf16 broadcast payload
) -> None:
    """w/m update + optional wire re-pack, semantics of ref src/optim/sgd.py:59-92
    (momentum buffer m = mu*m + g; w -= lr*m; nesterov variant as torch)."""
    n = w.numel()
    _check_flat(w, "w", n)
    _check_flat(grad_sum, "grad_sum", n)
    _check_flat(momentum_buf, "momentum_buf", n)
    if wire_out is not None:
        _check_flat(wire_out, "wire_out", n)

    if w.is_cuda:
        if n % 4:
            raise ValueError("flat buffers must be padded to a multiple of 4")
        lib = require_lib()
        lib.ps_fused_sgd(
            w.data_ptr(), grad_sum.data_ptr(), momentum_buf.data_ptr(),
            0 if wire_out is None else wire_out.data_ptr(), n,
            float(lr), float(momentum), float(weight_decay), float(grad_scale),
            int(nesterov), dtype_tag(grad_sum.dtype),
            dtype_tag(wire_out.dtype) if wire_out is not None else 0,
            current_stream_ptr())
        return

    # CPU reference path (also the golden model for the kernel test)
    g = grad_sum.to(torch.float32) * grad_scale
    if weight_decay:
        g = g.add(w, alpha=weight_decay)
    momentum_buf.mul_(momentum).add_(g)
    upd = g.add(momentum_buf, alpha=momentum) if nesterov else momentum_buf
    w.add_(upd, alpha=-lr)
    if wire_out is not None:
        wire_out.copy_(w.to(wire_out.dtype))



def fused_adam_step(
    w: torch.Tensor, grad_sum: torch.Tensor, exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor, t: int, lr: float, beta1: float, beta2: float,
    eps: float, weight_decay: float = 0.0, grad_scale: float = 1.0,
    max_exp_avg_sq: Optional[torch.Tensor] = None,
    wire_out: Optional[torch.Tensor] = None,
) -> None:
    """Fused flat Adam (ref src/optim/adam.py:48-95 semantics, incl.
    amsgrad); one kernel pass on GPU, torch ops on CPU."""
    n = w.numel()
    _check_flat(w, "w", n)
    _check_flat(grad_sum, "grad_sum", n)
    bc1 = 1.0 - beta1 ** t
    bc2 = 1.0 - beta2 ** t
    if w.is_cuda:
        if n % 4:
            raise ValueError("flat buffers must be padded to a multiple of 4")
        lib = require_lib()
        lib.ps_fused_adam(
            w.data_ptr(), grad_sum.data_ptr(), exp_avg.data_ptr(),
            exp_avg_sq.data_ptr(),
            0 if max_exp_avg_sq is None else max_exp_avg_sq.data_ptr(),
            0 if wire_out is None else wire_out.data_ptr(), n,
            float(lr / bc1), float(beta1), float(beta2), float(1.0 / bc2),
            float(eps), float(weight_decay), float(grad_scale),
            dtype_tag(grad_sum.dtype),
            dtype_tag(wire_out.dtype) if wire_out is not None else 0,
            current_stream_ptr())
        return
    g = grad_sum.to(torch.float32)
    if grad_scale != 1.0:
        g = g * grad_scale
    if weight_decay:
        g = g.add(w, alpha=weight_decay)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    if max_exp_avg_sq is not None:
        torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
        denom = (max_exp_avg_sq / bc2).sqrt_().add_(eps)
    else:
        denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    w.addcdiv_(exp_avg, denom, value=-lr / bc1)
    if wire_out is not None:
        wire_out.copy_(w.to(wire_out.dtype))

def pack_wire(dst: torch.Tensor, src: torch.Tensor) -> None:
    """Wire pack: f32 -> bf16 (the GPU 'compression' path, ref compression.py
    g_compress role) or plain copy when dtypes match."""
    n = src.numel()
    _check_flat(dst, "dst", n)
    if dst.dtype == src.dtype:
        dst.copy_(src)
        return
    if src.is_cuda:
        lib = require_lib()
        if src.dtype == torch.float32 and dst.dtype == torch.bfloat16:
            lib.ps_pack_bf16(dst.data_ptr(), src.data_ptr(), n, current_stream_ptr())
        elif src.dtype == torch.bfloat16 and dst.dtype == torch.float32:
            lib.ps_unpack_bf16(dst.data_ptr(), src.data_ptr(), n, current_stream_ptr())
        else:
            raise TypeError(f"unsupported pack {src.dtype} -> {dst.dtype}")
        return
    dst.copy_(src.to(dst.dtype))


unpack_wire = pack_wire  # symmetric: direction decided by dtypes


# ---- block-scaled int8 codec (the blosc-role GPU compressor) ----
# Scheme/layout documented in ops/kernels/quant.hip: 256-elem blocks,
# per-block f32 scale = max|x|/127, payload = [int8 x align4(n)][f32 x nblk].
# The CPU path below is the numerics REFERENCE for the HIP kernels and the
# gloo test path; both round identically (rint == torch.round, half-to-even).

Q8_BLOCK = 256


def q8_layout(n: int):
    """(scales_byte_offset, total_payload_bytes) for n f32 values."""
    qb = (n + 3) & ~3
    nblk = (n + Q8_BLOCK - 1) // Q8_BLOCK
    return qb, qb + 4 * nblk


def pack_q8(payload: torch.Tensor, src: torch.Tensor) -> None:
    """Compress flat f32 `src` into the uint8 `payload` buffer (4x vs f32)."""
    n = src.numel()
    qb, tot = q8_layout(n)
    if payload.dtype != torch.uint8 or payload.numel() != tot:
        raise ValueError(f"payload must be uint8[{tot}], got "
                         f"{payload.dtype}[{payload.numel()}]")
    if src.dtype not in (torch.float32, torch.bfloat16):
        raise TypeError("q8 source must be f32 or bf16")
    _check_flat(src, "src", n)
    if src.is_cuda:
        require_lib().ps_pack_q8(payload.data_ptr(), src.data_ptr(), n,
                                 dtype_tag(src.dtype), current_stream_ptr())
        return
    nblk = (n + Q8_BLOCK - 1) // Q8_BLOCK
    x = torch.zeros(nblk * Q8_BLOCK, dtype=torch.float32)
    x[:n] = src.to(torch.float32)
    xb = x.view(nblk, Q8_BLOCK)
    m = xb.abs().amax(dim=1)
    # NB: divide f32-tensor by f32-tensor — torch computes python-scalar /
    # tensor in f64 and casts, which is 1 ulp off strict f32 division and
    # breaks bitwise parity with the GPU kernel's IEEE f32 divide.
    c127 = torch.full_like(m, 127.0)
    inv = torch.where(m > 0, c127 / m, torch.zeros_like(m))
    q = torch.clamp(torch.round(xb * inv[:, None]), -127, 127).to(torch.int8)
    payload[:n] = q.view(-1)[:n].view(torch.uint8)
    payload[qb:qb + 4 * nblk].view(torch.float32).copy_(m / c127)


def unpack_q8(dst: torch.Tensor, payload: torch.Tensor,
              accumulate: bool = False) -> None:
    """Decompress `payload` into f32 `dst` (dst += values if accumulate)."""
    n = dst.numel()
    qb, tot = q8_layout(n)
    if payload.numel() != tot:
        raise ValueError(f"payload size {payload.numel()} != {tot}")
    if dst.dtype != torch.float32:
        raise TypeError("q8 destination must be f32")
    _check_flat(dst, "dst", n)
    if dst.is_cuda:
        require_lib().ps_unpack_q8(dst.data_ptr(), payload.data_ptr(), n,
                                   int(accumulate), current_stream_ptr())
        return
    nblk = (n + Q8_BLOCK - 1) // Q8_BLOCK
    q = payload[:n].view(torch.int8).to(torch.float32)
    scales = payload[qb:qb + 4 * nblk].view(torch.float32)
    vals = q * scales.repeat_interleave(Q8_BLOCK)[:n]
    if accumulate:
        dst.add_(vals)
    else:
        dst.copy_(vals)


def acc_into(acc: torch.Tensor, src: torch.Tensor) -> None:
    """acc += src (f32 acc; f32/bf16 src) — PS fan-in accumulate for
    uncompressed gather mode (the reference's Waitany += loop,
    sync_replicas_master_nn.py:157-186, on-device)."""
    n = acc.numel()
    _check_flat(acc, "acc", n)
    _check_flat(src, "src", n)
    if acc.dtype != torch.float32:
        raise TypeError("accumulator must be f32")
    if acc.is_cuda:
        require_lib().ps_acc(acc.data_ptr(), src.data_ptr(), n,
                             dtype_tag(src.dtype), current_stream_ptr())
        return
    acc.add_(src.to(torch.float32))
