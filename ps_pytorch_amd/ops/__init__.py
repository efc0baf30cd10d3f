"""ctypes loader for the in-tree gfx950 kernel library.

Policy (per-framework contract): on a GPU box the HIP path is THE path — if
the extension is missing or fails to load while CUDA/ROCm devices are
visible, ops raise immediately rather than falling back to eager torch.
CPU-only processes (tests, the build container) use torch fallbacks in
ops.functional.
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path
from typing import Optional

import torch

_LIB: Optional[ctypes.CDLL] = None
_LIB_ERR: Optional[str] = None

PS_F32 = 0
PS_BF16 = 1

_DTYPE_TAG = {torch.float32: PS_F32, torch.bfloat16: PS_BF16}


def _so_path() -> Path:
    return Path(__file__).resolve().parent / "libps_hip.so"


def _try_load() -> None:
    global _LIB, _LIB_ERR
    if _LIB is not None or _LIB_ERR is not None:
        return
    p = _so_path()
    if not p.exists():
        _LIB_ERR = f"{p} not built (run ps_pytorch_amd.ops.build or __graft_entry__.build())"
        return
    try:
        lib = ctypes.CDLL(str(p))
        lib.ps_fused_sgd.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_long, ctypes.c_float, ctypes.c_float, ctypes.c_float,
            ctypes.c_float, ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lib.ps_pack_bf16.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_long, ctypes.c_void_p]
        lib.ps_unpack_bf16.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_long, ctypes.c_void_p]
        lib.ps_pack_q8.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                   ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_unpack_q8.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_acc.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                               ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_conv_fwd.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_int] * 11 + [ctypes.c_void_p]
        lib.ps_conv_dgrad.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int] * 11 + [ctypes.c_void_p]
        lib.ps_conv_wgrad.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int] * 12 + [ctypes.c_void_p]
        lib.ps_conv_bias_grad.argtypes = [ctypes.c_void_p] * 3 + [
            ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_wt_transpose.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lib.ps_fused_adam.argtypes = [ctypes.c_void_p] * 6 + [
            ctypes.c_long] + [ctypes.c_float] * 7 + [
            ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lib.ps_bn_fwd.argtypes = [ctypes.c_void_p] * 13 + [
            ctypes.c_int,
            ctypes.c_long, ctypes.c_long, ctypes.c_float, ctypes.c_float,
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        lib.ps_bn_bwd.argtypes = [ctypes.c_void_p] * 12 + [
            ctypes.c_long, ctypes.c_long, ctypes.c_int, ctypes.c_int,
            ctypes.c_void_p]
        lib.ps_softmax_ce_fwd.argtypes = [ctypes.c_void_p] * 5 + [
            ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_softmax_ce_bwd.argtypes = [ctypes.c_void_p] * 5 + [
            ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_maxpool_fwd.argtypes = [ctypes.c_void_p] * 3 + [
            ctypes.c_int] * 10 + [ctypes.c_void_p]
        lib.ps_maxpool_bwd.argtypes = [ctypes.c_void_p] * 3 + [
            ctypes.c_int] * 10 + [ctypes.c_void_p]
        lib.ps_gavgpool_fwd.argtypes = [ctypes.c_void_p] * 2 + [
            ctypes.c_int] * 3 + [ctypes.c_void_p]
        lib.ps_gavgpool_bwd.argtypes = [ctypes.c_void_p] * 2 + [
            ctypes.c_int] * 3 + [ctypes.c_void_p]
        lib.ps_pad4.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                ctypes.c_long, ctypes.c_int, ctypes.c_void_p]
        lib.ps_padc.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                ctypes.c_long, ctypes.c_int, ctypes.c_int,
                                ctypes.c_void_p]
        _LIB = lib
    except OSError as e:  # pragma: no cover
        _LIB_ERR = str(e)


def lib_or_none() -> Optional[ctypes.CDLL]:
    _try_load()
    return _LIB


def require_lib() -> ctypes.CDLL:
    """The GPU compute path: loud failure if the native library is absent."""
    _try_load()
    if _LIB is None:
        raise RuntimeError(
            f"ps_pytorch_amd HIP extension unavailable on a GPU host: {_LIB_ERR}")
    return _LIB


def dtype_tag(dt: torch.dtype) -> int:
    try:
        return _DTYPE_TAG[dt]
    except KeyError:
        raise TypeError(f"unsupported wire/compute dtype {dt}")


def current_stream_ptr() -> int:
    return torch.cuda.current_stream().cuda_stream


from . import functional  # noqa: E402,F401
