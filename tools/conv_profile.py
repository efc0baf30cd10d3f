"""Minimal driver for rocprofv3 PMC runs: executes a handful of conv ops
N times each so per-kernel counters are clean. Shapes via --shape."""
import argparse
import sys

import torch

sys.path.insert(0, '.')
from ps_pytorch_amd.ops import require_lib, current_stream_ptr  # noqa: E402
from ps_pytorch_amd.ops.conv import _wgrad_split  # noqa: E402

_CL = torch.channels_last

SHAPES = {
    'l1': (64, 32, 32, 64, 3, 1, 1),
    'l3': (256, 8, 8, 256, 3, 1, 1),
    'stem': (3, 32, 32, 64, 3, 1, 1),
    'l2d': (64, 32, 32, 128, 3, 2, 1),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--shape', default='l1')
    ap.add_argument('--op', default='wgrad', choices=['fwd', 'dgrad', 'wgrad'])
    ap.add_argument('--batch', type=int, default=1024)
    ap.add_argument('--iters', type=int, default=10)
    args = ap.parse_args()
    lib = require_lib()
    C, H, W, K, R, stride, pad = SHAPES[args.shape]
    Nb = args.batch
    P = (H + 2 * pad - R) // stride + 1
    x = torch.randn(Nb, C, H, W, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=_CL)
    w = torch.randn(K, C, R, R, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=_CL)
    dout = torch.randn(Nb, K, P, P, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=_CL)
    out = torch.empty_like(dout)
    dx = torch.empty_like(x)
    dw = torch.empty_like(w)
    wt = w.permute(2, 3, 1, 0).contiguous()
    M = Nb * P * P
    split = _wgrad_split(M, K, C, R, R, stride, pad, P, P)
    partial = torch.empty(split * K * R * R * C, dtype=torch.float32,
                          device='cuda')
    strm = current_stream_ptr()
    torch.cuda.synchronize()
    for _ in range(args.iters):
        if args.op == 'fwd':
            lib.ps_conv_fwd(x.data_ptr(), w.data_ptr(), 0, out.data_ptr(),
                            0, Nb, H, W, C, K, P, P, R, R, stride, pad, strm)
        elif args.op == 'dgrad':
            lib.ps_conv_dgrad(dout.data_ptr(), wt.data_ptr(), dx.data_ptr(),
                              0, Nb, H, W, C, K, P, P, R, R, stride, pad,
                              strm)
        else:
            lib.ps_conv_wgrad(dout.data_ptr(), x.data_ptr(),
                              partial.data_ptr(), dw.data_ptr(),
                              Nb, H, W, C, K, P, P, R, R, stride, pad,
                              split, strm)
    torch.cuda.synchronize()
    print('done', args.shape, args.op, 'split', split)


if __name__ == '__main__':
    main()
