"""Per-shape conv microbench: PsConv2d MFMA kernels vs torch/MIOpen.

Times fwd / dgrad / wgrad separately on the ResNet-18/CIFAR b=1024 shapes.
The ps ops are timed through the C API directly (the autograd Function
computes dgrad+wgrad together when both leaves require grad, which
conflates the per-op numbers).

Run on a GPU box:  python tools/conv_microbench.py [--batch 1024]
"""
import argparse
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, '.')
from ps_pytorch_amd.ops import require_lib, current_stream_ptr  # noqa: E402
from ps_pytorch_amd.ops.conv import _wgrad_split  # noqa: E402

_CL = torch.channels_last

SHAPES = [
    ("stem 3->64", 3, 32, 32, 64, 3, 1, 1),
    ("l1 64->64", 64, 32, 32, 64, 3, 1, 1),
    ("l2d 64->128 s2", 64, 32, 32, 128, 3, 2, 1),
    ("l2 128->128", 128, 16, 16, 128, 3, 1, 1),
    ("sc 64->128 1x1s2", 64, 32, 32, 128, 1, 2, 0),
    ("l3d 128->256 s2", 128, 16, 16, 256, 3, 2, 1),
    ("l3 256->256", 256, 8, 8, 256, 3, 1, 1),
    ("l4d 256->512 s2", 256, 8, 8, 512, 3, 2, 1),
    ("l4 512->512", 512, 4, 4, 512, 3, 1, 1),
]


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6   # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batch', type=int, default=1024)
    ap.add_argument('--iters', type=int, default=20)
    args = ap.parse_args()
    torch.backends.cudnn.benchmark = True
    lib = require_lib()
    Nb = args.batch
    print(f"{'shape':>20} {'impl':>6} {'fwd us':>9} {'dgrad us':>9} "
          f"{'wgrad us':>9} {'TF fwd':>7} {'TF dx':>7} {'TF dw':>7}")
    for name, C, H, W, K, R, stride, pad in SHAPES:
        P = (H + 2 * pad - R) // stride + 1
        x = torch.randn(Nb, C, H, W, device='cuda', dtype=torch.bfloat16) \
            .contiguous(memory_format=_CL)
        w = torch.randn(K, C, R, R, device='cuda', dtype=torch.bfloat16) \
            .mul(0.05).contiguous(memory_format=_CL)
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
        flops = 2.0 * M * K * R * R * C
        strm = current_stream_ptr()

        for impl in ('ps', 'miopen'):
            if impl == 'ps':
                def fwd():
                    lib.ps_conv_fwd(x.data_ptr(), w.data_ptr(), 0,
                                    out.data_ptr(), 0, Nb, H, W, C, K, P, P,
                                    R, R, stride, pad, strm)

                def bwd_x():
                    lib.ps_conv_dgrad(dout.data_ptr(), wt.data_ptr(),
                                      dx.data_ptr(), 0, Nb, H, W, C, K, P, P,
                                      R, R, stride, pad, strm)

                def bwd_w():
                    lib.ps_conv_wgrad(dout.data_ptr(), x.data_ptr(),
                                      partial.data_ptr(), dw.data_ptr(),
                                      Nb, H, W, C, K, P, P, R, R, stride,
                                      pad, split, strm)
            else:
                xg = x.clone().requires_grad_(True)
                og = F.conv2d(xg, w, None, stride=stride, padding=pad)

                def fwd():
                    return F.conv2d(x, w, None, stride=stride, padding=pad)

                def bwd_x(og=og, xg=xg):
                    return torch.autograd.grad(og, xg, dout,
                                               retain_graph=True)

                wg = w.clone().requires_grad_(True)
                og2 = F.conv2d(x, wg, None, stride=stride, padding=pad)

                def bwd_w(og2=og2, wg=wg):
                    return torch.autograd.grad(og2, wg, dout,
                                               retain_graph=True)
            t_f = timeit(fwd, args.iters)
            t_dx = timeit(bwd_x, args.iters)
            t_dw = timeit(bwd_w, args.iters)
            print(f"{name:>20} {impl:>6} {t_f:>9.1f} {t_dx:>9.1f} "
                  f"{t_dw:>9.1f} {flops / t_f / 1e6:>7.0f} "
                  f"{flops / t_dx / 1e6:>7.0f} {flops / t_dw / 1e6:>7.0f}")


if __name__ == '__main__':
    main()
