"""Per-shape conv microbench: PsConv2d MFMA kernels vs torch/MIOpen.

Times fwd / dgrad / wgrad separately on the ResNet-18/CIFAR b=1024 shapes.
Run on a GPU box:  python tools/conv_microbench.py [--batch 1024]
"""
import argparse
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, '.')
from ps_pytorch_amd.ops.conv import _ConvFn  # noqa: E402

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
    Nb = args.batch
    print(f"{'shape':>20} {'impl':>6} {'fwd us':>9} {'dgrad us':>9} "
          f"{'wgrad us':>9} {'TF fwd':>7}")
    for name, C, H, W, K, R, stride, pad in SHAPES:
        P = (H + 2 * pad - R) // stride + 1
        x = torch.randn(Nb, C, H, W, device='cuda', dtype=torch.bfloat16) \
            .contiguous(memory_format=_CL)
        w = torch.randn(K, C, R, R, device='cuda', dtype=torch.bfloat16) \
            .mul(0.05).contiguous(memory_format=_CL)
        dout = torch.randn(Nb, K, P, P, device='cuda', dtype=torch.bfloat16) \
            .contiguous(memory_format=_CL)
        flops = 2.0 * Nb * P * P * K * R * R * C

        for impl in ('ps', 'miopen'):
            if impl == 'ps':
                def fwd():
                    return _ConvFn.apply(x, w, None, stride, pad)
            else:
                def fwd():
                    return F.conv2d(x, w, None, stride=stride, padding=pad)
            t_f = timeit(fwd, args.iters)
            xg = x.clone().requires_grad_(True)
            wg = w.clone().requires_grad_(True)
            out = (_ConvFn.apply(xg, wg, None, stride, pad) if impl == 'ps'
                   else F.conv2d(xg, wg, None, stride=stride, padding=pad))
            gfn = torch.autograd.grad

            def bwd_x():
                return gfn(out, xg, dout, retain_graph=True)

            def bwd_w():
                return gfn(out, wg, dout, retain_graph=True)
            t_dx = timeit(bwd_x, args.iters)
            t_dw = timeit(bwd_w, args.iters)
            print(f"{name:>20} {impl:>6} {t_f:>9.1f} {t_dx:>9.1f} "
                  f"{t_dw:>9.1f} {flops / t_f / 1e6:>7.0f}")


if __name__ == '__main__':
    main()
