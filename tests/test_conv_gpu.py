"""PsConv2d MFMA kernels vs plain-torch fp32 reference (SURVEY.md §4 rule:
numerics tests for a HIP kernel compare it against a torch fp32 reference
of the same op). Covers the ResNet-18/CIFAR shape family, the 1x1-stride-2
downsample, the C=3 stem, and LeNet's 5x5 no-pad convs with bias."""
import pytest
import torch
import torch.nn.functional as F

from ps_pytorch_amd.ops.conv import _ConvFn

pytestmark = pytest.mark.gpu

_CL = torch.channels_last

SHAPES = [
    # Nb, C, H, W, K, R, stride, pad, bias
    (8, 64, 32, 32, 64, 3, 1, 1, False),     # layer1 3x3
    (8, 64, 32, 32, 128, 3, 2, 1, False),    # layer2 downsample 3x3 s2
    (8, 128, 16, 16, 128, 3, 1, 1, False),
    (8, 64, 32, 32, 128, 1, 2, 0, False),    # 1x1 s2 shortcut
    (4, 256, 8, 8, 512, 3, 2, 1, False),
    (4, 3, 32, 32, 64, 3, 1, 1, False),      # stem C=3 (scalar path)
    (4, 1, 28, 28, 20, 5, 1, 0, True),       # LeNet conv1
    (4, 20, 12, 12, 50, 5, 1, 0, True),      # LeNet conv2 (C,K % 8 != 0)
    (3, 16, 9, 7, 24, 3, 1, 1, False),       # ragged M (tile tails)
    (2, 3, 64, 64, 64, 7, 2, 3, False),      # R50-style stem 7x7 s2
                                             # (RSC=147 multi-chunk flattened)
]


def _rel_err(a: torch.Tensor, b: torch.Tensor) -> float:
    d = (a.detach().float() - b.detach().float()).abs().max()
    return float(d / b.float().abs().max().clamp_min(1e-6))


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_fwd_bwd_matches_torch(shape):
    Nb, C, H, W, K, R, stride, pad, bias = shape
    g = torch.Generator().manual_seed(hash(shape) % (2 ** 31))
    x = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL).requires_grad_(True)
    w = (torch.randn(K, C, R, R, generator=g) / (R * R * C) ** 0.5) \
        .to('cuda', torch.bfloat16).contiguous(memory_format=_CL) \
        .requires_grad_(True)
    b = (torch.randn(K, generator=g).to('cuda', torch.bfloat16)
         .requires_grad_(True) if bias else None)

    out = _ConvFn.apply(x, w, b, stride, pad)
    dout = torch.randn(out.shape, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    out.backward(dout)

    # fp32 reference on the SAME bf16-rounded values
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True) if bias else None
    outr = F.conv2d(xr, wr, br, stride=stride, padding=pad)
    outr.backward(dout.float())

    assert out.shape == outr.shape
    assert _rel_err(out, outr) < 0.03, f"fwd {_rel_err(out, outr)}"
    assert _rel_err(x.grad, xr.grad) < 0.03, f"dgrad {_rel_err(x.grad, xr.grad)}"
    assert _rel_err(w.grad, wr.grad) < 0.03, f"wgrad {_rel_err(w.grad, wr.grad)}"
    if bias:
        assert _rel_err(b.grad, br.grad) < 0.03, "bias grad"


def test_conv_wgrad_deterministic():
    """Two identical backward passes give bit-identical dw (fixed-order
    split-K reduce, no atomics) — PS replicas must agree bitwise."""
    g = torch.Generator().manual_seed(7)
    x = torch.randn(8, 64, 32, 32, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    w = torch.randn(64, 64, 3, 3, generator=g).mul(0.05) \
        .to('cuda', torch.bfloat16).contiguous(memory_format=_CL)
    dout = torch.randn(8, 64, 32, 32, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    grads = []
    for _ in range(2):
        wv = w.clone().requires_grad_(True)
        out = _ConvFn.apply(x, wv, None, 1, 1)
        out.backward(dout)
        grads.append(wv.grad.clone())
    assert torch.equal(grads[0], grads[1])


CARRY_SHAPES = [
    (8, 64, 32, 32, 64, 3, 1, 1),     # BasicBlock conv1 s1 (gemm ACCF)
    (8, 64, 32, 32, 128, 3, 2, 1),    # downsample conv1 s2 (dgrad2 ACCF)
    (8, 64, 32, 32, 128, 1, 2, 0),    # 1x1 s2 (empty parity classes: carry copy)
]


@pytest.mark.parametrize("shape", CARRY_SHAPES)
def test_conv_carry_fuses_residual_grad(shape):
    """_ConvCarryFn: grad w.r.t. x must equal dgrad(dy) + dcarry — the
    residual-fork accumulation fused into the dx epilogue (ACCF kernels)."""
    from ps_pytorch_amd.ops.conv import _ConvCarryFn
    Nb, C, H, W, K, R, stride, pad = shape
    g = torch.Generator().manual_seed(hash(shape) % (2 ** 31))
    x = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL).requires_grad_(True)
    w = (torch.randn(K, C, R, R, generator=g) / (R * R * C) ** 0.5) \
        .to('cuda', torch.bfloat16).contiguous(memory_format=_CL) \
        .requires_grad_(True)
    scale = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)

    out, xp = _ConvCarryFn.apply(x, w, None, stride, pad)
    dout = torch.randn(out.shape, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    # residual branch consumes xp; its grad (scale) rides dcarry
    loss = (out.float() * dout.float()).sum() + (xp.float() * scale.float()).sum()
    loss.backward()

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    outr = F.conv2d(xr, wr, None, stride=stride, padding=pad)
    lossr = (outr * dout.float()).sum() + (xr * scale.float()).sum()
    lossr.backward()

    assert _rel_err(x.grad, xr.grad) < 0.03, f"dx+carry {_rel_err(x.grad, xr.grad)}"
    assert _rel_err(w.grad, wr.grad) < 0.03


@pytest.mark.parametrize("shape4", [
    (8, 3, 32, 32, 64, 3, 1, 1),     # CIFAR stem (RSC4=36, one-step)
    (4, 3, 64, 64, 64, 7, 2, 3),     # R50-style stem 7x7 s2 (RSC4=196)
    (4, 1, 28, 28, 20, 5, 1, 0),     # LeNet conv1 (C=1 -> pad 4)
])
def test_conv_c4_padded_stem_path(shape4):
    """C<=3 stems with a non-grad input take the padded-channel (C4) fast
    gather; fwd output and dw (computed at C=4, sliced) must match torch."""
    Nb, C, H, W, K, R, stride, pad = shape4
    g = torch.Generator().manual_seed(hash(shape4) % (2 ** 31))
    x = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)            # NO requires_grad: C4 path
    w = (torch.randn(K, C, R, R, generator=g) / (R * R * C) ** 0.5) \
        .to('cuda', torch.bfloat16).contiguous(memory_format=_CL) \
        .requires_grad_(True)
    out = _ConvFn.apply(x, w, None, stride, pad)
    dout = torch.randn(out.shape, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    out.backward(dout)

    xr = x.detach().float()
    wr = w.detach().float().requires_grad_(True)
    outr = F.conv2d(xr, wr, None, stride=stride, padding=pad)
    outr.backward(dout.float())
    assert _rel_err(out, outr) < 0.03, f"c4 fwd {_rel_err(out, outr)}"
    assert _rel_err(w.grad, wr.grad) < 0.03, f"c4 wgrad {_rel_err(w.grad, wr.grad)}"
