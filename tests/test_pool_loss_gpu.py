"""Pool + softmax-CE HIP kernels vs plain-torch fp32 references."""
import pytest
import torch
import torch.nn.functional as F

from ps_pytorch_amd.ops.pool import _MaxPoolFn, _GlobalAvgPoolFn
from ps_pytorch_amd.ops.loss import _SoftmaxCEFn

pytestmark = pytest.mark.gpu

_CL = torch.channels_last


def _rel_err(a, b):
    d = (a.detach().float() - b.detach().float()).abs().max()
    return float(d / b.float().abs().max().clamp_min(1e-6))


MP_SHAPES = [
    # Nb, C, H, W, k, stride, pad
    (8, 20, 24, 24, 2, 2, 0),       # LeNet pool1 (C % 8 != 0)
    (8, 50, 8, 8, 2, 2, 0),         # LeNet pool2
    (4, 64, 112, 112, 3, 2, 1),     # ImageNet stem maxpool (overlapping)
    (3, 24, 9, 7, 3, 2, 1),         # ragged
]


@pytest.mark.parametrize("shape", MP_SHAPES)
def test_maxpool_matches_torch(shape):
    Nb, C, H, W, k, st, pd = shape
    g = torch.Generator().manual_seed(hash(shape) % (2 ** 31))
    x = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL).requires_grad_(True)
    y = _MaxPoolFn.apply(x, k, st, pd)
    dy = torch.randn(y.shape, generator=g).to('cuda', torch.bfloat16) \
        .contiguous(memory_format=_CL)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    yr = F.max_pool2d(xr, k, st, pd)
    yr.backward(dy.float())
    assert y.shape == yr.shape
    # max of bf16 values is exact; grads route to the same argmax except on
    # bf16-equal ties, which torch also resolves first-in-scan-order
    assert _rel_err(y, yr) < 1e-6
    assert _rel_err(x.grad, xr.grad) < 0.03


def test_global_avg_pool_matches_torch():
    g = torch.Generator().manual_seed(11)
    for Nb, C, H, W in [(8, 512, 4, 4), (4, 2048, 7, 7), (3, 24, 5, 3)]:
        x = torch.randn(Nb, C, H, W, generator=g).to('cuda', torch.bfloat16) \
            .contiguous(memory_format=_CL).requires_grad_(True)
        y = _GlobalAvgPoolFn.apply(x)
        dy = torch.randn(y.shape, generator=g).to('cuda', torch.bfloat16)
        y.backward(dy)
        xr = x.detach().float().requires_grad_(True)
        yr = F.adaptive_avg_pool2d(xr, 1).flatten(1)
        yr.backward(dy.float())
        assert y.shape == yr.shape
        assert _rel_err(y, yr) < 0.02
        assert _rel_err(x.grad, xr.grad) < 0.02


@pytest.mark.parametrize("MC", [(1024, 10), (512, 100), (256, 1000),
                                (33, 17), (8192, 10)])
def test_softmax_ce_matches_torch(MC):
    M, C = MC
    g = torch.Generator().manual_seed(M * 1000 + C)
    x = (torch.randn(M, C, generator=g) * 3).to('cuda', torch.bfloat16) \
        .requires_grad_(True)
    t = torch.randint(0, C, (M,), generator=g).cuda()
    loss = _SoftmaxCEFn.apply(x, t)
    loss.backward()

    xr = x.detach().float().requires_grad_(True)
    lr = F.cross_entropy(xr, t)
    lr.backward()
    assert abs(float(loss) - float(lr)) < 2e-3 * max(1.0, abs(float(lr)))
    assert _rel_err(x.grad, xr.grad) < 0.03


def test_softmax_ce_deterministic():
    g = torch.Generator().manual_seed(5)
    x = (torch.randn(1024, 10, generator=g) * 2).to('cuda', torch.bfloat16)
    t = torch.randint(0, 10, (1024,), generator=g).cuda()
    vals = [float(_SoftmaxCEFn.apply(x, t)) for _ in range(2)]
    assert vals[0] == vals[1]
