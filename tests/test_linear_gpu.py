"""PsLinear MFMA GEMMs vs plain-torch fp32 reference (numerics rule,
SURVEY.md §4). Shapes cover every Linear in the model zoo: ResNet
classifiers (512/2048 -> 10/1000), LeNet fc (800->500->10), VGG (512->10),
plus ragged sizes for tile-tail coverage."""
import pytest
import torch
import torch.nn.functional as F

from ps_pytorch_amd.ops.linear import _LinearFn

pytestmark = pytest.mark.gpu

SHAPES = [
    # M, K, N, bias
    (1024, 512, 10, True),      # ResNet-18/CIFAR classifier
    (256, 2048, 1000, True),    # ResNet-50/ImageNet classifier
    (1024, 800, 500, True),     # LeNet fc1
    (1024, 500, 10, True),      # LeNet fc2
    (128, 512, 10, True),       # VGG CIFAR classifier
    (33, 100, 17, True),        # ragged everything
    (64, 64, 64, False),        # aligned, no bias
]


def _rel_err(a, b):
    d = (a.detach().float() - b.detach().float()).abs().max()
    return float(d / b.float().abs().max().clamp_min(1e-6))


@pytest.mark.parametrize("shape", SHAPES)
def test_linear_fwd_bwd_matches_torch(shape):
    M, K, N, bias = shape
    g = torch.Generator().manual_seed(hash(shape) % (2 ** 31))
    x = torch.randn(M, K, generator=g).div(K ** 0.5) \
        .to('cuda', torch.bfloat16).requires_grad_(True)
    w = torch.randn(N, K, generator=g).div(K ** 0.5) \
        .to('cuda', torch.bfloat16).requires_grad_(True)
    b = (torch.randn(N, generator=g).to('cuda', torch.bfloat16)
         .requires_grad_(True) if bias else None)

    y = _LinearFn.apply(x, w, b)
    dy = torch.randn(M, N, generator=g).to('cuda', torch.bfloat16)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True) if bias else None
    yr = F.linear(xr, wr, br)
    yr.backward(dy.float())

    assert y.shape == yr.shape
    assert _rel_err(y, yr) < 0.03, f"fwd {_rel_err(y, yr)}"
    assert _rel_err(x.grad, xr.grad) < 0.03, f"dgrad {_rel_err(x.grad, xr.grad)}"
    assert _rel_err(w.grad, wr.grad) < 0.03, f"wgrad {_rel_err(w.grad, wr.grad)}"
    if bias:
        assert _rel_err(b.grad, br.grad) < 0.03, "bias grad"


def test_linear_3d_input_and_determinism():
    g = torch.Generator().manual_seed(3)
    x = torch.randn(4, 32, 256, generator=g).to('cuda', torch.bfloat16)
    w = torch.randn(64, 256, generator=g).div(16).to('cuda', torch.bfloat16)
    b = torch.zeros(64, device='cuda', dtype=torch.bfloat16)
    y = _LinearFn.apply(x, w, b)
    assert y.shape == (4, 32, 64)
    dy = torch.randn(y.shape, generator=g).to('cuda', torch.bfloat16)
    grads = []
    for _ in range(2):
        wv = w.clone().requires_grad_(True)
        out = _LinearFn.apply(x, wv, b)
        out.backward(dy)
        grads.append(wv.grad.clone())
    assert torch.equal(grads[0], grads[1])    # deterministic split-K reduce
