"""GPU q8 codec vs the CPU reference: bitwise-identical payloads and
matching decode/accumulate (both sides use rint / round-half-even and the
same reduction values)."""
import pytest
import torch

from ps_pytorch_amd.ops import functional as F

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("n", [257, 4096, 1 << 20, (1 << 20) + 13])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_q8_pack_bitwise_matches_cpu(n, dtype):
    g = torch.Generator().manual_seed(n)
    x = (torch.randn(n, generator=g) * 0.01).to(dtype)
    _, tot = F.q8_layout(n)
    p_cpu = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(p_cpu, x)
    p_gpu = torch.zeros(tot, dtype=torch.uint8, device='cuda')
    F.pack_q8(p_gpu, x.cuda())
    assert torch.equal(p_gpu.cpu(), p_cpu)


def test_q8_unpack_and_acc_match_cpu():
    n = 300000
    g = torch.Generator().manual_seed(1)
    x = torch.randn(n, generator=g)
    _, tot = F.q8_layout(n)
    p = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(p, x)
    y_cpu = torch.empty(n)
    F.unpack_q8(y_cpu, p)
    y_gpu = torch.empty(n, device='cuda')
    F.unpack_q8(y_gpu, p.cuda())
    assert torch.equal(y_gpu.cpu(), y_cpu)
    base = torch.randn(n, generator=g)
    a_cpu = base.clone()
    F.unpack_q8(a_cpu, p, accumulate=True)
    a_gpu = base.cuda()
    F.unpack_q8(a_gpu, p.cuda(), accumulate=True)
    assert torch.allclose(a_gpu.cpu(), a_cpu, atol=1e-6, rtol=0)


def test_acc_into_gpu():
    n = 123457
    g = torch.Generator().manual_seed(2)
    a = torch.randn(n, generator=g)
    b = torch.randn(n, generator=g).to(torch.bfloat16)
    acc = a.cuda()
    F.acc_into(acc, b.cuda())
    assert torch.allclose(acc.cpu(), a + b.float(), atol=1e-6, rtol=0)
