"""PsBatchNorm2d: CPU fallback math vs nn.BatchNorm2d, and (gpu) the
hand-written NHWC kernels vs a plain fp32 torch reference."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from ps_pytorch_amd.ops.modules import PsBatchNorm2d


def _torch_ref(x, bn, residual=None, relu=False):
    y = bn(x)
    if residual is not None:
        y = y + residual
    if relu:
        y = F.relu(y)
    return y


@pytest.mark.parametrize('relu', [False, True])
@pytest.mark.parametrize('with_res', [False, True])
def test_cpu_fallback_matches_torch(relu, with_res):
    torch.manual_seed(0)
    m = PsBatchNorm2d(16, relu=relu)
    ref = nn.BatchNorm2d(16)
    ref.load_state_dict({k: v for k, v in m.state_dict().items()})
    x = torch.randn(4, 16, 8, 8, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    res = torch.randn(4, 16, 8, 8, requires_grad=True) if with_res else None
    res2 = (res.detach().clone().requires_grad_(True) if with_res else None)
    y = m(x, residual=res)
    yr = _torch_ref(x2, ref, res2, relu)
    assert torch.allclose(y, yr, atol=1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(m.weight.grad, ref.weight.grad, atol=1e-5)
    assert torch.allclose(m.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(m.running_var, ref.running_var, atol=1e-6)
    if with_res:
        assert torch.allclose(res.grad, res2.grad, atol=1e-6)


def test_eval_mode_uses_running_stats():
    torch.manual_seed(0)
    m = PsBatchNorm2d(8)
    m.running_mean.uniform_(-1, 1)
    m.running_var.uniform_(0.5, 2)
    m.eval()
    x = torch.randn(2, 8, 4, 4)
    y = m(x)
    ref = (x - m.running_mean.view(1, -1, 1, 1)) / torch.sqrt(
        m.running_var.view(1, -1, 1, 1) + m.eps)
    ref = ref * m.weight.view(1, -1, 1, 1) + m.bias.view(1, -1, 1, 1)
    assert torch.allclose(y, ref, atol=1e-5)


@pytest.mark.gpu
@pytest.mark.parametrize('C,HW,N', [(64, 16, 32), (256, 8, 16), (2048, 2, 8)])
@pytest.mark.parametrize('relu', [False, True])
@pytest.mark.parametrize('with_res', [False, True])
def test_gpu_kernel_vs_fp32_reference(C, HW, N, relu, with_res):
    torch.manual_seed(0)
    dev = torch.device('cuda')
    cl = torch.channels_last
    x32 = torch.randn(N, C, HW, HW, device=dev) * 2 + 0.5
    res32 = torch.randn(N, C, HW, HW, device=dev) if with_res else None

    # fp32 torch reference (plain BatchNorm2d + add + relu)
    ref_bn = nn.BatchNorm2d(C).to(dev)
    with torch.no_grad():
        ref_bn.weight.uniform_(0.5, 1.5)
        ref_bn.bias.uniform_(-0.5, 0.5)
    xr = x32.clone().requires_grad_(True)
    rr = res32.clone().requires_grad_(True) if with_res else None
    yr = _torch_ref(xr, ref_bn, rr, relu)
    g32 = torch.randn_like(yr)
    yr.backward(g32)

    # our kernel in bf16 channels_last
    m = PsBatchNorm2d(C, relu=relu).to(dev, torch.bfloat16)
    with torch.no_grad():
        m.weight.copy_(ref_bn.weight)
        m.bias.copy_(ref_bn.bias)
    m._ensure_f32_stats()
    xk = x32.to(torch.bfloat16).contiguous(memory_format=cl).requires_grad_(True)
    rk = (res32.to(torch.bfloat16).contiguous(memory_format=cl).requires_grad_(True)
          if with_res else None)
    yk = m(xk, residual=rk)
    yk.backward(g32.to(torch.bfloat16).contiguous(memory_format=cl))
    torch.cuda.synchronize()

    def close(a, b, tol):
        d = (a.float() - b.float()).abs()
        s = b.float().abs().max().clamp(min=1)
        assert (d.max() / s) < tol, (d.max().item(), s.item())

    close(yk, yr, 2e-2)              # bf16 io => ~1e-2 relative
    close(m.running_mean, ref_bn.running_mean, 2e-2)
    close(m.running_var, ref_bn.running_var, 2e-2)

    if not relu:
        close(xk.grad, xr.grad, 3e-2)
        close(m.weight.grad, ref_bn.weight.grad, 3e-2)
        close(m.bias.grad, ref_bn.bias.grad, 3e-2)
        if with_res:
            close(rk.grad, rr.grad, 2e-2)
    else:
        # With relu, the kernel masks on bf16-rounded y while torch autograd
        # masks on fp32 y — boundary elements legitimately flip, which moves
        # whole O(1) dy terms in/out of every sum. Reference: manual fp32 BN
        # backward computed WITH the kernel's own mask.
        Mcnt = x32.shape[0] * x32.shape[2] * x32.shape[3]
        mean = x32.mean(dim=(0, 2, 3), keepdim=True)
        var = x32.var(dim=(0, 2, 3), unbiased=False, keepdim=True)
        invstd = (var + m.eps).rsqrt()
        xhat = (x32 - mean) * invstd
        dy_eff = torch.where(yk.float() > 0, g32, torch.zeros_like(g32))
        dgamma = (dy_eff * xhat).sum(dim=(0, 2, 3))
        dbeta = dy_eff.sum(dim=(0, 2, 3))
        gam = ref_bn.weight.detach().view(1, -1, 1, 1)
        dx_ref = (gam * invstd) * (
            dy_eff - dbeta.view(1, -1, 1, 1) / Mcnt
            - xhat * dgamma.view(1, -1, 1, 1) / Mcnt)
        close(xk.grad, dx_ref, 3e-2)
        close(m.weight.grad, dgamma, 3e-2)
        close(m.bias.grad, dbeta, 3e-2)
        if with_res:
            close(rk.grad, dy_eff, 2e-2)


@pytest.mark.gpu
def test_gpu_kernel_f32_exactish():
    # f32 io: kernel must match torch reference tightly
    torch.manual_seed(1)
    dev = torch.device('cuda')
    cl = torch.channels_last
    C = 128
    x = (torch.randn(8, C, 8, 8, device=dev)
         .contiguous(memory_format=cl).requires_grad_(True))
    m = PsBatchNorm2d(C, relu=True).to(dev)
    ref = nn.BatchNorm2d(C).to(dev)
    x2 = x.detach().clone().requires_grad_(True)
    y = m(x)
    yr = F.relu(ref(x2))
    g = torch.randn_like(yr)
    y.backward(g.contiguous(memory_format=cl))
    yr.backward(g)
    torch.cuda.synchronize()
    assert torch.allclose(y, yr, atol=1e-4), (y - yr).abs().max()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(m.weight.grad, ref.weight.grad, atol=1e-3)


@pytest.mark.gpu
def test_bn_eval_fused_matches_torch():
    """Eval-mode fused BN (running-stats normalize + residual + relu) vs
    torch eval batch_norm."""
    torch.manual_seed(11)
    m = PsBatchNorm2d(64, relu=True).cuda().to(torch.bfloat16)
    m.running_mean.data = torch.randn(64, device='cuda') * 0.1
    m.running_var.data = torch.rand(64, device='cuda') + 0.5
    m.eval()
    x = torch.randn(8, 64, 16, 16, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    res = torch.randn_like(x)
    with torch.no_grad():
        y = m(x, residual=res)
        ref = F.relu(F.batch_norm(
            x.float(), m.running_mean, m.running_var, m.weight.float(),
            m.bias.float(), False, 0.1, m.eps) + res.float())
    err = (y.float() - ref).abs().max() / ref.abs().max().clamp_min(1e-6)
    assert float(err) < 0.03, float(err)
