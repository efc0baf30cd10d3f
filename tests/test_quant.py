"""Block-scaled int8 codec (the blosc-role compressor, ops/kernels/quant.hip):
CPU-reference numerics + payload layout. The GPU-vs-CPU bitwise test lives in
test_quant_gpu.py (gpu marker)."""
import pytest
import torch

from ps_pytorch_amd.ops import functional as F


@pytest.mark.parametrize("n", [1, 3, 255, 256, 257, 1024, 4096 + 7, 100000])
def test_q8_roundtrip_error_bound(n):
    g = torch.Generator().manual_seed(n)
    x = torch.randn(n, generator=g) * torch.rand(1, generator=g).exp()
    _, tot = F.q8_layout(n)
    payload = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(payload, x)
    y = torch.empty(n)
    F.unpack_q8(y, payload)
    # per-block max error <= scale/2 = max|x_block| / 254
    xb = torch.zeros(((n + 255) // 256) * 256)
    xb[:n] = x
    m = xb.view(-1, 256).abs().amax(dim=1)
    bound = (m / 254.0 + 1e-8).repeat_interleave(256)[:n]
    assert ((y - x).abs() <= bound).all()


def test_q8_payload_is_4x():
    n = 1 << 20
    _, tot = F.q8_layout(n)
    assert tot < 4 * n * 0.26 + 64   # ~4x smaller than f32


def test_q8_zero_block_and_accumulate():
    n = 600
    x = torch.zeros(n)
    x[300:] = torch.linspace(-2, 2, 300)
    _, tot = F.q8_layout(n)
    payload = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(payload, x)
    acc = torch.ones(n)
    F.unpack_q8(acc, payload, accumulate=True)
    y = torch.empty(n)
    F.unpack_q8(y, payload)
    assert torch.equal(acc, 1.0 + y)
    assert torch.equal(y[:256], torch.zeros(256))   # all-zero block stays exact


def test_q8_extremes_map_to_127():
    x = torch.zeros(256)
    x[0], x[77] = 3.0, -3.0
    _, tot = F.q8_layout(256)
    payload = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(payload, x)
    q = payload[:256].view(torch.int8)
    assert q[0] == 127 and q[77] == -127
    y = torch.empty(256)
    F.unpack_q8(y, payload)
    assert torch.allclose(y[0], torch.tensor(3.0))


def test_q8_bf16_source_matches_f32_of_same_values():
    g = torch.Generator().manual_seed(7)
    xb = torch.randn(1000, generator=g).to(torch.bfloat16)
    xf = xb.to(torch.float32)
    _, tot = F.q8_layout(1000)
    p1 = torch.zeros(tot, dtype=torch.uint8)
    p2 = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(p1, xb)
    F.pack_q8(p2, xf)
    assert torch.equal(p1, p2)


def test_acc_into():
    g = torch.Generator().manual_seed(3)
    a = torch.randn(501, generator=g)
    b = torch.randn(501, generator=g)
    acc = a.clone()
    F.acc_into(acc, b)
    assert torch.allclose(acc, a + b)
    bb = b.to(torch.bfloat16)
    acc = a.clone()
    F.acc_into(acc, bb)
    assert torch.allclose(acc, a + bb.float())


# property test: q8 roundtrip error bound holds for arbitrary sizes/scales
try:
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=30, deadline=None)
    @given(n=st.integers(min_value=1, max_value=5000),
           scale=st.floats(min_value=1e-6, max_value=1e3),
           seed=st.integers(min_value=0, max_value=2 ** 31 - 1))
    def test_q8_roundtrip_bound_hypothesis(n, scale, seed):
        from ps_pytorch_amd.ops import functional as F
        g = torch.Generator().manual_seed(seed)
        x = torch.randn(n, generator=g) * scale
        _, tot = F.q8_layout(n)
        p = torch.zeros(tot, dtype=torch.uint8)
        F.pack_q8(p, x)
        y = torch.empty(n)
        F.unpack_q8(y, p)
        # per-256-block error <= half a quantization step of that block
        xb = torch.zeros(((n + 255) // 256) * 256)
        xb[:n] = x
        step = xb.view(-1, 256).abs().amax(dim=1) / 127.0
        bound = step.repeat_interleave(256)[:n] * 0.5 + 1e-12
        assert torch.all((y - x).abs() <= bound + 1e-6 * x.abs())
except ImportError:       # pragma: no cover
    pass


def test_hip_library_loads_and_symbols_resolve():
    """The in-tree libps_hip.so must load (ROCm runtime links on CPU-only
    hosts too) with every declared entry point present — catches
    symbol/signature drift at CPU-test time instead of on a GPU box.
    Skips only if the .so was never built in this checkout."""
    import os
    import pytest
    from ps_pytorch_amd.ops import _so_path, lib_or_none
    if not os.path.exists(_so_path()):
        pytest.skip("libps_hip.so not built")
    lib = lib_or_none()
    assert lib is not None, "built .so failed to load"
    for sym in ('ps_fused_sgd', 'ps_fused_adam', 'ps_conv_fwd',
                'ps_conv_dgrad', 'ps_conv_wgrad', 'ps_conv_bias_grad',
                'ps_wt_transpose', 'ps_bn_fwd', 'ps_bn_bwd',
                'ps_softmax_ce_fwd', 'ps_softmax_ce_bwd', 'ps_maxpool_fwd',
                'ps_maxpool_bwd', 'ps_gavgpool_fwd', 'ps_gavgpool_bwd',
                'ps_pack_bf16', 'ps_unpack_bf16', 'ps_pack_q8',
                'ps_unpack_q8', 'ps_acc', 'ps_pad4', 'ps_padc'):
        assert getattr(lib, sym, None) is not None, sym
