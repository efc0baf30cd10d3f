"""Steal-mode gradient plumbing (parallel/flat.py attach_grads(steal=True)):
Ps ops write weight/bias grads directly into flat_g slices; fallback ops are
caught by ensure_grad_in_flat/harvest_grads pointer checks."""
import warnings

import pytest
import torch
import torch.nn.functional as F

from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.parallel.flat import FlatSpace


def _grads_ref(model_name, x, y, **kw):
    torch.manual_seed(0)
    net = build_model(model_name, **kw)
    fs = FlatSpace(net)
    fs.attach_grads()          # view mode: autograd accumulates into flat_g
    loss = F.cross_entropy(net(x), y)
    loss.backward()
    return fs.flat_g.clone()


def test_steal_harvest_cpu_fallback_matches_view_mode():
    """On CPU every op is a torch fallback, so steal mode relies entirely on
    the harvest copy path; the result must equal view-mode accumulation."""
    x = torch.randn(4, 1, 28, 28, generator=torch.Generator().manual_seed(3))
    y = torch.randint(0, 10, (4,), generator=torch.Generator().manual_seed(4))
    ref = _grads_ref('LeNet', x, y, in_channels=1)

    torch.manual_seed(0)
    net = build_model('LeNet', in_channels=1)
    fs = FlatSpace(net)
    fs.attach_grads(steal=True)
    for p in fs.params:
        assert p.grad is None and hasattr(p, '_ps_flat_grad_fn')
    loss = F.cross_entropy(net(x), y)
    loss.backward()
    # grads landed in autograd-owned tensors, flat_g still zero
    assert fs.flat_g.abs().sum() == 0
    with warnings.catch_warnings(record=True) as rec:
        warnings.simplefilter('always')
        fs.harvest_grads()
    assert any('outside flat_g' in str(w.message) for w in rec)
    assert torch.allclose(fs.flat_g, ref, atol=1e-6)
    # zero_grads resets p.grad to None for the next step
    fs.zero_grads()
    assert all(p.grad is None for p in fs.params)
    assert fs.flat_g.abs().sum() == 0


@pytest.mark.gpu
def test_steal_gpu_kernels_write_in_place():
    """On GPU the Ps ops must land every param grad in flat_g themselves —
    pointer-verified (no foreign-grad copies), values vs view mode."""
    dev = torch.device('cuda')
    x = torch.randn(8, 3, 32, 32, generator=torch.Generator().manual_seed(5)) \
        .to(dev, torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (8,), generator=torch.Generator().manual_seed(6)).to(dev)

    def run(steal):
        torch.manual_seed(0)
        net = build_model('ResNet18')
        from ps_pytorch_amd.parallel.flat import prep_model
        net = prep_model(net, dev, torch.bfloat16)
        fs = FlatSpace(net)
        fs.attach_grads(steal=steal)
        from ps_pytorch_amd.ops.loss import cross_entropy
        loss = cross_entropy(net(x), y)
        loss.backward()
        fs.harvest_grads()
        return fs, net

    fs_s, _ = run(True)
    # every param's grad must have landed at its flat slice pointer (the
    # kernels wrote in place — no silent fallback)
    for pid, p in enumerate(fs_s.params):
        assert p.grad is not None
        assert p.grad.data_ptr() == fs_s._gptrs[pid], fs_s.names[pid]
    fs_v, _ = run(False)
    assert torch.equal(fs_s.flat_g, fs_v.flat_g)
