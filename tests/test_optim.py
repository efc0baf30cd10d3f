"""FlatSGD/FlatAdam numerics vs torch.optim on the same flat problem."""
import pytest
import torch

from ps_pytorch_amd.optim import FlatSGD, FlatAdam


def _run_flat_vs_torch(opt_name, steps=5, **kw):
    torch.manual_seed(7)
    n = 1003
    w0 = torch.randn(n)
    grads = [torch.randn(n) for _ in range(steps)]

    # torch reference
    p = torch.nn.Parameter(w0.clone())
    if opt_name == 'sgd':
        topt = torch.optim.SGD([p], lr=kw['lr'], momentum=kw.get('momentum', 0),
                               weight_decay=kw.get('weight_decay', 0),
                               nesterov=kw.get('nesterov', False))
    else:
        topt = torch.optim.Adam([p], lr=kw['lr'],
                                weight_decay=kw.get('weight_decay', 0),
                                amsgrad=kw.get('amsgrad', False))
    for g in grads:
        p.grad = g.clone()
        topt.step()

    # flat implementation
    w = w0.clone()
    if opt_name == 'sgd':
        fopt = FlatSGD(w, **kw)
    else:
        fopt = FlatAdam(w, **kw)
    for g in grads:
        fopt.step(g.clone())
    return p.data, w


@pytest.mark.parametrize('kw', [
    dict(lr=0.1),
    dict(lr=0.1, momentum=0.9),
    dict(lr=0.1, momentum=0.9, weight_decay=1e-4),
    dict(lr=0.1, momentum=0.9, nesterov=True),
])
def test_flat_sgd_matches_torch(kw):
    ref, got = _run_flat_vs_torch('sgd', **kw)
    assert torch.allclose(ref, got, atol=1e-6, rtol=1e-5)


@pytest.mark.parametrize('kw', [
    dict(lr=1e-3),
    dict(lr=1e-3, weight_decay=1e-4),
    dict(lr=1e-3, amsgrad=True),
])
def test_flat_adam_matches_torch(kw):
    ref, got = _run_flat_vs_torch('adam', **kw)
    assert torch.allclose(ref, got, atol=1e-6, rtol=1e-5)


def test_grad_scale_is_average():
    # grad_scale = 1/W averages the summed gradients (PS semantics,
    # ref sync_replicas_master_nn.py:207)
    torch.manual_seed(0)
    n = 64
    w = torch.randn(n)
    w2 = w.clone()
    g1, g2 = torch.randn(n), torch.randn(n)
    a = FlatSGD(w, lr=0.5)
    a.step(g1 + g2, grad_scale=0.5)
    b = FlatSGD(w2, lr=0.5)
    b.step((g1 + g2) / 2)
    assert torch.allclose(w, w2, atol=1e-7)


def test_fused_wire_out():
    from ps_pytorch_amd.ops.functional import fused_sgd_step
    n = 32
    w = torch.randn(n)
    g = torch.randn(n)
    m = torch.zeros(n)
    wire = torch.empty(n)
    fused_sgd_step(w, g, m, lr=0.1, momentum=0.9, wire_out=wire)
    assert torch.equal(wire, w)


@pytest.mark.parametrize('opt_name', ['sgd', 'adam'])
def test_region_sliced_steps_equal_full_step(opt_name):
    """The pipelined PS updates bucket slices with region=(start, end) and
    advance only on the first slice (parallel/ps.py _step_pipelined); the
    result must be identical to one whole-buffer step."""
    import torch
    from ps_pytorch_amd.optim import FlatAdam, FlatSGD
    torch.manual_seed(17)
    n = 1000
    g = torch.randn(n)
    cuts = [0, 130, 512, 768, n]

    def make(name, w):
        return (FlatAdam(w, lr=1e-3) if name == 'adam'
                else FlatSGD(w, lr=0.1, momentum=0.9, weight_decay=1e-4))

    w_full = torch.randn(n)
    w_sliced = w_full.clone()
    full = make(opt_name, w_full)
    sliced = make(opt_name, w_sliced)
    for it in range(3):
        full.step(g, grad_scale=0.5)
        for i in range(len(cuts) - 1):
            sliced.step(g, grad_scale=0.5, region=(cuts[i], cuts[i + 1]),
                        advance=(i == 0))
        g = g.roll(7) * 0.9   # vary grads across iterations
    assert torch.equal(w_full, w_sliced)
