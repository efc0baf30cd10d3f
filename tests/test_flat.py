"""FlatSpace: view aliasing, bucket framing, grad attachment."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st
import torch.nn.functional as F

from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.parallel.flat import FlatSpace


def test_views_alias_storage():
    m = build_model('LeNet', in_channels=1)
    fs = FlatSpace(m)
    # mutating flat_w must mutate the live parameters
    fs.flat_w.fill_(0.5)
    for p in m.parameters():
        assert torch.all(p.data == 0.5)
    assert fs.num_params == 431080      # LeNet parameter count
    assert fs.total >= fs.num_params    # alignment gaps between params
    assert fs.total - fs.num_params < 8 * len(fs.params)
    assert fs.padded % 8 == 0
    # every param offset 16-B aligned (vectorized kernels)
    for off in fs.offsets:
        assert off % 8 == 0


def test_reverse_order_layout():
    m = build_model('LeNet', in_channels=1)
    fs = FlatSpace(m)
    names = fs.names
    # reverse parameter order: fc2 bias first, conv1 weight last
    assert 'fc2' in names[0]
    assert 'conv1' in names[-1]


def test_bucket_partition_covers_everything():
    m = build_model('ResNet18')
    fs = FlatSpace(m, bucket_bytes=4 * 1024 * 1024)
    assert fs.buckets[0].start == 0
    assert fs.buckets[-1].end == fs.padded
    for a, b in zip(fs.buckets, fs.buckets[1:]):
        assert a.end == b.start
    # bucket boundaries 16-B aligned for the gather-mode vector kernels
    for b in fs.buckets:
        assert b.start % 8 == 0 and b.end % 8 == 0
    covered = set()
    for b in fs.buckets:
        covered.update(b.param_ids)
    assert covered == set(range(len(fs.params)))
    assert len(fs.buckets) > 1


def test_grad_attachment_accumulates_into_flat():
    torch.manual_seed(0)
    m = build_model('LeNet', in_channels=1)
    fs = FlatSpace(m)
    fs.attach_grads()
    x = torch.randn(4, 1, 28, 28)
    y = torch.randint(0, 10, (4,))
    loss = F.cross_entropy(m(x), y)
    loss.backward()
    assert fs.flat_g.abs().sum() > 0
    # p.grad views alias flat_g
    for p, off in zip(fs.params, fs.offsets):
        assert torch.equal(p.grad.reshape(-1), fs.flat_g[off:off + p.numel()])
    fs.zero_grads()
    assert fs.flat_g.abs().sum() == 0
    for p in fs.params:
        assert p.grad.abs().sum() == 0


def test_state_dict_from_flat_roundtrip():
    torch.manual_seed(0)
    m = build_model('LeNet', in_channels=1)
    ref = {k: v.clone() for k, v in m.state_dict().items()}
    fs = FlatSpace(m)
    sd = fs.state_dict_from_flat()
    for k, v in sd.items():
        assert torch.equal(v, ref[k]), k


def test_layout_signature_deterministic():
    torch.manual_seed(0)
    a = FlatSpace(build_model('ResNet18'))
    b = FlatSpace(build_model('ResNet18'))
    assert a.layout_signature() == b.layout_signature()


def _role_layout_mismatch(rank: int, world: int, port: int):
    """Rank 1 builds a different model: transport init must raise (layout
    fingerprint broadcast, transport.py _verify_layout)."""
    import torch
    from ps_pytorch_amd.models import build_model
    from ps_pytorch_amd.parallel.flat import FlatSpace
    from ps_pytorch_amd.parallel.transport import PSTransport, init_distributed
    env = init_distributed(backend='gloo')
    torch.manual_seed(0)
    net = build_model('LeNet' if rank == 0 else 'ResNet18', num_classes=10,
                      in_channels=1 if rank == 0 else 3)
    fs = FlatSpace(net)
    try:
        PSTransport(fs, torch.float32, env['device'], rank, world)
        return 'no error'
    except RuntimeError as e:
        return 'raised' if 'fingerprint' in str(e) else f'other: {e}'


def test_layout_mismatch_raises_at_transport_init():
    from dist_utils import run_dist
    res = run_dist(_role_layout_mismatch, world=2)
    assert res[0] == 'no error'      # rank 0 is the reference fingerprint
    assert res[1] == 'raised', res


@given(
    sizes=st.lists(st.integers(min_value=1, max_value=5000),
                   min_size=1, max_size=20),
    bucket_kb=st.integers(min_value=1, max_value=64),
)
@settings(max_examples=40, deadline=None)
def test_partition_invariants_property(sizes, bucket_kb):
    """For arbitrary parameter sizes and bucket budgets: buckets tile
    [0, padded) contiguously, every boundary is 8-elem aligned, every
    param belongs to exactly one bucket, and no param straddles one."""
    import torch.nn as nn
    from ps_pytorch_amd.parallel.flat import FlatSpace

    class Blob(nn.Module):
        def __init__(self):
            super().__init__()
            for i, n in enumerate(sizes):
                setattr(self, f'p{i}', nn.Parameter(torch.zeros(n)))

    fs = FlatSpace(Blob(), bucket_bytes=bucket_kb * 1024)
    assert fs.padded % 8 == 0
    pos = 0
    seen = []
    for b in fs.buckets:
        assert b.start == pos and b.end > b.start
        assert b.start % 8 == 0
        seen += b.param_ids
        pos = b.end
    assert pos == fs.padded
    assert seen == list(range(len(fs.params)))
    for b in fs.buckets:
        for pid in b.param_ids:
            off = fs.offsets[pid]
            assert b.start <= off
            assert off + fs.params[pid].numel() <= b.end
