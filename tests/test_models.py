"""Model zoo shape/param-count parity (ref counts: SURVEY.md §2.1 row 6)."""
import pytest
import torch

from ps_pytorch_amd.models import build_model


def n_params(m):
    return sum(p.numel() for p in m.parameters())


def test_lenet_param_count():
    m = build_model('LeNet', num_classes=10, in_channels=1)
    assert n_params(m) == 431080               # ref lenet.py:16-37
    assert len(list(m.parameters())) == 8


def test_resnet18_param_count():
    m = build_model('ResNet18', num_classes=10)
    assert n_params(m) == 11173962             # ref resnet.py CIFAR variant
    assert len(list(m.parameters())) == 62


def test_resnet_alias():
    m = build_model('ResNet', num_classes=10)
    assert n_params(m) == 11173962


@pytest.mark.parametrize('name,shape,nc', [
    ('LeNet', (2, 1, 28, 28), 10),
    ('ResNet18', (2, 3, 32, 32), 10),
    ('ResNet34', (2, 3, 32, 32), 100),
    ('ResNet50', (2, 3, 32, 32), 10),
    ('VGG11', (2, 3, 32, 32), 10),
    ('VGG16_BN', (2, 3, 32, 32), 10),
])
def test_forward_shapes(name, shape, nc):
    in_ch = shape[1]
    m = build_model(name, num_classes=nc, in_channels=in_ch)
    y = m(torch.randn(*shape))
    assert y.shape == (shape[0], nc)


def test_resnet50_imagenet_stem():
    m = build_model('ResNet50', num_classes=1000)
    y = m(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)


def test_unknown_network():
    with pytest.raises(ValueError):
        build_model('AlexNet')


def test_psconv_fallback_slices_prepadded_input():
    """On the torch fallback path (CPU here), a pre-padded input from an
    out_pad producer upstream is sliced back to in_channels (ops/conv.py
    PsConv2d.forward) so CPU flows stay at nominal channel counts."""
    import torch
    from ps_pytorch_amd.ops.conv import PsConv2d
    torch.manual_seed(2)
    conv = PsConv2d(20, 50, kernel_size=5)
    x = torch.randn(2, 24, 12, 12)        # 4 pad channels of garbage
    out_pad = conv(x)
    out_ref = conv(x[:, :20])
    assert out_pad.shape == out_ref.shape == (2, 50, 8, 8)
    assert torch.equal(out_pad, out_ref)


def test_pad_target_modes():
    from ps_pytorch_amd.ops import conv as C
    assert C._pad_target(20) == 64 and C._pad_target(50) == 64
    assert C._pad_target(500) == 512 and C._pad_target(64) == 64
