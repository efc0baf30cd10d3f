"""Single-machine trainer: loss decreases on the synthetic problem
(the correctness oracle path, ref nn_ops.py / single_machine.py)."""
import torch

from ps_pytorch_amd.config import JobConfig
from ps_pytorch_amd.data import prepare_data
from ps_pytorch_amd.trainer import NNTrainer


def test_lenet_mnist_loss_decreases():
    cfg = JobConfig(network='LeNet', dataset='MNIST', batch_size=64,
                    lr=0.05, momentum=0.9, max_steps=30, log_interval=1000)
    tr = NNTrainer(cfg, device=torch.device('cpu'))
    tr.build_model()
    train_loader, _ = prepare_data(cfg, device=tr.device, train_size=2048,
                                   test_size=256)
    losses = []
    for i, (x, y) in enumerate(train_loader):
        if i >= 30:
            break
        losses.append(tr.train_step(x, y))
    assert losses[-1] < losses[0] * 0.8, losses[:3] + losses[-3:]


def test_validate_runs():
    cfg = JobConfig(network='LeNet', dataset='MNIST', batch_size=64,
                    test_batch_size=128)
    tr = NNTrainer(cfg, device=torch.device('cpu'))
    tr.build_model()
    _, test_loader = prepare_data(cfg, device=tr.device, train_size=128,
                                  test_size=256)
    acc = tr.validate(test_loader)
    assert 0.0 <= acc <= 100.0


def test_single_machine_adam_optimizer():
    """--optimizer adam wires FlatAdam into the single-machine engine."""
    import torch
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.optim import FlatAdam
    from ps_pytorch_amd.trainer import NNTrainer
    cfg = JobConfig(network='LeNet', dataset='MNIST', batch_size=8,
                    lr=1e-3, optimizer='adam', compute_dtype='fp32')
    tr = NNTrainer(cfg, device=torch.device('cpu'))
    tr.build_model()
    assert isinstance(tr.optimizer, FlatAdam)
    x = torch.randn(8, 1, 28, 28)
    y = torch.randint(0, 10, (8,))
    before = tr.master_w.clone()
    tr.train_step(x, y)
    assert not torch.equal(before, tr.master_w)
