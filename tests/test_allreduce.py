"""All-reduce DP engine golden test (gloo, world 2): K steps equal a serial
gradient-averaged FlatSGD simulation on the same batches (the same oracle
pattern as test_dist_ps.py)."""
import torch
import torch.nn.functional as F

from ps_pytorch_amd.config import JobConfig
from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.optim import FlatSGD
from ps_pytorch_amd.parallel.flat import FlatSpace

from dist_utils import run_dist

STEPS = 3
BS = 16
SEED = 9
LR, MOM = 0.05, 0.9


def _cfg(overlap: bool) -> JobConfig:
    return JobConfig(network='LeNet', dataset='MNIST', batch_size=BS,
                     lr=LR, momentum=MOM, seed=SEED, max_steps=STEPS,
                     compute_dtype='fp32', overlap=overlap, bucket_mb=0.25,
                     log_interval=10 ** 9, eval_freq=10 ** 9)


def _batches(rank: int):
    g = torch.Generator().manual_seed(400 + rank)
    xs = [torch.randn(BS, 1, 28, 28, generator=g) for _ in range(STEPS)]
    ys = [torch.randint(0, 10, (BS,), generator=g) for _ in range(STEPS)]
    return xs, ys


def _role(rank: int, world: int, port: int, overlap: bool):
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.allreduce import AllReduceTrainer
    env = init_distributed(backend='gloo')
    tr = AllReduceTrainer(_cfg(overlap), rank, world, env['device'])
    tr.build_model(10)
    xs, ys = _batches(rank)
    for i in range(STEPS):
        tr.train_step(xs[i], ys[i])
    return tr.master_w[:tr.flat.total].clone()


def _serial() -> torch.Tensor:
    torch.manual_seed(SEED)
    net = build_model('LeNet', num_classes=10, in_channels=1)
    fs = FlatSpace(net, bucket_bytes=int(0.25 * 2 ** 20))
    fs.attach_grads()
    master = fs.flat_w.detach().to(torch.float32).clone()
    opt = FlatSGD(master, lr=LR, momentum=MOM)
    data = {r: _batches(r) for r in (0, 1)}
    for step in range(STEPS):
        grad_sum = torch.zeros_like(master)
        for r in (0, 1):
            fs.load_flat(master)
            fs.zero_grads()
            xs, ys = data[r]
            F.cross_entropy(net(xs[step]).float(), ys[step]).backward()
            grad_sum += fs.flat_g
        opt.step(grad_sum, grad_scale=0.5)
    return master[:fs.total]


def test_allreduce_golden_overlap():
    res = run_dist(_role, world=2, args=(True,))
    ref = _serial()
    for r in (0, 1):   # every rank holds the same master
        got = torch.from_numpy(res[r])
        assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), \
            (got - ref).abs().max()


def test_allreduce_golden_no_overlap():
    res = run_dist(_role, world=2, args=(False,))
    ref = _serial()
    got = torch.from_numpy(res[0])
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5)
