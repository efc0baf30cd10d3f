"""Distributed PS protocol over gloo (CPU, world_size 3): golden-step test —
after K synchronous steps, the PS master weights equal a serial simulation
of gradient-averaged SGD on the same per-worker batches (SURVEY.md §4:
"PS aggregation == local DP average")."""
import torch
import torch.nn.functional as F

from ps_pytorch_amd.config import JobConfig
from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.optim import FlatSGD
from ps_pytorch_amd.parallel.flat import FlatSpace

from dist_utils import run_dist

STEPS = 3
BS = 16
SEED = 5
LR, MOM = 0.1, 0.9


def _cfg(overlap: bool, comm_type: str = 'Bcast') -> JobConfig:
    return JobConfig(network='LeNet', dataset='MNIST', batch_size=BS,
                     lr=LR, momentum=MOM, seed=SEED, max_steps=STEPS,
                     compress_grad='None', wire_dtype='fp32',
                     compute_dtype='fp32', overlap=overlap,
                     comm_type=comm_type,
                     bucket_mb=0.25,    # several buckets even for LeNet
                     log_interval=10 ** 9, eval_freq=10 ** 9)


def _worker_batches(rank: int):
    g = torch.Generator().manual_seed(1000 + rank)
    xs = [torch.randn(BS, 1, 28, 28, generator=g) for _ in range(STEPS)]
    ys = [torch.randint(0, 10, (BS,), generator=g) for _ in range(STEPS)]
    return xs, ys


def _role(rank: int, world: int, port: int, overlap: bool,
          comm_type: str = 'Bcast'):
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    cfg = _cfg(overlap, comm_type)
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        for _ in range(STEPS):
            ps.step()
        return ps.master_w[:ps.flat.total].clone()
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    xs, ys = _worker_batches(rank)
    for i in range(STEPS):
        w.train_step(xs[i], ys[i])
    return None


def _serial_reference() -> torch.Tensor:
    torch.manual_seed(SEED)
    net = build_model('LeNet', num_classes=10, in_channels=1)
    fs = FlatSpace(net)
    fs.attach_grads()
    master = fs.flat_w.detach().to(torch.float32).clone()
    opt = FlatSGD(master, lr=LR, momentum=MOM)
    data = {r: _worker_batches(r) for r in (1, 2)}
    for step in range(STEPS):
        grad_sum = torch.zeros_like(master)
        for r in (1, 2):
            fs.load_flat(master)
            fs.zero_grads()
            xs, ys = data[r]
            loss = F.cross_entropy(net(xs[step]).float(), ys[step])
            loss.backward()
            grad_sum += fs.flat_g
        opt.step(grad_sum, grad_scale=0.5)
    return master[:fs.total]


def test_golden_step_no_overlap():
    results = run_dist(_role, world=3, args=(False,))
    got = torch.from_numpy(results[0])
    ref = _serial_reference()
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), \
        (got - ref).abs().max()


def test_golden_step_with_overlap():
    results = run_dist(_role, world=3, args=(True,))
    got = torch.from_numpy(results[0])
    ref = _serial_reference()
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), \
        (got - ref).abs().max()


def test_golden_step_async_comm_type():
    """--comm-type Async: P2P weight fan-out instead of broadcast
    (ref distributed_worker.py:201-219) — same golden result."""
    results = run_dist(_role, world=3, args=(False, 'Async'))
    got = torch.from_numpy(results[0])
    ref = _serial_reference()
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), \
        (got - ref).abs().max()


def test_bcast_pipeline_mode_matrix(monkeypatch):
    """The pipelined per-bucket update/broadcast engages exactly for
    collective + Bcast (and can be killed with PS_BCAST_PIPE=0); gather and
    Async keep the single-collective broadcast."""
    import torch
    from ps_pytorch_amd.models import build_model
    from ps_pytorch_amd.parallel.flat import FlatSpace
    from ps_pytorch_amd.parallel.transport import PSTransport
    fs = FlatSpace(build_model('LeNet', in_channels=1))
    dev = torch.device('cpu')

    def mk(**kw):
        return PSTransport(fs, torch.float32, dev, 0, 1, **kw)

    assert mk(mode='collective', comm_type='Bcast').bcast_bucketed
    assert not mk(mode='gather', comm_type='Bcast').bcast_bucketed
    assert not mk(mode='collective', comm_type='Async').bcast_bucketed
    monkeypatch.setenv('PS_BCAST_PIPE', '0')
    assert not mk(mode='collective', comm_type='Bcast').bcast_bucketed


def test_golden_step_bcast_pipeline_off(monkeypatch):
    """PS_BCAST_PIPE=0 (single-collective broadcast) must produce the same
    golden master as the default pipelined per-bucket path."""
    monkeypatch.setenv('PS_BCAST_PIPE', '0')
    res = run_dist(_role, world=3, args=(True,))
    got = torch.from_numpy(res[0])
    ref = _serial_reference()
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), (got - ref).abs().max()


def _role_adam(rank: int, world: int, port: int):
    """PS with --optimizer adam (the reference ships optim/adam.py but
    hardwires SGD; here the flag wires the fused flat Adam in)."""
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    cfg = _cfg(True)
    cfg.optimizer = 'adam'
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        for _ in range(STEPS):
            ps.step()
        return ps.master_w[:ps.flat.total].clone()
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    xs, ys = _worker_batches(rank)
    for i in range(STEPS):
        w.train_step(xs[i], ys[i])
    return None


def test_golden_step_adam_optimizer():
    from ps_pytorch_amd.optim import FlatAdam
    results = run_dist(_role_adam, world=3)
    got = torch.from_numpy(results[0])
    # serial reference with FlatAdam, same per-worker batches
    torch.manual_seed(SEED)
    net = build_model('LeNet', num_classes=10, in_channels=1)
    fs = FlatSpace(net)
    fs.attach_grads()
    master = fs.flat_w.detach().to(torch.float32).clone()
    opt = FlatAdam(master, lr=LR)
    data = {r: _worker_batches(r) for r in (1, 2)}
    for step in range(STEPS):
        grad_sum = torch.zeros_like(master)
        for r in (1, 2):
            fs.load_flat(master)
            fs.zero_grads()
            xs, ys = data[r]
            F.cross_entropy(net(xs[step]).float(), ys[step]).backward()
            grad_sum += fs.flat_g
        opt.step(grad_sum, grad_scale=0.5)
    ref = master[:fs.total]
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), \
        (got - ref).abs().max()
