"""Gather-mode PS protocol over gloo (CPU): arrival-order aggregation,
--num-aggregate selection, q8-compressed payloads, and the straggler kill /
timeout abort protocols (ref sync_replicas_master_nn.py:157-207,
resnet_split.py:503-728)."""
import time

import torch
import torch.nn.functional as F

from ps_pytorch_amd.config import JobConfig
from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.optim import FlatSGD
from ps_pytorch_amd.parallel.flat import FlatSpace
from ps_pytorch_amd.ops import functional as ops_f

from dist_utils import run_dist

STEPS = 3
BS = 16
SEED = 5
LR, MOM = 0.1, 0.9


def _cfg(**kw) -> JobConfig:
    base = dict(network='LeNet', dataset='MNIST', batch_size=BS,
                lr=LR, momentum=MOM, seed=SEED, max_steps=STEPS,
                compress_grad='None', wire_dtype='fp32', compute_dtype='fp32',
                overlap=True, aggregation='gather', num_aggregate=64,
                bucket_mb=0.25, log_interval=10 ** 9, eval_freq=10 ** 9)
    base.update(kw)
    return JobConfig(**base)


def _batches(rank: int, same: bool = False):
    g = torch.Generator().manual_seed(1000 + (0 if same else rank))
    xs = [torch.randn(BS, 1, 28, 28, generator=g) for _ in range(STEPS)]
    ys = [torch.randint(0, 10, (BS,), generator=g) for _ in range(STEPS)]
    return xs, ys


def _role(rank: int, world: int, port: int, cfg_kw: dict, same_data: bool,
          slow_rank: int = -1, slow_s: float = 0.0):
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    cfg = _cfg(**cfg_kw)
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        for _ in range(STEPS):
            ps.step()
        return ps.master_w[:ps.flat.total].clone()
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    if rank == slow_rank and slow_s:
        # straggle INSIDE the step (after the synchronizing weight
        # broadcast): slow forward, so the PS hits quota from the fast
        # worker and signals while this rank is still computing
        w.network.register_forward_pre_hook(
            lambda *a: time.sleep(slow_s))
    xs, ys = _batches(rank, same=same_data)
    kills = 0
    for i in range(STEPS):
        loss = w.train_step(xs[i], ys[i])
        if loss is None:
            kills += 1
    return kills


def _serial(grad_fn, workers=(1, 2), same_data: bool = False,
            scale: float = None) -> torch.Tensor:
    """Serial simulation: per-step summed (optionally codec-roundtripped)
    worker grads -> FlatSGD. grad_fn maps a worker's flat grad to its wire
    contribution."""
    torch.manual_seed(SEED)
    net = build_model('LeNet', num_classes=10, in_channels=1)
    fs = FlatSpace(net, bucket_bytes=int(0.25 * 2 ** 20))  # match _cfg bucket_mb
    fs.attach_grads()
    master = fs.flat_w.detach().to(torch.float32).clone()
    opt = FlatSGD(master, lr=LR, momentum=MOM)
    data = {r: _batches(r, same=same_data) for r in workers}
    scale = scale if scale is not None else 1.0 / len(workers)
    for step in range(STEPS):
        grad_sum = torch.zeros_like(master)
        for r in workers:
            fs.load_flat(master)
            fs.zero_grads()
            xs, ys = data[r]
            loss = F.cross_entropy(net(xs[step]).float(), ys[step])
            loss.backward()
            grad_sum += grad_fn(fs.flat_g.clone(), fs)
        opt.step(grad_sum, grad_scale=scale)
    return master[:fs.total]


def _identity(g, fs):
    return g


def _q8_roundtrip(g, fs):
    """Exactly what the wire does: per-bucket q8 encode/decode."""
    out = torch.zeros_like(g)
    for b in fs.buckets:
        n = b.numel
        _, tot = ops_f.q8_layout(n)
        payload = torch.zeros(tot, dtype=torch.uint8)
        ops_f.pack_q8(payload, g[b.start:b.end])
        ops_f.unpack_q8(out[b.start:b.end], payload)
    return out


def test_gather_golden_uncompressed():
    res = run_dist(_role, world=3, args=({}, False))
    got = torch.from_numpy(res[0])
    ref = _serial(_identity)
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), (got - ref).abs().max()


def test_gather_golden_q8_compressed():
    res = run_dist(_role, world=3, args=({'compress_grad': 'compress'}, False))
    got = torch.from_numpy(res[0])
    ref = _serial(_q8_roundtrip)
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), (got - ref).abs().max()


def test_gather_num_aggregate_first_k():
    """num_aggregate=1 with identical worker batches: whichever arrival wins,
    the update equals the single-worker serial reference."""
    res = run_dist(_role, world=3,
                   args=({'num_aggregate': 1}, True))
    got = torch.from_numpy(res[0])
    ref = _serial(_identity, workers=(1,), same_data=True, scale=1.0)
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), (got - ref).abs().max()


def test_kill_mode_aborts_straggler():
    """kill mode + num_aggregate=1: the PS reaches quota from the fast worker
    while rank 2 sleeps; rank 2 must see the signal and abort its backward
    (train_step returns None) on every step; training stays live."""
    res = run_dist(_role, world=3,
                   args=({'num_aggregate': 1, 'mode': 'kill'},
                         True, 2, 1.5),
                   timeout=300.0)
    assert res[2] >= 1, f"straggler was never killed: {res}"
    assert res[0] is not None    # PS finished all steps


def test_timeout_mode_aborts_locally():
    """timeout mode with an instant threshold: every worker aborts at the
    first backward hook and pushes zero payloads; the PS still completes."""
    res = run_dist(_role, world=3,
                   args=({'mode': 'timeout', 'kill_threshold': 0.0}, False))
    assert res[1] == STEPS and res[2] == STEPS
    assert res[0] is not None


def test_kill_with_collective_normalizes_to_gather():
    """--mode kill with --aggregation collective would deadlock (the PS's
    kill verdict only fires from the gather drain); JobConfig normalizes the
    combination to gather with a warning."""
    import warnings
    with warnings.catch_warnings(record=True) as rec:
        warnings.simplefilter('always')
        cfg = _cfg(mode='kill', aggregation='collective')
    assert cfg.aggregation == 'gather'
    assert any('gather' in str(w.message) for w in rec)
    # the valid combination stays untouched, no warning
    with warnings.catch_warnings(record=True) as rec:
        warnings.simplefilter('always')
        cfg = _cfg(mode='kill', aggregation='gather')
    assert cfg.aggregation == 'gather' and not rec


def _role_one_timeout(rank: int, world: int, port: int, cfg_kw: dict):
    """Like _role, but ONLY rank 2 gets an instant timeout threshold."""
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    kw = dict(cfg_kw)
    if rank == 2:
        kw['kill_threshold'] = 0.0
    cfg = _cfg(**kw)
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        for _ in range(STEPS):
            ps.step()
        return ps.master_w[:ps.flat.total].clone()
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    xs, ys = _batches(rank, same=True)
    kills = 0
    for i in range(STEPS):
        if w.train_step(xs[i], ys[i]) is None:
            kills += 1
    return kills


def test_timeout_killed_worker_excluded_from_quota():
    """A timeout-aborted worker's zero payloads must NOT occupy
    --num-aggregate quota slots or dilute the average: with
    rank 2 instantly aborting and num_aggregate=2, the update must equal
    the single-real-worker serial reference exactly (count-corrected
    scaling), not a half-diluted one."""
    res = run_dist(_role_one_timeout, world=3,
                   args=({'mode': 'timeout', 'kill_threshold': 30.0,
                          'num_aggregate': 2},))
    assert res[2] == STEPS            # rank 2 aborted every step
    assert res[1] == 0                # rank 1 never aborted
    got = torch.from_numpy(res[0])
    ref = _serial(_identity, workers=(1,), same_data=True, scale=1.0)
    assert torch.allclose(got, ref, atol=1e-5, rtol=1e-5), (got - ref).abs().max()


def _role_dead_worker(rank: int, world: int, port: int):
    """Worker 2 'crashes' (never pushes); PS must raise the drain-liveness
    error instead of hanging forever (PS_DRAIN_TIMEOUT)."""
    import os
    os.environ['PS_DRAIN_TIMEOUT'] = '3'
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    cfg = _cfg(max_steps=1)
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        try:
            ps.step()
        except RuntimeError as e:
            return 'timeout' if 'stalled' in str(e) else f'other: {e}'
        return 'no-error'
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    if rank == 1:
        xs, ys = _batches(rank)
        try:
            w.train_step(xs[0], ys[0])   # blocks in wait_all (PS died) — ok
        except Exception:
            pass
    # rank 2: participate in the weight broadcast, then go silent
    elif rank == 2:
        try:
            w.fetch_weights()
        except Exception:
            pass
    return 'done'


def test_ps_drain_liveness_timeout():
    """A crashed worker must produce a diagnostic RuntimeError on the PS
    within PS_DRAIN_TIMEOUT, not a silent hang."""
    res = run_dist(_role_dead_worker, world=3, timeout=120.0)
    assert res[0] == 'timeout', res
