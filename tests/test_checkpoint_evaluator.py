"""Checkpoint layout + evaluator polling (ref: distributed_evaluator.py,
model_step_<k> NFS layout — a declared compat surface)."""
import os

import torch

from ps_pytorch_amd.config import JobConfig
from ps_pytorch_amd.data import prepare_data
from ps_pytorch_amd.evaluator import DistributedEvaluator
from ps_pytorch_amd.models import build_model
from ps_pytorch_amd.utils.checkpoint import (load_model_step, model_step_path,
                                             save_model_step)


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    m = build_model('LeNet', in_channels=1)
    path = save_model_step(m, str(tmp_path), 50)
    assert path == model_step_path(str(tmp_path), 50)
    assert os.path.basename(path) == 'model_step_50'
    m2 = build_model('LeNet', in_channels=1)
    load_model_step(m2, str(tmp_path), 50)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)


def test_evaluator_consumes_checkpoints(tmp_path):
    cfg = JobConfig(network='LeNet', dataset='MNIST', eval_freq=10,
                    max_steps=10, train_dir=str(tmp_path),
                    test_batch_size=64)
    torch.manual_seed(0)
    m = build_model('LeNet', in_channels=1)
    save_model_step(m, str(tmp_path), 10)
    ev = DistributedEvaluator(cfg, poll_interval=0.01, max_polls=3)
    _, test_loader = prepare_data(cfg, device=torch.device('cpu'),
                                  test_size=128)
    ev.evaluate(test_loader)   # consumes step 10 then stops at max_steps


def test_bf16_model_checkpoint_saved_as_f32(tmp_path):
    m = build_model('LeNet', in_channels=1).to(torch.bfloat16)
    save_model_step(m, str(tmp_path), 1)
    sd = torch.load(model_step_path(str(tmp_path), 1), weights_only=True)
    assert all(v.dtype == torch.float32 for v in sd.values()
               if torch.is_floating_point(v))


def test_resume_step_warm_start(tmp_path):
    """--resume-step K: a fresh engine loads model_step_<K> parameters
    (beyond-reference feature; the reference never resumes training)."""
    import torch
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.trainer import NNTrainer
    from ps_pytorch_amd.utils.checkpoint import save_model_step
    cfg = JobConfig(network='LeNet', dataset='MNIST', batch_size=8, lr=0.05,
                    momentum=0.9, compute_dtype='fp32',
                    train_dir=str(tmp_path))
    tr = NNTrainer(cfg, device=torch.device('cpu'))
    tr.build_model()
    x = torch.randn(8, 1, 28, 28)
    y = torch.randint(0, 10, (8,))
    for _ in range(2):
        tr.train_step(x, y)
    save_model_step(tr.network, str(tmp_path), 2)

    cfg2 = JobConfig(network='LeNet', dataset='MNIST', batch_size=8, lr=0.05,
                     momentum=0.9, compute_dtype='fp32',
                     train_dir=str(tmp_path), resume_step=2, seed=999)
    tr2 = NNTrainer(cfg2, device=torch.device('cpu'))
    tr2.build_model()
    # params match the checkpoint despite the different init seed
    for (n, a), (_, b) in zip(tr.network.named_parameters(),
                              tr2.network.named_parameters()):
        assert torch.equal(a, b), n
    assert torch.equal(tr2.master_w[:tr2.flat.total],
                       tr.master_w[:tr.flat.total])


def _role_bn_ckpt(rank: int, world: int, port: int, ckdir: str):
    """BN-net checkpoint division of labor: worker rank 1 (not the PS)
    saves ResNet/VGG so running stats come from a worker (ref
    distributed_worker.py:175-177 / sync_replicas_master_nn.py:194-196)."""
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.data import prepare_data
    from ps_pytorch_amd.parallel.transport import init_distributed
    from ps_pytorch_amd.parallel.ps import ParameterServer
    from ps_pytorch_amd.parallel.worker import DistributedWorker
    cfg = JobConfig(network='ResNet18', dataset='Cifar10', batch_size=8,
                    lr=0.05, momentum=0.9, seed=3, max_steps=2, eval_freq=2,
                    train_dir=ckdir, compress_grad='None', wire_dtype='fp32',
                    compute_dtype='fp32', log_interval=10 ** 9)
    env = init_distributed(backend='gloo')
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, env['device'])
        ps.build_model(10)
        ps.start()
        return None
    w = DistributedWorker(cfg, rank, world, env['device'])
    w.build_model(10)
    train_loader, _ = prepare_data(cfg, rank=rank, num_shards=world - 1,
                                   device=env['device'],
                                   dtype=w.compute_dtype,
                                   train_size=64, test_size=16)
    w.train(train_loader)
    return None


def test_bn_net_checkpoint_saved_by_worker(tmp_path):
    import os
    import torch
    from dist_utils import run_dist
    ck = str(tmp_path / 'ck')
    os.makedirs(ck)
    run_dist(_role_bn_ckpt, world=3, args=(ck,))
    path = os.path.join(ck, 'model_step_2')
    assert os.path.isfile(path), os.listdir(ck)
    sd = torch.load(path, map_location='cpu', weights_only=True)
    # full state_dict incl. BN buffers, with stats actually updated
    assert 'bn1.running_mean' in sd
    assert 'bn1.num_batches_tracked' in sd
    assert int(sd['bn1.num_batches_tracked']) == 2
    assert float(sd['bn1.running_var'].mean()) != 1.0
    # and the evaluator's buffer-aware strict load accepts it
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.evaluator import DistributedEvaluator
    ev = DistributedEvaluator(JobConfig(network='ResNet18',
                                        dataset='Cifar10', train_dir=ck))
    ev._load_model(path)
    assert int(ev.network.bn1.num_batches_tracked) == 2
