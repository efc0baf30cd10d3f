"""hipGraph-captured training step: replay numerics match eager stepping."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _mk(seed=17):
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.trainer import NNTrainer
    cfg = JobConfig(network='ResNet18', dataset='Cifar10', batch_size=64,
                    lr=0.05, momentum=0.9, enable_gpu=True, seed=3)
    tr = NNTrainer(cfg, device=torch.device('cuda', 0))
    tr.build_model()
    torch.manual_seed(seed)
    x = torch.randn(64, 3, 32, 32, device='cuda', dtype=tr.compute_dtype)
    y = torch.randint(0, 10, (64,), device='cuda')
    return tr, x, y


def test_graph_step_matches_eager():
    # eager: 3 (graph-warmup equivalent) + 5 steps on the same batch
    tr_e, x, y = _mk()
    for _ in range(8):
        tr_e.train_step(x, y)
    torch.cuda.synchronize()

    # graph: enable_graph runs 3 warmup steps on (x, y), then 5 replays
    tr_g, x2, y2 = _mk()
    assert torch.equal(x.float(), x2.float())
    assert tr_g.enable_graph(x2, y2), "hipGraph capture failed"
    loss = None
    for _ in range(5):
        loss = tr_g.graph_step(x2, y2)
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    d = (tr_e.master_w - tr_g.master_w).abs().max()
    scale = tr_e.master_w.abs().max()
    assert d / scale < 1e-2, (d.item(), scale.item())


def test_graph_replay_updates_weights():
    tr, x, y = _mk()
    assert tr.enable_graph(x, y)
    w0 = tr.flat.flat_w.clone()
    tr.graph_step(x, y)
    torch.cuda.synchronize()
    assert not torch.equal(w0, tr.flat.flat_w)
