"""Tuning parser, speedup analysis, and autograd-driver parity components."""
import json
import subprocess
import sys

import torch
import torch.nn.functional as F

from ps_pytorch_amd.tuning_parser import parse_losses, main as tp_main
from ps_pytorch_amd.utils.logging import worker_log_line
from ps_pytorch_amd.utils.autograd import run_backward, loss_backward_via_logits


def test_tuning_parser_roundtrip(tmp_path):
    lines = [worker_log_line(r, s, 0, s * 16, 160, 2.0 / s, 0.1, 0.01, 0.02,
                             0.03, 0.04)
             for r in (1, 2) for s in (10, 20, 30)]
    log = tmp_path / 'lr_0.1.log'
    log.write_text('\n'.join(f'INFO:worker:{l}' for l in lines) + '\n')
    best = parse_losses(log.read_text().splitlines())
    assert set(best) == {1, 2}
    assert best[1] == (30, round(2.0 / 30, 4))
    assert tp_main(['--tuning-dir', str(log), '--tuning-lr', '0.1',
                    '--num-workers', '2']) == 0


def test_speedup_analysis_bench_mode(tmp_path):
    rows = [(1, 98876.0, 10.36), (2, 95000.0, 10.8), (8, 660000.0, 10.9)]
    paths = []
    for n, v, ms in rows:
        p = tmp_path / f'BENCH_{n}.json'
        p.write_text(json.dumps({'n_gpus': n, 'value': v,
                                 'ms_per_step': ms}) + '\n')
        paths.append(str(p))
    out = subprocess.run(
        [sys.executable, 'analysis/speedup.py', 'bench'] + paths,
        capture_output=True, text=True, cwd='.')
    assert out.returncode == 0, out.stderr
    assert '6.675' in out.stdout     # 660000/98876 speedup at N=8


def test_run_backward_matches_plain_backward():
    torch.manual_seed(0)
    w = torch.randn(5, 3, requires_grad=True)
    x = torch.randn(4, 5)
    y = torch.randint(0, 3, (4,))
    out = x @ w
    loss = F.cross_entropy(out, y)
    loss.backward()
    g_ref = w.grad.clone()
    w.grad = None
    out2 = x @ w
    l2 = loss_backward_via_logits(out2, F.cross_entropy, y)
    assert torch.allclose(l2, loss)
    assert torch.allclose(w.grad, g_ref, atol=1e-6)
    w.grad = None
    out3 = x @ w
    run_backward(out3, torch.ones_like(out3))
    assert w.grad.abs().sum() > 0


def test_speedup_bench_record_shapes(tmp_path):
    """from_bench accepts JSON-lines (bench.py), arrays, and wrapper
    objects (driver SCALE file shapes)."""
    import os
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), 'analysis'))
    from speedup import from_bench
    rec = {"n_gpus": 2, "value": 100.0, "ms_per_step": 1.0}
    files = []
    for i, text in enumerate([
            json.dumps(rec),
            json.dumps([rec, {**rec, "n_gpus": 4}]),
            json.dumps(rec) + "\n" + json.dumps({**rec, "n_gpus": 8}),
            json.dumps({"runs": [{**rec, "n_gpus": 1}]})]):
        p = tmp_path / f"b{i}.json"
        p.write_text(text)
        files.append(str(p))
    rows = from_bench(files)
    assert [n for n, _, _ in rows] == [1, 2, 2, 2, 4, 8]


def test_worker_line_matches_reference_format():
    """The worker log line is the de-facto metrics protocol (parsed by the
    tuning parser and analysis); it must stay byte-identical to the
    reference's template (src/distributed_worker.py:169)."""
    from ps_pytorch_amd.utils.logging import WORKER_LINE
    assert WORKER_LINE == (
        'Worker: {}, Step: {}, Epoch: {} [{}/{} ({:.0f}%)], '
        'Loss: {:.4f}, Time Cost: {:.4f}, FetchWeight: {:.4f}, '
        'Forward: {:.4f}, Backward: {:.4f}, Comm Cost: {:.4f}')
