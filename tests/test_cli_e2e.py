"""Reference-workflow end-to-end on CPU/gloo: torchrun the distributed CLI
(1 PS + 1 worker), confirm `model_step_<k>` checkpoints land in train_dir,
then run the polling evaluator over them (prec@1/prec@5 lines) — the full
train->NFS->evaluate loop of the reference."""
import os
import re
import subprocess
import sys

from dist_utils import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cli_train_checkpoint_evaluate(tmp_path):
    ck = str(tmp_path / 'ck')
    os.makedirs(ck)
    env = dict(os.environ)
    env.pop('RANK', None)
    env.pop('WORLD_SIZE', None)
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', str(free_port()), '--no-python', sys.executable,
         '-m', 'ps_pytorch_amd.distributed_nn', '--network', 'LeNet',
         '--dataset', 'MNIST', '--batch-size', '16', '--max-steps', '4',
         '--eval-freq', '2', '--train-dir', ck, '--compress-grad', 'None'],
        capture_output=True, text=True, timeout=420, cwd=REPO, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    files = sorted(os.listdir(ck))
    assert files == ['model_step_2', 'model_step_4'], files

    ev = subprocess.run(
        [sys.executable, '-m', 'ps_pytorch_amd.evaluator', '--network',
         'LeNet', '--dataset', 'MNIST', '--train-dir', ck, '--max-steps',
         '4', '--eval-freq', '2'],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert ev.returncode == 0, ev.stderr[-1500:]
    lines = re.findall(r'Evaluator step (\d+): loss [0-9.]+ prec@1 '
                       r'[0-9.]+ prec@5 [0-9.]+', ev.stdout + ev.stderr)
    assert lines == ['2', '4'], (ev.stdout[-800:], ev.stderr[-800:])
