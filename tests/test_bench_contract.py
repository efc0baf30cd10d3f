"""bench.py driver contract: single-process run emits ONE JSON line with
the required fields (the round driver parses this exactly)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'bench.py'),
         '--steps', '2', '--warmup', '1', '--batch-size', '4',
         '--network', 'LeNet', '--dataset', 'MNIST'],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith('{')]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    for key in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
                'ms_per_step', 'higher_is_better', 'scaling',
                'vs_baseline', 'dtype', 'data', 'config'):
        assert key in j, key
    assert j['n_gpus'] == 1
    assert j['steps'] == 2 and j['warmup'] == 1
    assert j['higher_is_better'] is True
    assert j['scaling'] == 'weak'
    assert j['value'] > 0
    cfg = j['config']
    for key in ('model', 'global_batch', 'per_worker_batch', 'parallelism'):
        assert key in cfg, key
