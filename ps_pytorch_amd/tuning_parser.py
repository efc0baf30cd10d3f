"""LR-tuning log parser (reference parity: src/tiny_tuning_parser.py).

Greps a trial's log for the LAST worker iteration line of every rank at (or
nearest below) --max-tuning-step and prints the averaged loss for the LR
candidate — the de-facto metric the reference's tune.sh grid consumed. The
line format is the declared compat surface written by
utils/logging.worker_log_line (ref distributed_worker.py:169-173).
"""
from __future__ import annotations

import argparse
import re
import sys

WORKER_RE = re.compile(
    r'Worker: (?P<rank>\d+), Step: (?P<step>\d+), Epoch: .*\[.*\(.*\)\], '
    r'Loss: (?P<loss>[-+0-9.naif]+), Time Cost: .*, FetchWeight: .*, '
    r'Forward: .*, Backward: .*, Comm Cost: .*')


def parse_losses(lines, max_step: int = 10 ** 9):
    """{rank: (last_step<=max_step, loss)} over the log."""
    best = {}
    for line in lines:
        m = WORKER_RE.search(line)
        if not m:
            continue
        rank, step = int(m.group('rank')), int(m.group('step'))
        if step <= max_step and step >= best.get(rank, (-1, 0.0))[0]:
            best[rank] = (step, float(m.group('loss')))
    return best


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description='Distributed tuning log parser')
    p.add_argument('--tuning-dir', type=str, required=True,
                   help='path of one trial log file')
    p.add_argument('--tuning-lr', type=float, default=0.125)
    p.add_argument('--num-workers', type=int, default=0,
                   help='expected worker count (0 = no check)')
    p.add_argument('--max-tuning-step', type=int, default=10 ** 9)
    args = p.parse_args(argv)
    with open(args.tuning_dir, 'r', errors='replace') as f:
        best = parse_losses(f, args.max_tuning_step)
    if not best:
        print('No worker lines found!', file=sys.stderr)
        return 1
    if args.num_workers and len(best) != args.num_workers:
        print('Illegal Number of Workers!', file=sys.stderr)
    avg = sum(l for _, l in best.values()) / len(best)
    print(f'Avged loss for lr candidate: {args.tuning_lr}=========>{avg}')
    return 0


if __name__ == '__main__':
    raise SystemExit(main())
