"""Scaling/speedup analysis (reference parity: analysis/*.ipynb).

The reference's notebooks regex-parse worker stdout logs and plot
distributed-over-single speedup ("normal" = straggler-bound max over
workers, "ideal" = min). Here the same curves come from bench.py's JSON
lines (one per N) or from worker logs:

  python analysis/speedup.py bench BENCH_1.json BENCH_2.json BENCH_8.json
  python analysis/speedup.py logs  run_n1.log run_n2.log ...

Prints a table: N GPUs, images/s, ms/step, speedup vs the first entry,
scaling efficiency, plus the reference's published speedup curve for
ResNet-18/CIFAR-10 b=1024 for side-by-side comparison.
"""
from __future__ import annotations

import json
import re
import sys

# reference published normal speedups (m4.2xlarge CPU cluster),
# ResNet-18/CIFAR-10 b=1024 @ {1,2,4,8,16,32} workers — BASELINE.md /
# analysis/Speedups_with_GradCompression.ipynb cell 3
REF_B1K = {1: 1.0, 2: 1.9737, 4: 3.0871, 8: 5.1896, 16: 4.2394, 32: 2.4774}

WORKER_RE = re.compile(r'Worker: \d+, Step: (\d+), .*Time Cost: ([0-9.]+),')


def _bench_records(text):
    """Yield bench-JSON dicts from a file's text: JSON-lines (one object
    per line, bench.py's output), a whole-file array, or an object
    wrapping a list (driver SCALE files)."""
    text = text.strip()
    if text.startswith('['):
        yield from json.loads(text)
        return
    if text.startswith('{') and '\n' not in text:
        d = json.loads(text)
        if 'value' in d:
            yield d
        else:   # wrapper object: find the first list of bench records
            for v in d.values():
                if isinstance(v, list) and v and isinstance(v[0], dict):
                    yield from v
                    return
        return
    for line in text.splitlines():
        line = line.strip()
        if line.startswith('{'):
            yield json.loads(line)


def from_bench(paths):
    rows = []
    for p in paths:
        with open(p) as f:
            for d in _bench_records(f.read()):
                if 'value' in d:
                    rows.append((d['n_gpus'], d['value'], d['ms_per_step']))
    return sorted(rows)


def from_logs(paths):
    rows = []
    for p in paths:
        times = {}
        n = 0
        with open(p, errors='replace') as f:
            for line in f:
                m = WORKER_RE.search(line)
                if m:
                    times.setdefault(int(m.group(1)), []).append(
                        float(m.group(2)))
        if not times:
            continue
        per_step = [max(v) for v in times.values()]   # "normal" = straggler max
        n = max(len(v) for v in times.values()) + 1   # workers + PS
        ms = 1000.0 * sum(per_step) / len(per_step)
        rows.append((n, float('nan'), ms))
    return sorted(rows)


def main(argv):
    if len(argv) < 3 or argv[1] not in ('bench', 'logs'):
        print(__doc__)
        return 1
    rows = from_bench(argv[2:]) if argv[1] == 'bench' else from_logs(argv[2:])
    if not rows:
        print('no data found')
        return 1
    base_n, base_v, base_ms = rows[0]
    print(f"{'N':>3} {'images/s':>12} {'ms/step':>9} {'speedup':>8} "
          f"{'efficiency':>10} {'ref speedup':>12}")
    for n, v, ms in rows:
        sp = (v / base_v) if v == v and base_v == base_v else base_ms / ms
        eff = sp / (n / base_n)
        ref = REF_B1K.get(max(n - 1, 1), float('nan'))
        print(f"{n:>3} {v:>12.0f} {ms:>9.2f} {sp:>8.3f} {eff:>10.1%} "
              f"{ref:>12.4f}")
    return 0


if __name__ == '__main__':
    raise SystemExit(main(sys.argv))
