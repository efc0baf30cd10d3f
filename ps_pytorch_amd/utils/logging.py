"""Structured log lines — the reference's de-facto metrics protocol.

The worker per-iteration line (ref: distributed_worker.py:169-173) is parsed
by tiny_tuning_parser.py and the analysis notebooks; keep the exact format.
"""
from __future__ import annotations

import logging
import sys

WORKER_LINE = ('Worker: {}, Step: {}, Epoch: {} [{}/{} ({:.0f}%)], '
               'Loss: {:.4f}, Time Cost: {:.4f}, FetchWeight: {:.4f}, '
               'Forward: {:.4f}, Backward: {:.4f}, Comm Cost: {:.4f}')

MASTER_LINE = 'Master Step: {}, Method Time Cost: {:.4f}'


def get_logger(name: str = 'ps_pytorch_amd') -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stdout)
        h.setFormatter(logging.Formatter('%(levelname)s:%(name)s:%(message)s'))
        logger.addHandler(h)
        logger.setLevel(logging.INFO)
    return logger


def worker_log_line(rank, step, epoch, seen, total, loss, time_cost,
                    fetch_weight, forward, backward, comm) -> str:
    pct = 100.0 * seen / max(total, 1)
    return WORKER_LINE.format(rank, step, epoch, seen, total, pct, loss,
                              time_cost, fetch_weight, forward, backward, comm)
