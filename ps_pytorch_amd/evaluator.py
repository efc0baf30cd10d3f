"""Standalone evaluator (reference parity: src/distributed_evaluator.py).

A separate, non-distributed process that polls `train_dir` for
`model_step_<k>` checkpoints (shared filesystem; ref :79-88 polls every 10 s),
loads each state_dict and reports prec@1/prec@5 on the test set.
"""
from __future__ import annotations

import argparse
import os
import time

import torch
import torch.nn.functional as F

from .config import JobConfig, add_fit_args, input_shape_of, num_classes_of
from .data import prepare_data
from .models import build_model
from .utils.checkpoint import model_step_path
from .utils.logging import get_logger
from .utils.metrics import accuracy

logger = get_logger('ps_pytorch_amd.evaluator')


class DistributedEvaluator:
    def __init__(self, cfg: JobConfig, poll_interval: float = 10.0,
                 max_polls: int = 0):
        self.cfg = cfg
        self.poll_interval = poll_interval
        self.max_polls = max_polls     # 0 = poll forever
        self.device = torch.device('cuda' if (cfg.enable_gpu and
                                              torch.cuda.is_available()) else 'cpu')
        nc = num_classes_of(cfg.dataset)
        in_ch = input_shape_of(cfg.dataset)[0]
        self.network = build_model(cfg.network, num_classes=nc,
                                   in_channels=in_ch).to(self.device)

    def evaluate(self, test_loader) -> None:
        cfg = self.cfg
        next_step = cfg.eval_freq
        polls = 0
        while cfg.max_steps == 0 or next_step <= cfg.max_steps:
            path = model_step_path(cfg.train_dir, next_step)
            if os.path.isfile(path):
                self._load_model(path)
                loss, p1, p5 = self._evaluate_model(test_loader)
                logger.info('Evaluator step %d: loss %.4f prec@1 %.2f prec@5 %.2f',
                            next_step, loss, p1, p5)
                next_step += cfg.eval_freq
                polls = 0
            else:
                polls += 1
                if self.max_polls and polls >= self.max_polls:
                    return
                time.sleep(self.poll_interval)

    def _load_model(self, path: str) -> None:
        """Load a model_step_<k> checkpoint. PS checkpoints carry parameters
        only (no BN buffers — the reference never syncs them and its master
        saves params, sync_replicas_master_nn.py:264-270); worker-saved
        ResNet/VGG checkpoints carry the full state_dict. Strictness adapts:
        missing BUFFER keys are tolerated iff the checkpoint has no buffers,
        while a missing parameter or an unexpected key is always an error
        (a silent strict=False here masked renamed keys — ADVICE r1)."""
        sd = torch.load(path, map_location='cpu', weights_only=True)
        param_keys = {k for k, _ in self.network.named_parameters()}
        buffer_keys = {k for k, _ in self.network.named_buffers()}
        has_buffers = any(k in buffer_keys for k in sd)
        missing, unexpected = self.network.load_state_dict(sd, strict=False)
        bad_missing = [k for k in missing
                       if k in param_keys or has_buffers]
        if bad_missing or unexpected:
            raise RuntimeError(
                f"checkpoint {path} does not match model "
                f"{self.cfg.network}: missing={bad_missing} "
                f"unexpected={list(unexpected)}")
        self.network.to(self.device)

    @torch.no_grad()
    def _evaluate_model(self, test_loader):
        self.network.eval()
        pdtype = next(self.network.parameters()).dtype
        tot, loss_sum, p1_sum, p5_sum = 0, 0.0, 0.0, 0.0
        for data, target in test_loader:
            data = data.to(self.device, pdtype)   # loader dtype may differ
            target = target.to(self.device)
            out = self.network(data)
            loss_sum += float(F.cross_entropy(out, target, reduction='sum'))
            k = min(5, out.shape[1])
            p1, pk = accuracy(out, target, topk=(1, k))
            bs = target.size(0)
            p1_sum += float(p1) * bs
            p5_sum += float(pk) * bs
            tot += bs
        return (loss_sum / max(tot, 1), p1_sum / max(tot, 1),
                p5_sum / max(tot, 1))


def main(argv=None) -> None:
    parser = argparse.ArgumentParser(description='ps_pytorch_amd evaluator')
    add_fit_args(parser)
    # reference evaluator flag aliases (distributed_evaluator.py:45-50)
    parser.add_argument('--eval-batch-size', type=int, default=0,
                        help='validation batch size (alias of '
                             '--test-batch-size; reference flag parity)')
    parser.add_argument('--model-dir', type=str, default='',
                        help='checkpoint directory (alias of --train-dir; '
                             'reference flag parity)')
    args = parser.parse_args(argv)
    if args.eval_batch_size:
        args.test_batch_size = args.eval_batch_size
    if args.model_dir:
        args.train_dir = args.model_dir
    cfg = JobConfig.from_args(args)
    ev = DistributedEvaluator(cfg)
    _, test_loader = prepare_data(cfg, device=ev.device)
    ev.evaluate(test_loader)


if __name__ == '__main__':
    main()
