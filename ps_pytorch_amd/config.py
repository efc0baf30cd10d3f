"""CLI flag surface, compatible with the reference's argparse contract.

Flag names/semantics mirror /root/reference/src/distributed_nn.py:24-68 and
single_machine.py (the declared compatibility surface); new MI355X-specific
knobs are added under their own names and default to sane values.
"""
from __future__ import annotations

import argparse
import dataclasses
from typing import Optional


def str2bool(v: str) -> bool:
    # The reference used `type=bool`, where any non-empty string is truthy
    # (SURVEY.md §5 "Config / flag system"); we accept the same spellings but
    # parse them properly.
    if isinstance(v, bool):
        return v
    return v.lower() not in ("", "0", "false", "no", "none")


def add_fit_args(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    """Training flags (reference parity: distributed_nn.py:24-68)."""
    parser.add_argument('--batch-size', type=int, default=128, metavar='N',
                        help='per-worker input batch size for training')
    parser.add_argument('--test-batch-size', type=int, default=500, metavar='N',
                        help='input batch size for testing')
    parser.add_argument('--epochs', type=int, default=100, metavar='N',
                        help='number of epochs to train')
    parser.add_argument('--max-steps', type=int, default=10000, metavar='N',
                        help='the maximum number of iterations')
    parser.add_argument('--lr', type=float, default=0.01, metavar='LR',
                        help='learning rate')
    parser.add_argument('--momentum', type=float, default=0.5, metavar='M',
                        help='SGD momentum')
    parser.add_argument('--seed', type=int, default=1, metavar='S',
                        help='random seed')
    parser.add_argument('--log-interval', type=int, default=10, metavar='N',
                        help='batches between log lines')
    parser.add_argument('--network', type=str, default='LeNet', metavar='N',
                        help='LeNet | ResNet18/34/50/101/152 | VGG11/13/16/19[_BN]')
    parser.add_argument('--mode', type=str, default='normal', metavar='N',
                        help='normal | kill | timeout : straggler handling '
                             '(kill = PS signal on quota, ref backward_signal_kill; '
                             'timeout = local --kill-threshold abort)')
    parser.add_argument('--kill-threshold', type=float, default=7.0, metavar='KT',
                        help='timeout threshold (s) that triggers straggler kill')
    parser.add_argument('--dataset', type=str, default='MNIST', metavar='N',
                        help='MNIST | Cifar10 | Cifar100 | SVHN | ImageNet-syn')
    parser.add_argument('--comm-type', type=str, default='Bcast', metavar='N',
                        help='Bcast (collective weight distribution) | Async (P2P)')
    parser.add_argument('--num-aggregate', type=int, default=5, metavar='N',
                        help='how many worker gradients count per iteration')
    parser.add_argument('--eval-freq', type=int, default=50, metavar='N',
                        help='checkpoint every this many steps')
    parser.add_argument('--train-dir', type=str, default='output/models/', metavar='N',
                        help='shared directory for model_step_<k> checkpoints')
    parser.add_argument('--compress-grad', type=str, default='compress', metavar='N',
                        help='compress | None : wire compression for gradients')
    parser.add_argument('--data-dir', type=str, default=None, metavar='DIR',
                        help='directory holding real dataset files (MNIST idx / '
                             'CIFAR pickle batches / SVHN .mat); falls back to '
                             './<name>_data, then synthetic data')
    parser.add_argument('--enable-gpu', type=str2bool, nargs='?', const=True,
                        default=False, help='run compute on GPUs (one rank per GPU)')
    parser.add_argument('--no-cuda', action='store_true', default=False,
                        help='disables GPU training (overrides --enable-gpu; '
                             'reference flag parity, distributed_nn.py:42)')
    # --- MI355X-native knobs (new; not in the reference) ---
    parser.add_argument('--wire-dtype', type=str, default='fp32',
                        help='fp32 | bf16 : uncompressed wire dtype (one declared '
                             'dtype end-to-end; SURVEY.md §2.4 dtype-quirk note)')
    parser.add_argument('--compute-dtype', type=str, default='bf16',
                        help='bf16 | fp32 : worker compute dtype on GPU')
    parser.add_argument('--bucket-mb', type=float, default=4.0,
                        help='gradient bucket size (MB) for overlapped RCCL ops')
    parser.add_argument('--overlap', type=str2bool, nargs='?', const=True, default=True,
                        help='overlap gradient push with backward (side HIP stream)')
    parser.add_argument('--aggregation', type=str, default='collective',
                        help='collective (reduce-to-root) | gather (per-worker P2P, '
                             'enables arrival-order --num-aggregate selection)')
    parser.add_argument('--resume-step', type=int, default=0,
                        help='warm-start: load train-dir/model_step_<K> '
                             'into the global model before training '
                             '(parameters only — the reference checkpoints '
                             'no optimizer state; 0 = fresh start)')
    parser.add_argument('--optimizer', type=str, default='sgd',
                        help='PS-side optimizer: sgd | adam (the reference '
                             'ships optim/adam.py but hardwires SGD at '
                             'sync_replicas_master_nn.py:126 — here both '
                             'fused flat optimizers are selectable)')
    parser.add_argument('--engine', type=str, default='ps',
                        help='ps (1 PS + N-1 workers, the reference design) | '
                             'allreduce (collective DP on all N ranks — the '
                             'reference\'s vendored data_parallel_dist.py, built '
                             'properly)')
    return parser


def parse_args(argv=None) -> argparse.Namespace:
    parser = argparse.ArgumentParser(description='ps_pytorch_amd')
    add_fit_args(parser)
    return parser.parse_args(argv)


@dataclasses.dataclass
class JobConfig:
    """Normalized config shared by all roles."""
    network: str = 'LeNet'
    dataset: str = 'MNIST'
    batch_size: int = 128
    test_batch_size: int = 500
    epochs: int = 100
    max_steps: int = 10000
    lr: float = 0.01
    momentum: float = 0.5
    seed: int = 1
    log_interval: int = 10
    mode: str = 'normal'
    kill_threshold: float = 7.0
    comm_type: str = 'Bcast'
    num_aggregate: int = 5
    eval_freq: int = 50
    train_dir: str = 'output/models/'
    compress_grad: str = 'compress'
    data_dir: Optional[str] = None
    enable_gpu: bool = False
    wire_dtype: str = 'fp32'
    compute_dtype: str = 'bf16'
    bucket_mb: float = 4.0
    overlap: bool = True
    aggregation: str = 'collective'
    engine: str = 'ps'
    optimizer: str = 'sgd'
    resume_step: int = 0

    def __post_init__(self) -> None:
        if self.mode not in ('normal', 'kill', 'timeout'):
            raise ValueError(f"unknown --mode {self.mode!r}")
        if self.aggregation not in ('collective', 'gather'):
            raise ValueError(f"unknown --aggregation {self.aggregation!r}")
        if self.optimizer not in ('sgd', 'adam'):
            raise ValueError(f"unknown --optimizer {self.optimizer!r}")
        # --mode kill needs arrival-order fan-in: the PS's kill verdict fires
        # from the gather drain's first-k quota (parallel/ps.py); a collective
        # reduce has no arrival order and would never send a verdict, hanging
        # every worker at its end-of-step verdict recv. Normalize rather than
        # reject: the reference's --mode kill ran on its (gather-like)
        # per-layer P2P protocol without a separate aggregation flag.
        if self.mode == 'kill' and self.aggregation != 'gather':
            import warnings
            warnings.warn("--mode kill requires --aggregation gather "
                          "(arrival-order first-k drives the kill verdict); "
                          "switching aggregation to 'gather'")
            self.aggregation = 'gather'

    @property
    def compress(self) -> bool:
        return str(self.compress_grad).lower() in ('compress', 'true', '1')

    @classmethod
    def from_args(cls, args: argparse.Namespace) -> "JobConfig":
        known = {f.name for f in dataclasses.fields(cls)}
        kw = {k: v for k, v in vars(args).items() if k in known}
        if getattr(args, 'no_cuda', False):
            kw['enable_gpu'] = False
        return cls(**kw)


def num_classes_of(dataset: str) -> int:
    d = dataset.lower()
    if d in ('mnist', 'cifar10', 'svhn'):
        return 10
    if d == 'cifar100':
        return 100
    if d in ('imagenet-syn', 'imagenet'):
        return 1000
    raise ValueError(f"unknown dataset {dataset!r}")


def input_shape_of(dataset: str):
    d = dataset.lower()
    if d == 'mnist':
        return (1, 28, 28)
    if d in ('cifar10', 'cifar100', 'svhn'):
        return (3, 32, 32)
    if d in ('imagenet-syn', 'imagenet'):
        return (3, 224, 224)
    raise ValueError(f"unknown dataset {dataset!r}")
