"""Data layer: real datasets when their files are on disk, synthetic
fallback otherwise (offline parity with ref src/util.py prepare_data)."""
from __future__ import annotations

from typing import Optional

import torch

from .synthetic import SyntheticDataset, ResidentLoader
from .synthetic import prepare_data as prepare_synthetic
from .real import (RealDataset, RealResidentLoader, dataset_root)
from ..utils.logging import get_logger

logger = get_logger('ps_pytorch_amd.data')


def prepare_data(args_or_cfg, rank: int = 0, num_shards: int = 1,
                 device: torch.device = torch.device('cpu'),
                 dtype: torch.dtype = torch.float32,
                 train_size: Optional[int] = None,
                 test_size: Optional[int] = None):
    """(train_loader, test_loader) — real data if present, else synthetic.

    Mirrors ref util.py:21-106: MNIST/CIFAR-10/CIFAR-100/SVHN with the
    reference's transform stacks (GPU-side, data/real.py). Files are looked
    up under --data-dir / $PS_DATA_ROOT / ./<name>_data (the paths the
    reference's pre-download script used). ImageNet-syn and missing files
    use the synthetic generator (data/synthetic.py)."""
    cfg = args_or_cfg
    dataset = getattr(cfg, 'dataset', 'MNIST')
    root = dataset_root(dataset, getattr(cfg, 'data_dir', None))
    if root is None:
        if dataset.lower() in ('mnist', 'cifar10', 'cifar100', 'svhn'):
            logger.info('%s files not found on disk — using synthetic data '
                        '(drop the files under ./%s_data or --data-dir to '
                        'train on the real set)', dataset, dataset.lower())
        return prepare_synthetic(cfg, rank=rank, num_shards=num_shards,
                                 device=device, dtype=dtype,
                                 train_size=train_size, test_size=test_size)
    seed = getattr(cfg, 'seed', 1)
    bs = getattr(cfg, 'batch_size', 128)
    tbs = getattr(cfg, 'test_batch_size', 500)
    logger.info('%s: real data from %s', dataset, root)
    train = RealDataset(dataset, 'train', root, device=device, dtype=dtype)
    test = RealDataset(dataset, 'test', root, device=device, dtype=dtype)
    return (RealResidentLoader(train, bs, shuffle=True, seed=seed + rank),
            RealResidentLoader(test, tbs, shuffle=False, drop_last=False,
                               augment=False))


__all__ = ['SyntheticDataset', 'prepare_data', 'prepare_synthetic',
           'ResidentLoader', 'RealDataset', 'RealResidentLoader',
           'dataset_root']
