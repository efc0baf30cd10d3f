"""Synthetic datasets + device-resident loader.

Reference parity: src/util.py:21-106 `prepare_data` (MNIST / CIFAR-10 /
CIFAR-100 / SVHN via torchvision) and the vendored multiprocessing DataLoader
(src/data_loader_ops/my_data_loader.py). This environment has no network and
no torchvision, so datasets are synthetic tensors of the real shapes
(label-correlated so losses actually fall), generated deterministically per
(dataset, seed, rank).

MI355X-native loader design: 288 GB HBM3E per GPU makes host-side batching
obsolete for these workloads — the WHOLE shard lives on-device and a "batch"
is an index_select over a per-epoch permutation (zero H2D per step, no worker
processes, no pin-memory thread). ResidentLoader still exposes next_batch()
(the vendored loader's API, my_data_loader.py:318) for drop-in use; on CPU
it behaves identically with host tensors.
"""
from __future__ import annotations

import math
from typing import Iterator, Optional, Tuple

import torch

from ..config import input_shape_of, num_classes_of

# Dataset sizes mirror the real datasets (CIFAR: 50k/10k etc.) but are
# capped so CPU tests stay fast; synthetic => size is a free parameter.
_TRAIN_SIZE = {'mnist': 60000, 'cifar10': 50000, 'cifar100': 50000,
               'svhn': 73257, 'imagenet-syn': 8192}   # 224^2 f32: keep resident size sane
_TEST_SIZE = {'mnist': 10000, 'cifar10': 10000, 'cifar100': 10000,
              'svhn': 26032, 'imagenet-syn': 2000}


class SyntheticDataset:
    """Label-correlated gaussian images: x = noise + class_template[y].

    Deterministic for a given (dataset, seed, split); templates are shared
    across ranks so the learning problem is consistent cluster-wide."""

    def __init__(self, dataset: str = 'MNIST', split: str = 'train',
                 size: Optional[int] = None, seed: int = 1,
                 device: torch.device = torch.device('cpu'),
                 dtype: torch.dtype = torch.float32,
                 template_seed: Optional[int] = None):
        d = dataset.lower()
        self.name = d
        self.shape = input_shape_of(d)
        self.num_classes = num_classes_of(d)
        n = size if size is not None else (_TRAIN_SIZE if split == 'train'
                                           else _TEST_SIZE).get(d, 10000)
        g = torch.Generator().manual_seed(seed * 1000003 + (0 if split == 'train' else 1))
        # class templates must be keyed by the JOB seed only: the per-rank
        # sample seed must NOT leak in, or every shard (and the test split)
        # would model a different classification problem
        tg = torch.Generator().manual_seed(
            seed if template_seed is None else template_seed)
        templates = 0.5 * torch.randn((self.num_classes,) + self.shape, generator=tg)
        y = torch.randint(0, self.num_classes, (n,), generator=g)
        x = torch.randn((n,) + self.shape, generator=g) * 0.5 + templates[y]
        self.x = x.to(device=device, dtype=dtype)
        self.y = y.to(device=device)

    def __len__(self) -> int:
        return self.x.shape[0]


class ResidentLoader:
    """Batches by permutation-indexing a device-resident dataset.

    API-compatible surface with the vendored loader: iteration yields
    (data, target); next_batch() returns one batch and wraps epochs."""

    def __init__(self, ds: SyntheticDataset, batch_size: int,
                 shuffle: bool = True, seed: int = 1, drop_last: bool = True):
        self.ds = ds
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self._epoch = 0
        self._pos = 0
        self._seed = seed
        self._perm = self._make_perm()

    @property
    def dataset(self):
        return self.ds

    def _make_perm(self) -> torch.Tensor:
        if not self.shuffle:
            return torch.arange(len(self.ds), device=self.ds.x.device)
        g = torch.Generator().manual_seed(self._seed * 7919 + self._epoch)
        return torch.randperm(len(self.ds), generator=g).to(self.ds.x.device)

    def __len__(self) -> int:
        n = len(self.ds)
        return n // self.batch_size if self.drop_last else math.ceil(n / self.batch_size)

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        for i in range(len(self)):
            idx = self._perm[i * self.batch_size:(i + 1) * self.batch_size]
            yield self.ds.x.index_select(0, idx), self.ds.y.index_select(0, idx)
        self._epoch += 1
        self._perm = self._make_perm()

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        if self._pos >= len(self):
            self._pos = 0
            self._epoch += 1
            self._perm = self._make_perm()
        i = self._pos
        self._pos += 1
        idx = self._perm[i * self.batch_size:(i + 1) * self.batch_size]
        return self.ds.x.index_select(0, idx), self.ds.y.index_select(0, idx)


def prepare_data(args_or_cfg, rank: int = 0, num_shards: int = 1,
                 device: torch.device = torch.device('cpu'),
                 dtype: torch.dtype = torch.float32,
                 train_size: Optional[int] = None,
                 test_size: Optional[int] = None):
    """(train_loader, test_loader) for a rank's shard (ref util.py:21-106).

    Data locality parity: the reference pre-downloads data on every node so
    nothing moves over the wire (README.md:24); here each worker generates a
    deterministic shard keyed by its rank."""
    cfg = args_or_cfg
    dataset = getattr(cfg, 'dataset', 'MNIST')
    seed = getattr(cfg, 'seed', 1)
    bs = getattr(cfg, 'batch_size', 128)
    tbs = getattr(cfg, 'test_batch_size', 500)
    train = SyntheticDataset(dataset, 'train', size=train_size,
                             seed=seed * 131 + rank, device=device,
                             dtype=dtype, template_seed=seed)
    test = SyntheticDataset(dataset, 'test', size=test_size, seed=seed,
                            device=device, dtype=dtype, template_seed=seed)
    return (ResidentLoader(train, bs, shuffle=True, seed=seed + rank),
            ResidentLoader(test, tbs, shuffle=False, drop_last=False))
