"""CIFAR-100-shaped data pipeline, torchvision-free.

The reference uses torchvision CIFAR-100 with RandomCrop(32, padding=4) +
Normalize (reference utils/dataset.py:3-14) and plain Normalize for eval
(utils/dataset.py:17-26). There is no network in this environment, so the
default dataset is synthetic with the exact CIFAR shape/statistics; if a real
CIFAR-100 python-pickle dump exists under ``data_root`` it is read directly
(no torchvision dependency).

Design notes (MI355X-first): the augmentation is functional and seeded per
(epoch-independent) index so every access is deterministic — that is what lets
single-GPU vs multi-GPU parity tests compare loss curves exactly, and what
makes hipGraph replay of a captured step meaningful.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional, Tuple

import torch
from torch.utils.data import DataLoader, Dataset
from torch.utils.data.distributed import DistributedSampler

from ..config import CIFAR100_MEAN, CIFAR100_STD, TrainConfig

_MEAN = torch.tensor(CIFAR100_MEAN).view(3, 1, 1)
_STD = torch.tensor(CIFAR100_STD).view(3, 1, 1)


def normalize(img: torch.Tensor) -> torch.Tensor:
    """Channel-wise (x - mean) / std with CIFAR-100 statistics
    (reference utils/dataset.py:8 hard-codes the same numbers)."""
    return (img - _MEAN.to(img.dtype)) / _STD.to(img.dtype)


def random_crop_padded(img: torch.Tensor, size: int, padding: int,
                       gen: Optional[torch.Generator] = None) -> torch.Tensor:
    """RandomCrop(size, padding) equivalent (reference utils/dataset.py:6):
    zero-pad by `padding` on each side, then take a random size×size window."""
    c, h, w = img.shape
    padded = torch.zeros(c, h + 2 * padding, w + 2 * padding, dtype=img.dtype)
    padded[:, padding:padding + h, padding:padding + w] = img
    top = int(torch.randint(0, padded.shape[1] - size + 1, (1,), generator=gen))
    left = int(torch.randint(0, padded.shape[2] - size + 1, (1,), generator=gen))
    return padded[:, top:top + size, left:left + size]


class SyntheticCIFAR(Dataset):
    """CIFAR-100-shaped synthetic dataset: 3×32×32 float images, 100 classes.

    Each index is generated from its own seeded Generator, so ``ds[i]`` is
    deterministic across accesses and processes (needed for DDP parity tests)
    while distinct across indices. Train mode applies the same augmentation
    chain as the reference pipeline (crop-with-padding then normalize); eval
    mode applies normalize only (reference utils/dataset.py:17-26).
    """

    def __init__(self, n: int = 50000, num_classes: int = 100, size: int = 32,
                 train: bool = True, seed: int = 0, augment: Optional[bool] = None):
        self.n = n
        self.num_classes = num_classes
        self.size = size
        self.train = train
        self.seed = seed
        self.augment = train if augment is None else augment

    def __len__(self) -> int:
        return self.n

    def __getitem__(self, idx: int) -> Tuple[torch.Tensor, int]:
        gen = torch.Generator().manual_seed(
            (self.seed * 1_000_003 + idx) * 2 + int(self.train))
        img = torch.rand(3, self.size, self.size, generator=gen)
        label = int(torch.randint(0, self.num_classes, (1,), generator=gen))
        if self.augment:
            img = random_crop_padded(img, self.size, 4, gen=gen)
        return normalize(img), label


class PickleCIFAR100(Dataset):
    """Reads the standard CIFAR-100 python-pickle dump directly (the files
    torchvision would download: cifar-100-python/{train,test}) — no
    torchvision. Same transform chain as SyntheticCIFAR."""

    def __init__(self, root: str, train: bool = True, seed: int = 0):
        name = "train" if train else "test"
        path = os.path.join(root, "cifar-100-python", name)
        with open(path, "rb") as f:
            d = pickle.load(f, encoding="bytes")
        data = torch.frombuffer(bytearray(b"".join(d[b"data"])), dtype=torch.uint8) \
            if isinstance(d[b"data"], list) else torch.from_numpy(d[b"data"].copy())
        self.images = data.reshape(-1, 3, 32, 32).float().div_(255.0)
        self.labels = [int(x) for x in d[b"fine_labels"]]
        self.train = train
        self.seed = seed

    def __len__(self) -> int:
        return self.images.shape[0]

    def __getitem__(self, idx: int) -> Tuple[torch.Tensor, int]:
        img = self.images[idx]
        if self.train:
            gen = torch.Generator().manual_seed(self.seed * 1_000_003 + idx)
            img = random_crop_padded(img, 32, 4, gen=gen)
        return normalize(img), self.labels[idx]


def _real_cifar_available(root: str) -> bool:
    return os.path.exists(os.path.join(root, "cifar-100-python", "train"))


def build_datasets(cfg: TrainConfig) -> Tuple[Dataset, Dataset]:
    if not cfg.synthetic and _real_cifar_available(cfg.data_root):
        return (PickleCIFAR100(cfg.data_root, train=True, seed=cfg.seed),
                PickleCIFAR100(cfg.data_root, train=False))
    return (SyntheticCIFAR(50000, cfg.num_classes, cfg.image_size, train=True,
                           seed=cfg.seed),
            SyntheticCIFAR(10000, cfg.num_classes, cfg.image_size, train=False,
                           seed=cfg.seed))


def build_loaders(cfg: TrainConfig, world_size: int, rank: int,
                  distributed: bool = True,
                  ) -> Tuple[DataLoader, DataLoader, Optional[DistributedSampler]]:
    """Sharded train/test loaders. Global batch is divided by world size
    (reference distributed.py:67); both loaders get a DistributedSampler in
    distributed mode (reference distributed.py:70-75 — note the test-set
    sampler pads to divide evenly, so eval accuracy is approximate there too);
    train uses drop_last so every rank steps the same count."""
    train_ds, test_ds = build_datasets(cfg)
    per_rank = cfg.per_rank_batch(world_size)
    train_sampler: Optional[DistributedSampler] = None
    test_sampler: Optional[DistributedSampler] = None
    if distributed:
        train_sampler = DistributedSampler(train_ds, num_replicas=world_size,
                                           rank=rank, shuffle=True,
                                           seed=cfg.seed, drop_last=True)
        test_sampler = DistributedSampler(test_ds, num_replicas=world_size,
                                          rank=rank, shuffle=False)
    train_loader = DataLoader(
        train_ds, batch_size=per_rank, shuffle=(train_sampler is None),
        sampler=train_sampler, num_workers=cfg.num_workers,
        pin_memory=cfg.pin_memory and torch.cuda.is_available(),
        drop_last=True, persistent_workers=cfg.num_workers > 0)
    test_loader = DataLoader(
        test_ds, batch_size=per_rank, shuffle=False, sampler=test_sampler,
        num_workers=cfg.num_workers,
        pin_memory=cfg.pin_memory and torch.cuda.is_available(),
        persistent_workers=cfg.num_workers > 0)
    return train_loader, test_loader, train_sampler
