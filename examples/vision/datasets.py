"""Datasets for the vision examples.

This image has no network access, so the default is a synthetic dataset
of the right shape; real CIFAR-10/ImageNet directories are supported when
present on disk (parity surface with reference examples/vision/datasets.py
minus torchvision, which is not installed).
"""

from __future__ import annotations

import os

import torch
from torch.utils.data import DataLoader
from torch.utils.data import Dataset
from torch.utils.data.distributed import DistributedSampler


class SyntheticImages(Dataset):
    """Random images + labels of a fixed shape (reproducible per index)."""

    def __init__(
        self,
        n: int,
        shape: tuple[int, int, int],
        num_classes: int,
        seed: int = 0,
    ):
        self.n = n
        self.shape = shape
        self.num_classes = num_classes
        self.seed = seed

    def __len__(self) -> int:
        return self.n

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed + idx)
        x = torch.randn(*self.shape, generator=g)
        y = int(
            torch.randint(0, self.num_classes, (1,), generator=g).item(),
        )
        return x, y


def _loaders(
    train: Dataset,
    val: Dataset,
    batch_size: int,
    val_batch_size: int,
    workers: int = 2,
) -> tuple[DataLoader, DataLoader]:
    import torch.distributed as dist

    train_sampler = None
    val_sampler = None
    if dist.is_available() and dist.is_initialized():
        train_sampler = DistributedSampler(train)
        val_sampler = DistributedSampler(val, shuffle=False)
    train_loader = DataLoader(
        train,
        batch_size=batch_size,
        sampler=train_sampler,
        shuffle=train_sampler is None,
        num_workers=workers,
        pin_memory=torch.cuda.is_available(),
    )
    val_loader = DataLoader(
        val,
        batch_size=val_batch_size,
        sampler=val_sampler,
        shuffle=False,
        num_workers=workers,
        pin_memory=torch.cuda.is_available(),
    )
    return train_loader, val_loader


def get_cifar(
    data_dir: str | None,
    batch_size: int,
    val_batch_size: int,
    synthetic: bool = True,
    train_size: int = 50000,
    val_size: int = 10000,
) -> tuple[DataLoader, DataLoader]:
    """CIFAR-10 loaders; synthetic unless a real data dir exists."""
    if not synthetic and data_dir is not None and os.path.isdir(data_dir):
        try:
            from torchvision import datasets, transforms  # type: ignore

            tf = transforms.Compose(
                [
                    transforms.RandomCrop(32, padding=4),
                    transforms.RandomHorizontalFlip(),
                    transforms.ToTensor(),
                    transforms.Normalize(
                        (0.4914, 0.4822, 0.4465), (0.247, 0.243, 0.262),
                    ),
                ],
            )
            train = datasets.CIFAR10(data_dir, train=True, transform=tf)
            val = datasets.CIFAR10(
                data_dir, train=False, transform=transforms.ToTensor(),
            )
            return _loaders(train, val, batch_size, val_batch_size)
        except ImportError:
            pass
    train = SyntheticImages(train_size, (3, 32, 32), 10, seed=1)
    val = SyntheticImages(val_size, (3, 32, 32), 10, seed=2)
    return _loaders(train, val, batch_size, val_batch_size)


def get_imagenet(
    data_dir: str | None,
    batch_size: int,
    val_batch_size: int,
    synthetic: bool = True,
    train_size: int = 100000,
    val_size: int = 10000,
) -> tuple[DataLoader, DataLoader]:
    """ImageNet-shaped loaders (synthetic by default)."""
    train = SyntheticImages(train_size, (3, 224, 224), 1000, seed=1)
    val = SyntheticImages(val_size, (3, 224, 224), 1000, seed=2)
    return _loaders(train, val, batch_size, val_batch_size)
