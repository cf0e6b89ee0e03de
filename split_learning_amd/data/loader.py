"""data_loader(data_name, batch_size, distribution, train) — same call surface
as reference src/dataset/dataloader.py:124-134.

Resolution order: real local dataset files (data/real.py) if present, else
synthetic tensors of the same shape (data/synthetic.py).  `distribution` is the
per-label sample-count list the server assigns to layer-1 clients.
"""

from __future__ import annotations

from typing import List, Optional

import torch
from torch.utils.data import DataLoader, TensorDataset

from .synthetic import synthetic_tensors


class _AugmentedCifar(TensorDataset):
    """CIFAR10 train-time augmentation (reference RandomCrop(32, padding=4) +
    RandomHorizontalFlip) applied per item on the normalized tensors."""

    def __init__(self, x, y, seed: int):
        super().__init__(x, y)
        self._gen = torch.Generator().manual_seed(seed * 65537 + 11)

    def __getitem__(self, idx):
        x, y = super().__getitem__(idx)
        dy, dx = (int(v) for v in torch.randint(0, 9, (2,), generator=self._gen))
        xp = torch.nn.functional.pad(x, (4, 4, 4, 4))[:, dy:dy + 32, dx:dx + 32]
        if bool(torch.rand((), generator=self._gen) < 0.5):
            xp = torch.flip(xp, dims=(2,))
        return xp, y


def data_loader(data_name: str, batch_size: int,
                distribution: Optional[List[int]] = None, train: bool = True,
                drop_last: bool = True, seed: int = 0) -> DataLoader:
    try:
        from .real import load_real
        real = load_real(data_name, distribution, train, seed=seed)
    except Exception:
        real = None
    if real is not None:
        x, y = real
    else:
        x, y = synthetic_tensors(data_name, distribution, seed=seed)
    if real is not None and train and data_name == "CIFAR10":
        ds = _AugmentedCifar(x, y, seed)
    else:
        ds = TensorDataset(x, y)
    gen = torch.Generator().manual_seed(seed * 7919 + 13) if train else None
    return DataLoader(ds, batch_size=batch_size, shuffle=train, drop_last=drop_last,
                      num_workers=0, generator=gen)
