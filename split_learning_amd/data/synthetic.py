"""Synthetic datasets shaped exactly like the reference's real ones.

This environment has no network (no torchvision download, no HF datasets), so
the default data source generates correctly-shaped random tensors honouring the
per-label sample-count `distribution` the server assigns (reference
src/Server.py:87-101, src/dataset/dataloader.py:124-134).  Real-data loaders
(data/real.py) take over automatically when local files exist.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

# data_name -> (sample shape, num labels, dtype, vocab/None)
SHAPES = {
    "CIFAR10": ((3, 32, 32), 10, torch.float32, None),
    "MNIST": ((1, 28, 28), 10, torch.float32, None),
    "AGNEWS": ((128,), 4, torch.int64, 28996),
    "EMOTION": ((128,), 6, torch.int64, 30522),
    "SPEECHCOMMANDS": ((40, 98), 10, torch.float32, None),
}


def synthetic_tensors(data_name: str, distribution: Optional[List[int]],
                      total: int = 512, seed: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    if data_name not in SHAPES:
        raise ValueError(f"unknown data_name {data_name!r}")
    shape, n_labels, dtype, vocab = SHAPES[data_name]
    g = torch.Generator().manual_seed(seed)

    if distribution:
        counts = [int(c) for c in distribution[:n_labels]]
    else:
        counts = [total // n_labels] * n_labels
    labels = torch.cat([torch.full((c,), i, dtype=torch.int64)
                        for i, c in enumerate(counts) if c > 0])
    n = labels.numel()
    perm = torch.randperm(n, generator=g)
    labels = labels[perm]

    if dtype == torch.int64:
        x = torch.randint(1, vocab, (n, *shape), generator=g, dtype=torch.int64)
    else:
        x = torch.randn(n, *shape, generator=g, dtype=torch.float32)
    return x, labels
