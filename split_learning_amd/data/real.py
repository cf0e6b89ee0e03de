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

# reference normalization constants (src/dataset/dataloader.py CIFAR10
# transform) so checkpoints stay input-distribution-compatible
_CIFAR_MEAN = np.array([0.4914, 0.4822, 0.4465], dtype=np.float32)
_CIFAR_STD = np.array([0.2023, 0.1994, 0.2010], dtype=np.float32)


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


def _load_mnist(train: bool):
    """Raw idx files (train-images-idx3-ubyte etc.) under DATA_ROOT/MNIST/raw."""
    base = os.path.join(DATA_ROOT, "MNIST", "raw")
    stem = "train" if train else "t10k"
    img_p = os.path.join(base, f"{stem}-images-idx3-ubyte")
    lbl_p = os.path.join(base, f"{stem}-labels-idx1-ubyte")
    if not (os.path.exists(img_p) and os.path.exists(lbl_p)):
        return None
    with open(img_p, "rb") as f:
        data = np.frombuffer(f.read(), dtype=np.uint8, offset=16)
    with open(lbl_p, "rb") as f:
        labels = np.frombuffer(f.read(), dtype=np.uint8, offset=8)
    x = data.reshape(-1, 1, 28, 28).astype(np.float32) / 255.0
    x = (x - 0.1307) / 0.3081
    return torch.from_numpy(x.copy()), torch.tensor(labels, dtype=torch.int64)


def _load_agnews(train: bool):
    """AG_NEWS csv (class,title,description) tokenised with BertTokenizer to
    max_len 128 — reference src/dataset/AGNEWS.py / dataloader.py:16-59."""
    p = os.path.join(DATA_ROOT, "ag_news", "train.csv" if train else "test.csv")
    if not os.path.exists(p):
        return None
    import csv as _csv
    from transformers import BertTokenizer
    tok = BertTokenizer.from_pretrained("bert-base-cased")
    ids, labels = [], []
    with open(p, newline="") as f:
        for row in _csv.reader(f):
            label = int(row[0]) - 1
            text = " ".join(row[1:])
            enc = tok(text, max_length=128, truncation=True, padding="max_length")
            ids.append(enc["input_ids"])
            labels.append(label)
    return (torch.tensor(ids, dtype=torch.int64),
            torch.tensor(labels, dtype=torch.int64))


_SC_CLASSES = ["yes", "no", "up", "down", "left", "right", "on", "off", "stop", "go"]


def _load_speechcommands(train: bool, max_per_class: int = 400):
    """SpeechCommands v0.02 wavs -> numpy MFCC [40, 98]
    (reference src/dataset/SPEECHCOMMANDS.py; frontend in data/mfcc.py)."""
    base = os.path.join(DATA_ROOT, "SpeechCommands", "speech_commands_v0.02")
    if not os.path.isdir(base):
        return None
    from scipy.io import wavfile
    from .mfcc import compute_mfcc
    test_set = set()
    tl = os.path.join(base, "testing_list.txt")
    if os.path.exists(tl):
        test_set = set(open(tl).read().split())
    xs, ys = [], []
    for ci, cls in enumerate(_SC_CLASSES):
        d = os.path.join(base, cls)
        if not os.path.isdir(d):
            continue
        count = 0
        for fn in sorted(os.listdir(d)):
            if not fn.endswith(".wav") or count >= max_per_class:
                continue
            rel = f"{cls}/{fn}"
            if train == (rel in test_set):
                continue
            _sr, wav = wavfile.read(os.path.join(d, fn))
            wav = wav.astype(np.float32) / 32768.0
            if len(wav) < 16000:
                wav = np.pad(wav, (0, 16000 - len(wav)))
            xs.append(compute_mfcc(wav[:16000]).astype(np.float32))
            ys.append(ci)
            count += 1
    if not xs:
        return None
    return torch.from_numpy(np.stack(xs)), torch.tensor(ys, dtype=torch.int64)


def load_real(data_name: str, distribution: Optional[List[int]],
              train: bool, seed: int = 0) -> Optional[Tuple[torch.Tensor, torch.Tensor]]:
    """seed: per-client draw so same-distribution clients select DIFFERENT
    sample subsets (the reference uses unseeded random.sample per client —
    ADVICE.md round-1 medium finding)."""
    loaders = {"CIFAR10": _load_cifar10, "MNIST": _load_mnist,
               "AGNEWS": _load_agnews, "SPEECHCOMMANDS": _load_speechcommands}
    fn = loaders.get(data_name)
    if fn is None:
        return None
    loaded = fn(train)
    if loaded is None:
        return None
    return _subset_by_distribution(*loaded, distribution, seed=seed)
