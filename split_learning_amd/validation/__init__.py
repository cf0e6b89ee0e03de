"""Server-side round validation: rebuild the full model, load the merged state
dict, run the test set, log loss/accuracy (reference src/val/get_val.py:5-16,
src/val/VGG16.py:8-38 — note get_val returns True for every known model
regardless of accuracy; kept for parity, with an opt-in NaN gate)."""

from __future__ import annotations

import math

import torch

from ..data import data_loader
from ..models import build_partition


KNOWN = {"VGG16", "BERT", "KWT", "MobileNetv1", "ViT"}


def get_val(model_name: str, data_name: str, state_dict_full, logger=None,
            device: str = "cpu", max_batches: int = 0) -> bool:
    """max_batches=0 (default) evaluates the WHOLE test set, matching the
    reference's full-test-set loop (src/val/VGG16.py:8-38); a positive value
    truncates (used by fast CPU tests only)."""
    if model_name not in KNOWN:
        if logger:
            logger.log_warning(f"get_val: unknown model {model_name}")
        return False
    model = build_partition(model_name, data_name, [0, 0])
    model.load_state_dict(state_dict_full)
    model.to(device).eval()

    loader = data_loader(data_name, batch_size=64, distribution=None, train=False)
    total = correct = 0
    loss_sum = 0.0
    with torch.no_grad():
        for bi, (x, y) in enumerate(loader):
            if max_batches > 0 and bi >= max_batches:
                break
            x, y = x.to(device), y.to(device)
            logits = model(x)
            loss_sum += torch.nn.functional.cross_entropy(
                logits, y, reduction="sum").item()
            correct += (logits.argmax(-1) == y).sum().item()
            total += y.numel()
    if total == 0:
        return True
    avg_loss = loss_sum / total
    acc = correct / total
    if logger:
        logger.log_info(f"validation: loss={avg_loss:.4f} acc={acc * 100:.2f}%")
    if math.isnan(avg_loss):
        if logger:
            logger.log_warning("validation: NaN loss")
        return False
    return True
