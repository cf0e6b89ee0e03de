"""Minimal LoRA for the BERT path (peft is not installed in this environment).

Mirrors the behaviour the reference gets from peft (src/RpcClient.py:61-66,99-103):
LoraConfig(r=8, lora_alpha=16, lora_dropout=0.1, target_modules=["query","key",
"value","dense"]) wraps matching Linear layers, freezes everything else, and
``merge_and_unload()`` folds B@A*scale back into the base weights before the
state dict is shipped to the server, so .pth files contain plain weights.
"""

from __future__ import annotations

import math
from typing import Iterable

import torch
import torch.nn as nn

from ..ops.modules import HipDropout


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, r: int, alpha: int, dropout: float):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad = False
        self.r = r
        self.scaling = alpha / r
        self.lora_dropout = HipDropout(dropout)
        self.lora_A = nn.Parameter(torch.empty(r, base.in_features))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))

    def forward(self, x):
        y = self.base(x)
        xd = self.lora_dropout(x)
        if x.is_cuda:
            from ..ops import functional as hf
            xa = hf.linear(xd, self.lora_A, None)
            delta = hf.linear(xa, self.lora_B, None)
        else:
            xa = xd @ self.lora_A.t()
            delta = xa @ self.lora_B.t()
        # fused scale+add (one aten kernel instead of mul + add)
        return torch.add(y, delta, alpha=self.scaling)

    def merged_weight(self) -> torch.Tensor:
        return self.base.weight + (self.lora_B @ self.lora_A) * self.scaling


def apply_lora(model: nn.Module, r: int = 8, alpha: int = 16, dropout: float = 0.1,
               target_modules: Iterable[str] = ("query", "key", "value", "dense"),
               trainable_extra: Iterable[str] = ()) -> nn.Module:
    """Wrap matching Linear submodules in-place; freeze all other parameters.

    ``trainable_extra``: dotted prefixes whose params stay trainable (e.g. the
    classifier head, reference src/RpcClient.py:101-103).
    """
    targets = set(target_modules)
    for p in model.parameters():
        p.requires_grad = False

    def visit(mod: nn.Module, prefix: str):
        for name, child in list(mod.named_children()):
            full = f"{prefix}{name}"
            if isinstance(child, nn.Linear) and name in targets:
                setattr(mod, name, LoRALinear(child, r, alpha, dropout))
            else:
                visit(child, full + ".")

    visit(model, "")
    for full_name, p in model.named_parameters():
        if any(full_name.startswith(pref) for pref in trainable_extra):
            p.requires_grad = True
    return model


def merge_and_unload(model: nn.Module) -> nn.Module:
    """Fold LoRA deltas into base weights and restore plain Linear modules."""
    def visit(mod: nn.Module):
        for name, child in list(mod.named_children()):
            if isinstance(child, LoRALinear):
                with torch.no_grad():
                    child.base.weight.copy_(child.merged_weight())
                for p in child.base.parameters():
                    p.requires_grad = True
                setattr(mod, name, child.base)
            else:
                visit(child)

    visit(model)
    for p in model.parameters():
        p.requires_grad = True
    return model
