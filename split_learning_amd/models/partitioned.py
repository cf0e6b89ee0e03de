"""Base class for numbered-unit partitioned models.

Reference semantics (src/model/VGG16_CIFAR10.py:4-8,119-230): a model is a list of
numbered units ``layer1..layerN``; an instance constructed with
``(start_layer, end_layer)`` materialises and runs exactly the units ``i`` with
``start_layer < i <= end_layer``; ``end_layer == -1`` means "through the last
unit" (src/model/BERT_AGNEWS.py:173).  State-dict keys are ``layer{i}.<param>``
(plus bare parameters such as ``cls_token``), which is what makes the saved
``{model}_{data}.pth`` files interoperable and what the server's partial-key
slicing (src/Server.py:241-254) relies on.
"""

from __future__ import annotations

import torch.nn as nn


class PartitionedModel(nn.Module):
    """Subclasses set TOTAL_UNITS and implement _build(); units are numbered 1..TOTAL."""

    TOTAL_UNITS: int = 0

    def __init__(self, start_layer: int = 0, end_layer: int | None = None):
        super().__init__()
        total = type(self).TOTAL_UNITS
        if end_layer is None:
            end_layer = total
        if end_layer == -1:
            end_layer = total
        self.start_layer = int(start_layer)
        self.end_layer = int(end_layer)
        self._build()

    # -- partition helpers ---------------------------------------------------
    def _active(self, i: int) -> bool:
        return self.start_layer < i <= self.end_layer

    def active_units(self):
        return [i for i in range(1, type(self).TOTAL_UNITS + 1) if self._active(i)]

    def _build(self) -> None:
        raise NotImplementedError


class SequentialUnits(PartitionedModel):
    """Partitioned model whose units are a flat module chain (VGG16, MobileNetv1).

    Subclasses provide UNIT_FACTORIES: a dict {unit_index: factory()} built once
    per class; forward applies active units in order.
    """

    @classmethod
    def unit_factories(cls):
        raise NotImplementedError

    def _build(self):
        factories = type(self).unit_factories()
        assert len(factories) == type(self).TOTAL_UNITS
        for i in range(1, type(self).TOTAL_UNITS + 1):
            if self._active(i):
                setattr(self, f"layer{i}", factories[i]())

    def forward(self, x):
        # stage-executor fusion: an adjacent BatchNorm2d -> ReLU pair inside
        # one partition runs as a single fused kernel on GPU (the cut can
        # still split the pair, in which case both run standalone); a Conv2d
        # feeding a training-mode BatchNorm skips its bias-gradient reduction
        # (analytically zero through batch normalisation — ops/functional.py)
        from ..ops.modules import HipBatchNorm2d, HipConv2d, HipReLU
        units = self.active_units()
        n = len(units)
        i = 0
        while i < n:
            mod = getattr(self, f"layer{units[i]}")
            nxt = getattr(self, f"layer{units[i + 1]}") if i + 1 < n else None
            if (isinstance(mod, HipBatchNorm2d) and x.is_cuda
                    and isinstance(nxt, HipReLU)):
                x = mod(x, fuse_relu=True)
                i += 2
                continue
            if (isinstance(mod, HipConv2d) and x.is_cuda and self.training
                    and isinstance(nxt, HipBatchNorm2d) and nxt.training):
                x = mod(x, bias_grad_zero=True)
                i += 1
                continue
            x = mod(x)
            i += 1
        return x
