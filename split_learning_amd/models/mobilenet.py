"""MobileNetv1-shaped partitioned models (84 units).

Matches other/Vanilla_SL/src/model/MobileNetv1_CIFAR10.py:10-185 — note the
reference uses FULL convolutions (no groups/depthwise): alternating 3x3
(stride 1 or 2) and 1x1 convs, each followed by BatchNorm2d + ReLU; tail is
MaxPool2d(2,2) -> Flatten -> Linear(1024, 10).
"""

from __future__ import annotations

import torch.nn as nn

from ..ops.modules import (HipBatchNorm2d, HipConv2d, HipLinear, HipMaxPool2d,
                           HipReLU)
from .partitioned import SequentialUnits

# (out_channels, kernel, stride) for each of the 27 convs, in order
# (other/Vanilla_SL/src/model/MobileNetv1_CIFAR10.py:11-167)
_CONVS = [
    (32, 3, 1), (32, 3, 1), (64, 1, 1),
    (64, 3, 2), (128, 1, 1), (128, 3, 1), (128, 1, 1),
    (128, 3, 2), (256, 1, 1), (256, 3, 1), (256, 1, 1),
    (256, 3, 2), (512, 1, 1),
    (512, 3, 1), (512, 1, 1), (512, 3, 1), (512, 1, 1), (512, 3, 1), (512, 1, 1),
    (512, 3, 1), (512, 1, 1), (512, 3, 1), (512, 1, 1),
    (512, 3, 2), (1024, 1, 1), (1024, 3, 1), (1024, 1, 1),
]


def _mobilenet_factories(in_channels: int):
    factories = {}
    idx = 0
    c_in = in_channels
    for (c_out, k, s) in _CONVS:
        ci, co = c_in, c_out
        idx += 1
        factories[idx] = (lambda ci=ci, co=co, k=k, s=s:
                          HipConv2d(ci, co, kernel_size=k, stride=s, padding=(1 if k == 3 else 0)))
        idx += 1
        factories[idx] = (lambda co=co: HipBatchNorm2d(co))
        idx += 1
        factories[idx] = (lambda: HipReLU())
        c_in = c_out
    idx += 1
    factories[idx] = (lambda: HipMaxPool2d(2, 2))
    idx += 1
    factories[idx] = (lambda: nn.Flatten(1, -1))
    idx += 1
    factories[idx] = (lambda: HipLinear(1024, 10))
    return factories


class MobileNetv1_CIFAR10(SequentialUnits):
    TOTAL_UNITS = 84
    _FACTORIES = None

    @classmethod
    def unit_factories(cls):
        if cls._FACTORIES is None:
            cls._FACTORIES = _mobilenet_factories(in_channels=3)
        return cls._FACTORIES


class MobileNetv1_MNIST(SequentialUnits):
    TOTAL_UNITS = 84
    _FACTORIES = None

    @classmethod
    def unit_factories(cls):
        if cls._FACTORIES is None:
            cls._FACTORIES = _mobilenet_factories(in_channels=1)
        return cls._FACTORIES
