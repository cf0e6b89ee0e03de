"""VGG16 partitioned models (CIFAR10: 52 units; MNIST: 51 units).

Unit numbering and shapes match the reference exactly
(src/model/VGG16_CIFAR10.py:9-117, other/Vanilla_SL/src/model/VGG16_MNIST.py):
each conv block is Conv3x3(s1,p1) -> BatchNorm2d -> ReLU, with MaxPool2d(2,2)
after blocks; classifier is Flatten/Dropout/Linear(512,4096)/ReLU/Dropout/
Linear(4096,4096)/ReLU/Linear(4096,10).  CIFAR10 has 5 pools (32->1 spatial),
MNIST has 4 (28->1).  All layers are the HIP-backed modules from ops.modules.
"""

from __future__ import annotations

from ..ops.modules import (HipBatchNorm2d, HipConv2d, HipDropout, HipLinear,
                           HipMaxPool2d, HipReLU)
import torch.nn as nn

from .partitioned import SequentialUnits

# conv plans: list of blocks; each block = list of out-channels; pool flag per block
_CONV_BLOCKS = [[64, 64], [128, 128], [256, 256, 256], [512, 512, 512], [512, 512, 512]]


def _vgg_factories(in_channels: int, pooled_blocks: int):
    """Build {unit_index: factory} for the VGG16 unit chain."""
    factories = {}
    idx = 0
    c_in = in_channels
    for bi, block in enumerate(_CONV_BLOCKS):
        for c_out in block:
            ci, co = c_in, c_out
            idx += 1
            factories[idx] = (lambda ci=ci, co=co: HipConv2d(ci, co, kernel_size=3, stride=1, padding=1))
            idx += 1
            factories[idx] = (lambda co=co: HipBatchNorm2d(co))
            idx += 1
            factories[idx] = (lambda: HipReLU())
            c_in = c_out
        if bi < pooled_blocks:
            idx += 1
            factories[idx] = (lambda: HipMaxPool2d(kernel_size=2, stride=2))
    idx += 1
    factories[idx] = (lambda: nn.Flatten(1, -1))
    idx += 1
    factories[idx] = (lambda: HipDropout(0.5))
    idx += 1
    factories[idx] = (lambda: HipLinear(512, 4096))
    idx += 1
    factories[idx] = (lambda: HipReLU())
    idx += 1
    factories[idx] = (lambda: HipDropout(0.5))
    idx += 1
    factories[idx] = (lambda: HipLinear(4096, 4096))
    idx += 1
    factories[idx] = (lambda: HipReLU())
    idx += 1
    factories[idx] = (lambda: HipLinear(4096, 10))
    return factories


class VGG16_CIFAR10(SequentialUnits):
    TOTAL_UNITS = 52
    _FACTORIES = None

    @classmethod
    def unit_factories(cls):
        if cls._FACTORIES is None:
            cls._FACTORIES = _vgg_factories(in_channels=3, pooled_blocks=5)
        return cls._FACTORIES


class VGG16_MNIST(SequentialUnits):
    TOTAL_UNITS = 51
    _FACTORIES = None

    @classmethod
    def unit_factories(cls):
        if cls._FACTORIES is None:
            cls._FACTORIES = _vgg_factories(in_channels=1, pooled_blocks=4)
        return cls._FACTORIES
