"""ViT partitioned models — 12 units.

Matches other/Vanilla_SL/src/model/ViT_CIFAR10.py:29-116 (and the MNIST variant:
28x28, 1 channel): layer1 patch-embed Conv(kernel=stride=4), layer2 Flatten(2)
then transpose; unit3 CLS token (bare param); unit4 pos_embed (bare param) +
``layer4 = nn.Identity()``; layers 5..10 transformer blocks (embed 128, 4 heads,
mlp 256); layer11 LayerNorm on CLS; layer12 Linear(128, 10).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.modules import HipConv2d, HipLayerNorm, HipLinear
from .kwt import TransformerEncoderBlock
from .partitioned import PartitionedModel


class _ViTBase(PartitionedModel):
    TOTAL_UNITS = 12
    IMG_SIZE = 32
    IN_CHANNELS = 3
    PATCH = 4
    EMBED_DIM = 128
    NUM_HEADS = 4
    MLP_DIM = 256
    NUM_CLASSES = 10

    def _build(self):
        E = self.EMBED_DIM
        n_patches = (self.IMG_SIZE // self.PATCH) ** 2
        if self._active(1):
            self.layer1 = HipConv2d(self.IN_CHANNELS, E, kernel_size=self.PATCH,
                                    stride=self.PATCH)
        if self._active(2):
            self.layer2 = nn.Flatten(2)
        if self._active(3):
            self.cls_token = nn.Parameter(torch.randn(1, 1, E))
        if self._active(4):
            self.pos_embed = nn.Parameter(torch.randn(1, n_patches + 1, E))
            self.layer4 = nn.Identity()
        for i in range(5, 11):
            if self._active(i):
                setattr(self, f"layer{i}",
                        TransformerEncoderBlock(E, self.NUM_HEADS, self.MLP_DIM))
        if self._active(11):
            self.layer11 = HipLayerNorm(E)
        if self._active(12):
            self.layer12 = HipLinear(E, self.NUM_CLASSES)

    def forward(self, x):
        if self._active(1):
            x = self.layer1(x)
        if self._active(2):
            x = self.layer2(x).transpose(1, 2)
        if self._active(3):
            cls = self.cls_token.expand(x.size(0), -1, -1)
            x = torch.cat([cls, x], dim=1)
        if self._active(4):
            x = self.layer4(x + self.pos_embed)
        for i in range(5, 11):
            if self._active(i):
                x = getattr(self, f"layer{i}")(x)
        if self._active(11):
            x = self.layer11(x[:, 0])
        if self._active(12):
            x = self.layer12(x)
        return x


class ViT_CIFAR10(_ViTBase):
    IMG_SIZE = 32
    IN_CHANNELS = 3


class ViT_MNIST(_ViTBase):
    IMG_SIZE = 28
    IN_CHANNELS = 1
