"""KWT (Keyword Transformer) partitioned model — 17 units.

Matches src/model/KWT_SPEECHCOMMANDS.py:26-109: layer1 Linear embed 40->64;
unit2 CLS token (bare parameter ``cls_token``); unit3 positional embedding
(bare parameter ``pos_embed``) + dropout (module named ``dropout``); layers
4..15 pre-LN transformer blocks (nn.MultiheadAttention layout, 1 head,
mlp 64->256->64); layer16 LayerNorm on the CLS token; layer17 Linear 64->10.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.modules import (HipDropout, HipGELU, HipLayerNorm, HipLinear,
                           HipMultiheadAttention)
from .partitioned import PartitionedModel


class TransformerEncoderBlock(nn.Module):
    def __init__(self, embed_dim, num_heads=1, mlp_dim=256):
        super().__init__()
        self.ln1 = HipLayerNorm(embed_dim)
        self.mha = HipMultiheadAttention(embed_dim, num_heads, batch_first=True)
        self.ln2 = HipLayerNorm(embed_dim)
        self.mlp = nn.Sequential(
            HipLinear(embed_dim, mlp_dim),
            HipGELU(),
            HipLinear(mlp_dim, embed_dim),
        )

    def forward(self, x):
        _x = self.ln1(x)
        x = x + self.mha(_x, _x, _x)[0]
        x = x + self.mlp(self.ln2(x))
        return x


class KWT_SPEECHCOMMANDS(PartitionedModel):
    TOTAL_UNITS = 17

    N_MFCC = 40
    TIME_STEPS = 98
    EMBED_DIM = 64
    NUM_HEADS = 1
    MLP_DIM = 256
    NUM_CLASSES = 10
    DROPOUT = 0.1

    def _build(self):
        E = self.EMBED_DIM
        if self._active(1):
            self.layer1 = HipLinear(self.N_MFCC, E)
        if self._active(2):
            self.cls_token = nn.Parameter(torch.randn(1, 1, E))
            nn.init.trunc_normal_(self.cls_token, std=0.02)
        if self._active(3):
            self.pos_embed = nn.Parameter(torch.randn(1, self.TIME_STEPS + 1, E))
            nn.init.trunc_normal_(self.pos_embed, std=0.02)
            self.dropout = HipDropout(self.DROPOUT)
        for i in range(4, 16):
            if self._active(i):
                setattr(self, f"layer{i}",
                        TransformerEncoderBlock(E, self.NUM_HEADS, self.MLP_DIM))
        if self._active(16):
            self.layer16 = HipLayerNorm(E)
        if self._active(17):
            self.layer17 = HipLinear(E, self.NUM_CLASSES)

    def forward(self, x):
        # input: (batch, n_mfcc, time_steps)
        if self._active(1):
            x = self.layer1(x.transpose(1, 2))
        if self._active(2):
            cls = self.cls_token.expand(x.size(0), -1, -1)
            x = torch.cat([cls, x], dim=1)
        if self._active(3):
            x = self.dropout(x + self.pos_embed)
        for i in range(4, 16):
            if self._active(i):
                x = getattr(self, f"layer{i}")(x)
        if self._active(16):
            x = self.layer16(x[:, 0])
        if self._active(17):
            x = self.layer17(x)
        return x
