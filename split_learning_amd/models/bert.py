"""BERT partitioned models.

BERT_AGNEWS (15 units — src/model/BERT_AGNEWS.py:167-219): layer1 embeddings,
layers 2..13 full encoder blocks, layer14 pooler, layer15 4-class classifier.

BERT_EMOTION (27 units — other/Vanilla_SL/src/model/BERT_EMOTION.py:184-430):
layer1 embeddings; layers 2..25 alternate [SelfAttention, SelfOutput] and
[Intermediate, Output] ModuleList halves of the 12 encoder blocks; layer26
pooler; layer27 classifier.

All dense math is HIP-backed (ops.modules); attention runs through
attention_core (MFMA batched GEMM + fused softmax on GPU).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.modules import (HipDropout, HipEmbedding, HipGELU, HipLayerNorm,
                           HipLinear, HipTanh, attention_core)
from .partitioned import PartitionedModel


class DotDict(dict):
    """Attribute-style dict used for HF-style .config compatibility
    (src/model/BERT_AGNEWS.py:5-9)."""

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError:
            raise AttributeError(k)

    def __setattr__(self, k, v):
        self[k] = v

    def __delattr__(self, k):
        del self[k]


class BertEmbeddings(nn.Module):
    def __init__(self, vocab_size, hidden_size, max_position_embeddings, type_vocab_size,
                 dropout_prob):
        super().__init__()
        self.word_embeddings = HipEmbedding(vocab_size, hidden_size, padding_idx=0)
        self.position_embeddings = HipEmbedding(max_position_embeddings, hidden_size)
        self.token_type_embeddings = HipEmbedding(type_vocab_size, hidden_size)
        self.LayerNorm = HipLayerNorm(hidden_size, eps=1e-12)
        self.dropout = HipDropout(dropout_prob)

    def forward(self, input_ids, token_type_ids=None):
        seq_len = input_ids.size(1)
        pos = torch.arange(seq_len, dtype=torch.long, device=input_ids.device)
        pos = pos.unsqueeze(0).expand_as(input_ids)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        emb = (self.word_embeddings(input_ids)
               + self.position_embeddings(pos)
               + self.token_type_embeddings(token_type_ids))
        return self.dropout(self.LayerNorm(emb))


class BertSdpaSelfAttention(nn.Module):
    def __init__(self, hidden_size, num_attention_heads, dropout_prob):
        super().__init__()
        self.num_attention_heads = num_attention_heads
        self.attention_head_size = hidden_size // num_attention_heads
        self.all_head_size = self.num_attention_heads * self.attention_head_size
        self.query = HipLinear(hidden_size, self.all_head_size)
        self.key = HipLinear(hidden_size, self.all_head_size)
        self.value = HipLinear(hidden_size, self.all_head_size)
        self.dropout = HipDropout(dropout_prob)

    def forward(self, hidden_states, attention_mask=None):
        B, S, E = hidden_states.shape
        H, hd = self.num_attention_heads, self.attention_head_size
        q = self.query(hidden_states).view(B, S, H, hd).permute(0, 2, 1, 3).reshape(B * H, S, hd).contiguous()
        k = self.key(hidden_states).view(B, S, H, hd).permute(0, 2, 1, 3).reshape(B * H, S, hd).contiguous()
        v = self.value(hidden_states).view(B, S, H, hd).permute(0, 2, 1, 3).reshape(B * H, S, hd).contiguous()
        ctx = attention_core(q, k, v, dropout_p=self.dropout.p, training=self.training)
        return ctx.reshape(B, H, S, hd).permute(0, 2, 1, 3).reshape(B, S, E)


def _dense_drop_res_ln(dense, dropout, ln, hidden_states, input_tensor,
                       training):
    """Shared sublayer epilogue: bias rides in the GEMM, then ONE fused
    dropout+residual+LayerNorm kernel on GPU (norm.hip drop_res_ln_fwd —
    SURVEY §2.4 fused bias-residual-LN row); stock ops on CPU."""
    h = dense(hidden_states)
    if h.is_cuda:
        from ..ops.functional import dropout_residual_layer_norm
        return dropout_residual_layer_norm(
            h, input_tensor, ln.weight, ln.bias, eps=ln.eps,
            p=dropout.p, training=training)
    return ln(dropout(h) + input_tensor)


class BertSelfOutput(nn.Module):
    def __init__(self, hidden_size, dropout_prob):
        super().__init__()
        self.dense = HipLinear(hidden_size, hidden_size)
        self.LayerNorm = HipLayerNorm(hidden_size, eps=1e-12)
        self.dropout = HipDropout(dropout_prob)

    def forward(self, hidden_states, input_tensor):
        return _dense_drop_res_ln(self.dense, self.dropout, self.LayerNorm,
                                  hidden_states, input_tensor, self.training)


class BertAttention(nn.Module):
    def __init__(self, hidden_size, num_attention_heads, dropout_prob):
        super().__init__()
        self.self = BertSdpaSelfAttention(hidden_size, num_attention_heads, dropout_prob)
        self.output = BertSelfOutput(hidden_size, dropout_prob)

    def forward(self, hidden_states):
        return self.output(self.self(hidden_states), hidden_states)


class BertIntermediate(nn.Module):
    def __init__(self, hidden_size, intermediate_size):
        super().__init__()
        self.dense = HipLinear(hidden_size, intermediate_size)
        self.intermediate_act_fn = HipGELU()

    def forward(self, hidden_states):
        return self.intermediate_act_fn(self.dense(hidden_states))


class BertOutput(nn.Module):
    def __init__(self, hidden_size, intermediate_size, dropout_prob):
        super().__init__()
        self.dense = HipLinear(intermediate_size, hidden_size)
        self.LayerNorm = HipLayerNorm(hidden_size, eps=1e-12)
        self.dropout = HipDropout(dropout_prob)

    def forward(self, hidden_states, input_tensor):
        return _dense_drop_res_ln(self.dense, self.dropout, self.LayerNorm,
                                  hidden_states, input_tensor, self.training)


class BertLayer(nn.Module):
    def __init__(self, hidden_size, num_attention_heads, intermediate_size, dropout_prob):
        super().__init__()
        self.attention = BertAttention(hidden_size, num_attention_heads, dropout_prob)
        self.intermediate = BertIntermediate(hidden_size, intermediate_size)
        self.output = BertOutput(hidden_size, intermediate_size, dropout_prob)

    def forward(self, hidden_states):
        attn = self.attention(hidden_states)
        return self.output(self.intermediate(attn), attn)


class BertPooler(nn.Module):
    def __init__(self, hidden_size):
        super().__init__()
        self.dense = HipLinear(hidden_size, hidden_size)
        self.activation = HipTanh()

    def forward(self, hidden_states):
        return self.activation(self.dense(hidden_states[:, 0]))


class BertClassifier(nn.Module):
    def __init__(self, hidden_size, num_labels, dropout_prob=0.1):
        super().__init__()
        self.dropout = HipDropout(dropout_prob)
        self.classifier = HipLinear(hidden_size, num_labels)

    def forward(self, pooled):
        return self.classifier(self.dropout(pooled))


class BERT_AGNEWS(PartitionedModel):
    TOTAL_UNITS = 15

    def __init__(self, vocab_size=28996, hidden_size=768, num_attention_heads=12,
                 intermediate_size=3072, max_position_embeddings=512, type_vocab_size=2,
                 dropout_prob=0.1, n_block=12, start_layer=0, end_layer=15):
        self._hp = (vocab_size, hidden_size, num_attention_heads, intermediate_size,
                    max_position_embeddings, type_vocab_size, dropout_prob)
        super().__init__(start_layer, end_layer)
        self.config = DotDict(
            model_type="bert", vocab_size=vocab_size, hidden_size=hidden_size,
            num_attention_heads=num_attention_heads, intermediate_size=intermediate_size,
            max_position_embeddings=max_position_embeddings,
            bos_token_id=101, eos_token_id=102, pad_token_id=0,
            is_encoder_decoder=False, tie_word_embeddings=False,
            use_return_dict=True, output_attentions=False, output_hidden_states=False,
        )

    def _build(self):
        (V, Hd, Ha, I, P, T, dp) = self._hp
        if self._active(1):
            self.layer1 = BertEmbeddings(V, Hd, P, T, dp)
        for i in range(2, 14):
            if self._active(i):
                setattr(self, f"layer{i}", BertLayer(Hd, Ha, I, dp))
        if self._active(14):
            self.layer14 = BertPooler(Hd)
        if self._active(15):
            self.layer15 = BertClassifier(Hd, 4)

    def forward(self, input_ids=None, token_type_ids=None, **kwargs):
        x = input_ids
        if self._active(1):
            x = self.layer1(x, token_type_ids)
        for i in range(2, 14):
            if self._active(i):
                x = getattr(self, f"layer{i}")(x)
        if self._active(14):
            x = self.layer14(x)
        if self._active(15):
            x = self.layer15(x)
        return x


class BERT_EMOTION(PartitionedModel):
    TOTAL_UNITS = 27

    def __init__(self, start_layer=0, end_layer=27, vocab_size=30522, hidden_size=768,
                 intermediate_size=3072, num_attention_heads=12, num_labels=4,
                 max_position_embeddings=512, type_vocab_size=2, dropout_prob=0.1,
                 num_hidden_layers=12):
        self._hp = (vocab_size, hidden_size, num_attention_heads, intermediate_size,
                    max_position_embeddings, type_vocab_size, dropout_prob, num_labels)
        super().__init__(start_layer, end_layer)

    def _build(self):
        (V, Hd, Ha, I, P, T, dp, L) = self._hp
        if self._active(1):
            self.layer1 = BertEmbeddings(V, Hd, P, T, dp)
        for i in range(2, 26):
            if not self._active(i):
                continue
            if i % 2 == 0:  # attention half
                setattr(self, f"layer{i}", nn.ModuleList([
                    BertSdpaSelfAttention(Hd, Ha, dp), BertSelfOutput(Hd, dp)]))
            else:           # mlp half
                setattr(self, f"layer{i}", nn.ModuleList([
                    BertIntermediate(Hd, I), BertOutput(Hd, I, dp)]))
        if self._active(26):
            self.layer26 = BertPooler(Hd)
        if self._active(27):
            self.layer27 = BertClassifier(Hd, L, self._hp[6])

    def forward(self, x, attention_mask=None, token_type_ids=None):
        if self._active(1):
            x = self.layer1(x, token_type_ids)
        for i in range(2, 26):
            if not self._active(i):
                continue
            mod = getattr(self, f"layer{i}")
            if i % 2 == 0:
                x = mod[1](mod[0](x, attention_mask), x)
            else:
                x = mod[1](mod[0](x), x)
        if self._active(26):
            x = self.layer26(x)
        if self._active(27):
            x = self.layer27(x)
        return x
