"""BERT (GluonNLP-style; parity target: the BERT-base pretrain config in
BASELINE.json).  Built from the framework's hot ops: Embedding, LayerNorm,
fused MHA (CDNA4 batched-MFMA + masked softmax), Dense (MFMA TN GEMM),
GELU, Dropout."""
from __future__ import annotations

import math

from .. import nn
from ..block import HybridBlock
from ... import ndarray as nd

__all__ = ["BERTEncoder", "BERTModel", "BERTForPretrain", "bert_base",
           "bert_large"]


class PositionwiseFFN(HybridBlock):
    def __init__(self, units, hidden_size, dropout=0.1, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self.ffn_1 = nn.Dense(hidden_size, flatten=False,
                                  in_units=units, prefix="ffn_1_")
            self.ffn_2 = nn.Dense(units, flatten=False,
                                  in_units=hidden_size, prefix="ffn_2_")
            self.dropout = nn.Dropout(dropout)
            self.layer_norm = nn.LayerNorm(in_channels=units)

    def forward(self, x):
        out = self.ffn_1(x)
        out = nd.gelu(out)
        out = self.ffn_2(out)
        out = self.dropout(out)
        return self.layer_norm(out + x)

    hybrid_forward = None


class MultiHeadAttentionCell(HybridBlock):
    def __init__(self, units, num_heads, dropout=0.1, causal=False,
                 **kwargs):
        super().__init__(**kwargs)
        self._units = units
        self._num_heads = num_heads
        self._dropout = dropout
        self._causal = causal
        with self.name_scope():
            self.proj_query = nn.Dense(units, flatten=False, in_units=units,
                                       prefix="query_")
            self.proj_key = nn.Dense(units, flatten=False, in_units=units,
                                     prefix="key_")
            self.proj_value = nn.Dense(units, flatten=False, in_units=units,
                                       prefix="value_")
            self.proj_out = nn.Dense(units, flatten=False, in_units=units,
                                     prefix="out_")

    def forward(self, q_in, kv_in=None, mask=None):
        kv_in = kv_in if kv_in is not None else q_in
        q = self.proj_query(q_in)
        k = self.proj_key(kv_in)
        v = self.proj_value(kv_in)
        out = nd.multi_head_attention(q, k, v, self._num_heads, mask=mask,
                                      dropout_p=self._dropout,
                                      causal=self._causal)
        return self.proj_out(out)

    hybrid_forward = None


class TransformerEncoderCell(HybridBlock):
    def __init__(self, units, hidden_size, num_heads, dropout=0.1,
                 causal=False, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self.attention_cell = MultiHeadAttentionCell(
                units, num_heads, dropout, causal=causal, prefix="attn_")
            self.proj_dropout = nn.Dropout(dropout)
            self.layer_norm = nn.LayerNorm(in_channels=units)
            self.ffn = PositionwiseFFN(units, hidden_size, dropout,
                                       prefix="ffn_")

    def forward(self, x, mask=None):
        attn = self.attention_cell(x, mask=mask)
        x = self.layer_norm(x + self.proj_dropout(attn))
        return self.ffn(x)

    hybrid_forward = None


class BERTEncoder(HybridBlock):
    def __init__(self, num_layers=12, units=768, hidden_size=3072,
                 num_heads=12, dropout=0.1, max_length=512, **kwargs):
        super().__init__(**kwargs)
        self._units = units
        with self.name_scope():
            self.position_weight = self.params.get(
                "position_weight", shape=(max_length, units))
            self.dropout_layer = nn.Dropout(dropout)
            self.layer_norm = nn.LayerNorm(in_channels=units)
            self.transformer_cells = nn.HybridSequential()
            for i in range(num_layers):
                self.transformer_cells.add(TransformerEncoderCell(
                    units, hidden_size, num_heads, dropout,
                    prefix=f"transformer{i}_"))

    def forward(self, x, mask=None):
        T = x.shape[1]
        pos = self.position_weight.data(x.context)[0:T]
        x = x + pos.expand_dims(0).astype(x.dtype)
        x = self.layer_norm(x)
        x = self.dropout_layer(x)
        for cell in self.transformer_cells:
            x = cell(x, mask)
        return x

    hybrid_forward = None


class BERTModel(HybridBlock):
    def __init__(self, vocab_size=30528, token_type_vocab_size=2,
                 units=768, num_layers=12, hidden_size=3072, num_heads=12,
                 dropout=0.1, max_length=512, **kwargs):
        super().__init__(**kwargs)
        self._units = units
        with self.name_scope():
            self.word_embed = nn.Embedding(vocab_size, units,
                                           prefix="word_embed_")
            self.token_type_embed = nn.Embedding(token_type_vocab_size,
                                                 units,
                                                 prefix="token_type_embed_")
            self.encoder = BERTEncoder(num_layers, units, hidden_size,
                                       num_heads, dropout, max_length,
                                       prefix="encoder_")
            self.pooler = nn.Dense(units, activation="tanh", flatten=False,
                                   in_units=units, prefix="pooler_")

    def forward(self, inputs, token_types=None, valid_length=None):
        emb = self.word_embed(inputs)
        if token_types is not None:
            emb = emb + self.token_type_embed(token_types)
        seq = self.encoder(emb)
        pooled = self.pooler(seq.slice_axis(1, 0, 1).squeeze(1))
        return seq, pooled

    hybrid_forward = None


class BERTForPretrain(HybridBlock):
    """MLM + NSP heads (GluonNLP model_zoo parity)."""

    def __init__(self, bert: BERTModel, vocab_size=30528, **kwargs):
        super().__init__(**kwargs)
        self.bert = bert
        units = bert._units
        with self.name_scope():
            self.mlm_decoder = nn.HybridSequential(prefix="decoder_")
            self.mlm_decoder.add(nn.Dense(units, flatten=False,
                                          in_units=units))
            self.mlm_decoder.add(nn.GELU())
            self.mlm_decoder.add(nn.LayerNorm(in_channels=units))
            self.mlm_decoder.add(nn.Dense(vocab_size, flatten=False,
                                          in_units=units))
            self.nsp_classifier = nn.Dense(2, in_units=units,
                                           prefix="nsp_")

    def forward(self, inputs, token_types=None):
        seq, pooled = self.bert(inputs, token_types)
        mlm_scores = self.mlm_decoder(seq)
        nsp_scores = self.nsp_classifier(pooled)
        return mlm_scores, nsp_scores

    hybrid_forward = None


def bert_tiny(vocab_size=30528, max_length=512, dropout=0.1, **kwargs):
    """CPU-smoke-sized configuration (examples/tests)."""
    return BERTModel(vocab_size=vocab_size, units=128, num_layers=2,
                     hidden_size=512, num_heads=2, dropout=dropout,
                     max_length=max_length, **kwargs)


def bert_base(vocab_size=30528, max_length=512, dropout=0.1, **kwargs):
    return BERTModel(vocab_size=vocab_size, units=768, num_layers=12,
                     hidden_size=3072, num_heads=12, dropout=dropout,
                     max_length=max_length, **kwargs)


def bert_large(vocab_size=30528, max_length=512, dropout=0.1, **kwargs):
    return BERTModel(vocab_size=vocab_size, units=1024, num_layers=24,
                     hidden_size=4096, num_heads=16, dropout=dropout,
                     max_length=max_length, **kwargs)
