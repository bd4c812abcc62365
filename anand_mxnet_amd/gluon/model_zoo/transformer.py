"""Transformer encoder-decoder for machine translation (WMT
"transformer-big" shape — the BASELINE.json config #5; parity target:
the Sockeye/GNMT-class seq2seq models trained with the reference).

Decoder self-attention is causal (CDNA4 masked-softmax kernel)."""
from __future__ import annotations

import math

from .. import nn
from ..block import HybridBlock
from .bert import MultiHeadAttentionCell, PositionwiseFFN
from ... import ndarray as nd
from .. import nn as _gnn


def _set_fp8(use_fp8):
    """Route the Dense layers built inside bert/transformer cells through
    Fp8Dense while constructing an fp8 model."""
    import anand_mxnet_amd.gluon.model_zoo.bert as _bert
    _bert.nn = _FP8NS if use_fp8 else _gnn


class _Fp8Namespace:
    def __getattr__(self, name):
        if name == "Dense":
            return _gnn.Fp8Dense
        return getattr(_gnn, name)


_FP8NS = _Fp8Namespace()

__all__ = ["TransformerEncoder", "TransformerDecoder", "TransformerModel",
           "transformer_big", "transformer_base"]


class TransformerDecoderCell(HybridBlock):
    def __init__(self, units, hidden_size, num_heads, dropout=0.1, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self.self_attn = MultiHeadAttentionCell(units, num_heads,
                                                    dropout, causal=True,
                                                    prefix="selfattn_")
            self.cross_attn = MultiHeadAttentionCell(units, num_heads,
                                                     dropout,
                                                     prefix="crossattn_")
            self.drop1 = nn.Dropout(dropout)
            self.drop2 = nn.Dropout(dropout)
            self.ln1 = nn.LayerNorm(in_channels=units)
            self.ln2 = nn.LayerNorm(in_channels=units)
            self.ffn = PositionwiseFFN(units, hidden_size, dropout,
                                       prefix="ffn_")

    def forward(self, x, mem):
        attn = self.self_attn(x)
        x = self.ln1(x + self.drop1(attn))
        attn = self.cross_attn(x, mem)
        x = self.ln2(x + self.drop2(attn))
        return self.ffn(x)

    hybrid_forward = None


class TransformerEncoder(HybridBlock):
    def __init__(self, num_layers=6, units=1024, hidden_size=4096,
                 num_heads=16, dropout=0.1, **kwargs):
        super().__init__(**kwargs)
        from .bert import TransformerEncoderCell
        with self.name_scope():
            self.cells = nn.HybridSequential()
            for i in range(num_layers):
                self.cells.add(TransformerEncoderCell(
                    units, hidden_size, num_heads, dropout,
                    prefix=f"enc{i}_"))
            self.ln = nn.LayerNorm(in_channels=units)

    def forward(self, x):
        for cell in self.cells:
            x = cell(x)
        return self.ln(x)

    hybrid_forward = None


class TransformerDecoder(HybridBlock):
    def __init__(self, num_layers=6, units=1024, hidden_size=4096,
                 num_heads=16, dropout=0.1, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self.cells = []
            for i in range(num_layers):
                cell = TransformerDecoderCell(units, hidden_size, num_heads,
                                              dropout, prefix=f"dec{i}_")
                self.register_child(cell, f"dec{i}")
                self.cells.append(cell)
            self.ln = nn.LayerNorm(in_channels=units)

    def forward(self, x, mem):
        for cell in self.cells:
            x = cell(x, mem)
        return self.ln(x)

    hybrid_forward = None


class TransformerModel(HybridBlock):
    def __init__(self, src_vocab=32768, tgt_vocab=32768, units=1024,
                 num_layers=6, hidden_size=4096, num_heads=16, dropout=0.1,
                 max_length=1024, tie_weights=True, **kwargs):
        super().__init__(**kwargs)
        self._units = units
        with self.name_scope():
            self.src_embed = nn.Embedding(src_vocab, units,
                                          prefix="src_embed_")
            self.tgt_embed = nn.Embedding(tgt_vocab, units,
                                          prefix="tgt_embed_")
            self.pos_weight = self.params.get("pos_weight",
                                              shape=(max_length, units))
            self.encoder = TransformerEncoder(num_layers, units,
                                              hidden_size, num_heads,
                                              dropout, prefix="encoder_")
            self.decoder = TransformerDecoder(num_layers, units,
                                              hidden_size, num_heads,
                                              dropout, prefix="decoder_")
            self.proj = nn.Dense(tgt_vocab, flatten=False, in_units=units,
                                 prefix="proj_")

    def _embed(self, embed, tokens):
        x = embed(tokens) * math.sqrt(self._units)
        T = tokens.shape[1]
        pos = self.pos_weight.data(tokens.context)[0:T]
        return x + pos.expand_dims(0).astype(x.dtype)

    def forward(self, src_tokens, tgt_tokens):
        mem = self.encoder(self._embed(self.src_embed, src_tokens))
        out = self.decoder(self._embed(self.tgt_embed, tgt_tokens), mem)
        return self.proj(out)

    hybrid_forward = None


def transformer_big(src_vocab=32768, tgt_vocab=32768, use_fp8=False,
                    **kwargs):
    """WMT transformer-big: 6+6 layers, d=1024, 16 heads, ffn 4096.
    use_fp8=True builds the attention/FFN projections as Fp8Dense
    (CDNA4 fp8 e4m3 MFMA forward — config #5)."""
    _set_fp8(use_fp8)
    try:
        return TransformerModel(src_vocab, tgt_vocab, units=1024,
                                num_layers=6, hidden_size=4096,
                                num_heads=16, dropout=0.1, **kwargs)
    finally:
        _set_fp8(False)


def transformer_base(src_vocab=32768, tgt_vocab=32768, **kwargs):
    return TransformerModel(src_vocab, tgt_vocab, units=512, num_layers=6,
                            hidden_size=2048, num_heads=8, dropout=0.1,
                            **kwargs)
