"""Transformer decoder stack (optionally auto-regressive, rel-pos bias).

Functional parity with reference unicore/modules/transformer_decoder.py:25-180,
with the same MI355X broadcast optimization as the encoder: the rel-pos bias
stays (1, H, q, k) and causal/padding masks stay small; the fused softmax
kernel broadcasts them instead of materializing (B*H, q, k) tensors.
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layer_norm import LayerNorm
from .transformer_decoder_layer import TransformerDecoderLayer
from .transformer_encoder import additive_padding_mask, build_rel_pos_table


def build_future_mask(seq_len):
    """Strictly-upper-triangular -inf mask (causal attention)."""
    return torch.triu(torch.full([seq_len, seq_len], float("-inf")), 1)


class TransformerDecoder(nn.Module):
    def __init__(
        self,
        decoder_layers=6,
        embed_dim=768,
        ffn_embed_dim=3072,
        attention_heads=8,
        emb_dropout=0.1,
        dropout=0.1,
        attention_dropout=0.1,
        activation_dropout=0.0,
        max_seq_len=256,
        activation_fn="gelu",
        rel_pos=True,
        rel_pos_bins=32,
        max_rel_pos=128,
        post_ln=False,
        auto_regressive=True,
    ):
        super().__init__()
        self.emb_dropout, self.max_seq_len = emb_dropout, max_seq_len
        self.embed_dim, self.attention_heads = embed_dim, attention_heads
        self.emb_layer_norm = LayerNorm(embed_dim)
        self.auto_regressive = auto_regressive
        self._future_mask = (
            build_future_mask(max_seq_len) if auto_regressive else None
        )
        self.final_layer_norm = None if post_ln else LayerNorm(embed_dim)

        self.layers = nn.ModuleList(
            TransformerDecoderLayer(
                embed_dim=embed_dim, ffn_embed_dim=ffn_embed_dim,
                attention_heads=attention_heads, dropout=dropout,
                attention_dropout=attention_dropout,
                activation_dropout=activation_dropout,
                activation_fn=activation_fn, post_ln=post_ln,
            )
            for _ in range(decoder_layers)
        )

        self.rel_pos = rel_pos
        if rel_pos:
            assert rel_pos_bins % 2 == 0
            self.rel_pos_bins, self.max_rel_pos = rel_pos_bins, max_rel_pos
            self.relative_attention_bias = nn.Embedding(
                rel_pos_bins, attention_heads
            )
            self.rp_bucket = build_rel_pos_table(
                max_seq_len, rel_pos_bins, max_rel_pos
            )

    def get_rel_pos_bias(self, h):
        if self.rp_bucket.device != h.device:
            self.rp_bucket = self.rp_bucket.to(h.device)
        L = h.size(1)
        per_pair = F.embedding(
            self.rp_bucket[:L, :L], self.relative_attention_bias.weight
        )
        return per_pair.permute([2, 0, 1]).contiguous()  # (H, q, k)

    def _causal_bias(self, h, seq_len):
        if self._future_mask.device != h.device:
            self._future_mask = self._future_mask.to(h.device)
        if self._future_mask.dtype != h.dtype:
            self._future_mask = self._future_mask.type_as(h)
        return self._future_mask[:seq_len, :seq_len].view(
            1, 1, seq_len, seq_len
        )

    def forward(
        self,
        emb,
        encoder_out: Optional[torch.Tensor] = None,
        padding_mask: Optional[torch.Tensor] = None,
        encoder_padding_mask: Optional[torch.Tensor] = None,
        attn_mask: Optional[torch.Tensor] = None,
        encoder_attn_mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        bsz, seq_len = emb.size(0), emb.size(1)
        h = self.emb_layer_norm(emb)
        h = F.dropout(h, p=self.emb_dropout, training=self.training)

        if padding_mask is not None:
            h = h * (1 - padding_mask.unsqueeze(-1).type_as(h))

        attn_bias = None
        if self.rel_pos:
            attn_bias = self.get_rel_pos_bias(h).unsqueeze(0)  # (1, H, q, k)
        if self.auto_regressive:
            causal = self._causal_bias(h, seq_len)
            attn_bias = causal if attn_bias is None else attn_bias + causal
        if attn_mask is not None:
            # user-provided additive mask, reference API shape (B*H, q, k)
            attn_mask = attn_mask.view(bsz, -1, seq_len, seq_len)
            attn_bias = attn_mask if attn_bias is None else attn_mask + attn_bias

        pad_bias = additive_padding_mask(padding_mask, h)
        enc_pad_bias = additive_padding_mask(encoder_padding_mask, h)

        for layer in self.layers:
            h = layer(
                h,
                encoder_out=encoder_out,
                padding_mask=pad_bias,
                attn_bias=attn_bias,
                encoder_padding_mask=enc_pad_bias,
                encoder_attn_bias=encoder_attn_mask,
            )

        if self.final_layer_norm is not None:
            h = self.final_layer_norm(h)
        return h
