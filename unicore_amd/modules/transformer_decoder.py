"""Transformer decoder stack (optionally auto-regressive, rel-pos bias).

Functional parity with reference unicore/modules/transformer_decoder.py:25-180,
with the same MI355X broadcast optimization as the encoder: rel-pos bias stays
(1, H, q, k) and the causal/padding masks stay small; the fused softmax kernel
broadcasts them.
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layer_norm import LayerNorm
from .transformer_decoder_layer import TransformerDecoderLayer
from .transformer_encoder import relative_position_bucket


def build_future_mask(seq_len):
    return torch.triu(
        torch.full([seq_len, seq_len], float("-inf")), 1
    )


class TransformerDecoder(nn.Module):
    def __init__(
        self,
        decoder_layers: int = 6,
        embed_dim: int = 768,
        ffn_embed_dim: int = 3072,
        attention_heads: int = 8,
        emb_dropout: float = 0.1,
        dropout: float = 0.1,
        attention_dropout: float = 0.1,
        activation_dropout: float = 0.0,
        max_seq_len: int = 256,
        activation_fn: str = "gelu",
        rel_pos: bool = True,
        rel_pos_bins: int = 32,
        max_rel_pos: int = 128,
        post_ln: bool = False,
        auto_regressive: bool = True,
    ) -> None:
        super().__init__()
        self.emb_dropout = emb_dropout
        self.max_seq_len = max_seq_len
        self.embed_dim = embed_dim
        self.attention_heads = attention_heads
        self.emb_layer_norm = LayerNorm(self.embed_dim)
        self.auto_regressive = auto_regressive
        if self.auto_regressive:
            self._future_mask = build_future_mask(self.max_seq_len)
        else:
            self._future_mask = None
        if not post_ln:
            self.final_layer_norm = LayerNorm(self.embed_dim)
        else:
            self.final_layer_norm = None

        self.layers = nn.ModuleList(
            [
                TransformerDecoderLayer(
                    embed_dim=self.embed_dim,
                    ffn_embed_dim=ffn_embed_dim,
                    attention_heads=attention_heads,
                    dropout=dropout,
                    attention_dropout=attention_dropout,
                    activation_dropout=activation_dropout,
                    activation_fn=activation_fn,
                    post_ln=post_ln,
                )
                for _ in range(decoder_layers)
            ]
        )

        self.rel_pos = rel_pos
        if self.rel_pos:
            assert rel_pos_bins % 2 == 0
            self.rel_pos_bins = rel_pos_bins
            self.max_rel_pos = max_rel_pos
            self.relative_attention_bias = nn.Embedding(
                self.rel_pos_bins, self.attention_heads
            )
            seq_len = self.max_seq_len
            context_position = torch.arange(seq_len, dtype=torch.long)[:, None]
            memory_position = torch.arange(seq_len, dtype=torch.long)[None, :]
            relative_position = memory_position - context_position
            self.rp_bucket = relative_position_bucket(
                relative_position,
                num_buckets=self.rel_pos_bins,
                max_distance=self.max_rel_pos,
            )
            self.rp_bucket -= self.rp_bucket.min()

    def get_rel_pos_bias(self, x):
        if self.rp_bucket.device != x.device:
            self.rp_bucket = self.rp_bucket.to(x.device)
        seq_len = x.size(1)
        rp_bucket = self.rp_bucket[:seq_len, :seq_len]
        values = F.embedding(rp_bucket, self.relative_attention_bias.weight)
        values = values.permute([2, 0, 1])
        return values.contiguous()  # (H, q, k)

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
        x = self.emb_layer_norm(emb)
        x = F.dropout(x, p=self.emb_dropout, training=self.training)

        if padding_mask is not None:
            x = x * (1 - padding_mask.unsqueeze(-1).type_as(x))

        attn_bias = None
        if self.rel_pos:
            attn_bias = self.get_rel_pos_bias(x).unsqueeze(0)  # (1, H, q, k)
        if self.auto_regressive:
            if self._future_mask.device != x.device:
                self._future_mask = self._future_mask.to(x.device)
            if self._future_mask.dtype != x.dtype:
                self._future_mask = self._future_mask.type_as(x)
            causal = self._future_mask[:seq_len, :seq_len].view(1, 1, seq_len, seq_len)
            attn_bias = causal if attn_bias is None else attn_bias + causal
        if attn_mask is not None:
            attn_mask = attn_mask.view(bsz, -1, seq_len, seq_len)
            attn_bias = attn_mask if attn_bias is None else attn_mask + attn_bias

        additive_pad = None
        if padding_mask is not None:
            additive_pad = torch.zeros(
                (bsz, 1, 1, seq_len), dtype=x.dtype, device=x.device
            )
            additive_pad.masked_fill_(
                padding_mask.view(bsz, 1, 1, seq_len).to(torch.bool), float("-inf")
            )

        additive_enc_pad = None
        if encoder_padding_mask is not None:
            enc_len = encoder_padding_mask.size(-1)
            additive_enc_pad = torch.zeros(
                (bsz, 1, 1, enc_len), dtype=x.dtype, device=x.device
            )
            additive_enc_pad.masked_fill_(
                encoder_padding_mask.view(bsz, 1, 1, enc_len).to(torch.bool),
                float("-inf"),
            )

        for layer in self.layers:
            x = layer(
                x,
                encoder_out=encoder_out,
                padding_mask=additive_pad,
                attn_bias=attn_bias,
                encoder_padding_mask=additive_enc_pad,
                encoder_attn_bias=encoder_attn_mask,
            )

        if self.final_layer_norm is not None:
            x = self.final_layer_norm(x)

        return x
