"""Transformer encoder stack with relative-position bucket bias.

Functional parity with reference unicore/modules/transformer_encoder.py
(init_bert_params:16, relative_position_bucket:33, TransformerEncoder:51-163),
with one MI355X-first change: the per-head relative-position bias is NOT
repeated across the batch and the key-padding mask is NOT materialized into
a (B*H, q, k) tensor — both stay small ((1,H,q,k) and (B,1,1,k)) and the
fused softmax kernel applies them via its broadcast addressing, saving a
B*H*q*k-sized HBM round-trip per layer.

The rel-pos machinery is shared with the decoder through the module-level
helpers below.
"""

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layer_norm import LayerNorm
from .transformer_encoder_layer import TransformerEncoderLayer


def _bert_normal_(data):
    # draw on CPU so initialization is identical regardless of device
    data.copy_(data.cpu().normal_(mean=0.0, std=0.02).to(data.device))


def init_bert_params(m):
    """BERT-style init: N(0, 0.02) weights, zero biases/padding rows."""
    if not getattr(m, "can_global_init", True):
        return
    if isinstance(m, nn.Linear):
        _bert_normal_(m.weight.data)
        if m.bias is not None:
            m.bias.data.zero_()
    if isinstance(m, nn.Embedding):
        _bert_normal_(m.weight.data)
        if m.padding_idx is not None:
            m.weight.data[m.padding_idx].zero_()


def relative_position_bucket(relative_position, num_buckets=32,
                             max_distance=128):
    """T5-style signed log-bucketing of relative positions: half the buckets
    cover exact small offsets, the rest grow logarithmically out to
    max_distance; the sign separates look-back from look-ahead."""
    direction = torch.sign(relative_position)
    half = num_buckets // 2
    dist = torch.abs(relative_position)

    exact_span = half // 2
    log_span = half - 1 - exact_span
    scaled = exact_span + torch.ceil(
        torch.log(dist.float() / exact_span)
        / math.log((max_distance - 1) / exact_span)
        * log_span
    ).long()
    scaled = torch.min(scaled, torch.full_like(scaled, half - 1))
    return torch.where(dist < exact_span, dist, scaled) * direction


def build_rel_pos_table(max_seq_len, num_buckets, max_distance):
    """(L, L) bucket-index table, shifted to start at 0."""
    positions = torch.arange(max_seq_len, dtype=torch.long)
    offsets = positions[None, :] - positions[:, None]
    table = relative_position_bucket(
        offsets, num_buckets=num_buckets, max_distance=max_distance
    )
    return table - table.min()


def additive_padding_mask(padding_mask, like):
    """(B, L) bool padding -> small additive (B, 1, 1, L) float mask the
    fused softmax broadcasts over heads and query positions."""
    if padding_mask is None:
        return None
    b, k = padding_mask.size(0), padding_mask.size(-1)
    out = torch.zeros((b, 1, 1, k), dtype=like.dtype, device=like.device)
    out.masked_fill_(
        padding_mask.view(b, 1, 1, k).to(torch.bool), float("-inf")
    )
    return out


class TransformerEncoder(nn.Module):
    def __init__(
        self,
        encoder_layers=6,
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
    ):
        super().__init__()
        self.emb_dropout, self.max_seq_len = emb_dropout, max_seq_len
        self.embed_dim, self.attention_heads = embed_dim, attention_heads
        self.emb_layer_norm = LayerNorm(embed_dim)
        # pre-LN keeps a final norm; post-LN ends normalized already
        self.final_layer_norm = None if post_ln else LayerNorm(embed_dim)

        self.layers = nn.ModuleList(
            TransformerEncoderLayer(
                embed_dim=embed_dim, ffn_embed_dim=ffn_embed_dim,
                attention_heads=attention_heads, dropout=dropout,
                attention_dropout=attention_dropout,
                activation_dropout=activation_dropout,
                activation_fn=activation_fn, post_ln=post_ln,
            )
            for _ in range(encoder_layers)
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
        # assumes tokens arrive in position order
        if self.rp_bucket.device != h.device:
            self.rp_bucket = self.rp_bucket.to(h.device)
        L = h.size(1)
        per_pair = F.embedding(
            self.rp_bucket[:L, :L], self.relative_attention_bias.weight
        )
        return per_pair.permute([2, 0, 1]).contiguous()  # (H, q, k)

    def forward(
        self,
        emb: torch.Tensor,
        attn_mask: Optional[torch.Tensor] = None,
        padding_mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        bsz, seq_len = emb.size(0), emb.size(1)
        h = self.emb_layer_norm(emb)
        h = F.dropout(h, p=self.emb_dropout, training=self.training)

        # padded positions contribute nothing to the representation
        if padding_mask is not None:
            h = h * (1 - padding_mask.unsqueeze(-1).type_as(h))

        # (1, H, q, k) bias — broadcast over batch by the fused kernel,
        # never repeated to (B*H, q, k)
        attn_bias = None
        if self.rel_pos:
            attn_bias = self.get_rel_pos_bias(h).unsqueeze(0)
        if attn_mask is not None:
            # user-provided additive mask, reference API shape (B*H, q, k)
            attn_mask = attn_mask.view(bsz, -1, seq_len, seq_len)
            attn_bias = attn_mask if attn_bias is None else attn_mask + attn_bias

        pad_bias = additive_padding_mask(padding_mask, h)

        for layer in self.layers:
            h = layer(h, padding_mask=pad_bias, attn_bias=attn_bias)

        if self.final_layer_norm is not None:
            h = self.final_layer_norm(h)
        return h
