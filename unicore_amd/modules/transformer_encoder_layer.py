"""Transformer encoder layer built around the fused kernel set.

Functional parity with reference
unicore/modules/transformer_encoder_layer.py:56-98 (pre/post-LN BERT/XLM
layer). On GPU with the extension loaded, the three Linears run bias-free
and each bias rides the following fused elementwise op (dropout+residual /
gelu+dropout), whose backward emits the bias grad as a deterministic
column sum — no eager activation-sized ``.sum`` re-reads.
"""

import os
from typing import Optional

import torch.nn.functional as F
from torch import Tensor, nn

from unicore_amd import utils

from .dropout_add import dropout_add
from .dropout_add_ln import dropout_add_ln
from .gelu_dropout import gelu_dropout
from .layer_norm import LayerNorm
from .multihead_attention import SelfMultiheadAttention


class TransformerEncoderLayer(nn.Module):
    def __init__(
        self,
        embed_dim=768,
        ffn_embed_dim=3072,
        attention_heads=8,
        dropout=0.1,
        attention_dropout=0.1,
        activation_dropout=0.0,
        activation_fn="gelu",
        post_ln=False,
    ):
        super().__init__()
        self.embed_dim, self.attention_heads = embed_dim, attention_heads
        self.dropout, self.attention_dropout = dropout, attention_dropout
        self.activation_dropout = activation_dropout
        self.act = utils.get_activation_fn(activation_fn)
        self._fuse_gelu = activation_fn == "gelu"

        self.self_attn = SelfMultiheadAttention(
            embed_dim, attention_heads, dropout=attention_dropout
        )
        self.self_attn_layer_norm = LayerNorm(embed_dim)
        self.fc1 = nn.Linear(embed_dim, ffn_embed_dim)
        self.fc2 = nn.Linear(ffn_embed_dim, embed_dim)
        self.final_layer_norm = LayerNorm(embed_dim)
        self.post_ln = post_ln

    def _can_fold_biases(self, h):
        """True when every Linear bias in this layer can ride a fused op."""
        if not h.is_cuda or os.environ.get("UNICORE_FOLD_BIAS", "1") != "1":
            return False
        from unicore_amd import ops

        if not (ops.gpu_kernels_available() and self._fuse_gelu):
            return False
        biases = (self.self_attn.out_proj.bias, self.fc1.bias, self.fc2.bias)
        return all(
            b is not None and ops.colsum_supported(b.numel()) for b in biases
        )

    def forward(
        self,
        x: Tensor,
        attn_bias: Optional[Tensor] = None,
        padding_mask: Optional[Tensor] = None,
        return_attn: bool = False,
    ) -> Tensor:
        """Pre-LN or post-LN ordering per ``self.post_ln``."""
        fold = self._can_fold_biases(x)

        skip = x
        h = x if self.post_ln else self.self_attn_layer_norm(x)
        h = self.self_attn(
            query=h,
            key_padding_mask=padding_mask,
            attn_bias=attn_bias,
            return_attn=return_attn,
            skip_out_bias=fold,
        )
        if return_attn:
            h, attn_weights, attn_probs = h
        attn_bias_fold = self.self_attn.out_proj.bias if fold else None
        if self.post_ln:
            # single kernel: dropout + residual + LN (the sum never makes
            # an extra HBM round-trip)
            h = dropout_add_ln(
                h, skip, self.self_attn_layer_norm, self.dropout,
                self.training, bias=attn_bias_fold,
            )
        else:
            h = dropout_add(
                h, skip, self.dropout, self.training, bias=attn_bias_fold,
            )

        skip = h
        if not self.post_ln:
            h = self.final_layer_norm(h)
        if fold:
            h = F.linear(h, self.fc1.weight)
            h = gelu_dropout(h, self.activation_dropout, self.training,
                             bias=self.fc1.bias)
            h = F.linear(h, self.fc2.weight)
            fc2_bias_fold = self.fc2.bias
        else:
            h = self.fc1(h)
            if self._fuse_gelu and h.is_cuda:
                h = gelu_dropout(h, self.activation_dropout, self.training)
            else:
                h = F.dropout(self.act(h), p=self.activation_dropout,
                              training=self.training)
            h = self.fc2(h)
            fc2_bias_fold = None
        if self.post_ln:
            h = dropout_add_ln(
                h, skip, self.final_layer_norm, self.dropout, self.training,
                bias=fc2_bias_fold,
            )
        else:
            h = dropout_add(h, skip, self.dropout, self.training,
                            bias=fc2_bias_fold)

        if return_attn:
            return h, attn_weights, attn_probs
        return h
