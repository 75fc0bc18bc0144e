"""Transformer encoder layer (post-LN) with fused LayerNorm + softmax_dropout.

Functional parity with reference unicore/modules/transformer_encoder_layer.py:56-98.
"""

from typing import Optional

import os

import torch.nn.functional as F
from torch import Tensor, nn

from unicore_amd import utils

from .dropout_add import dropout_add
from .gelu_dropout import gelu_dropout
from .layer_norm import LayerNorm
from .multihead_attention import SelfMultiheadAttention


class TransformerEncoderLayer(nn.Module):
    """
    Implements a Transformer Encoder Layer used in BERT/XLM style pre-trained
    models.
    """

    def __init__(
        self,
        embed_dim: int = 768,
        ffn_embed_dim: int = 3072,
        attention_heads: int = 8,
        dropout: float = 0.1,
        attention_dropout: float = 0.1,
        activation_dropout: float = 0.0,
        activation_fn: str = "gelu",
        post_ln=False,
    ) -> None:
        super().__init__()

        # Initialize parameters
        self.embed_dim = embed_dim
        self.attention_heads = attention_heads
        self.attention_dropout = attention_dropout

        self.dropout = dropout
        self.activation_dropout = activation_dropout
        self.activation_fn = utils.get_activation_fn(activation_fn)
        self._fuse_gelu = activation_fn == "gelu" 

        self.self_attn = SelfMultiheadAttention(
            self.embed_dim,
            attention_heads,
            dropout=attention_dropout,
        )
        # layer norm associated with the self attention layer
        self.self_attn_layer_norm = LayerNorm(self.embed_dim)
        self.fc1 = nn.Linear(self.embed_dim, ffn_embed_dim)
        self.fc2 = nn.Linear(ffn_embed_dim, self.embed_dim)
        self.final_layer_norm = LayerNorm(self.embed_dim)
        self.post_ln = post_ln

    def forward(
        self,
        x: Tensor,
        attn_bias: Optional[Tensor] = None,
        padding_mask: Optional[Tensor] = None,
        return_attn: bool = False,
    ) -> Tensor:
        """
        LayerNorm is applied either before or after the self-attention/ffn
        modules similar to the original Transformer implementation.
        """
        # On GPU with the kernel extension, the Linears run bias-free and
        # each bias rides the following fused op (add is free there, and
        # the backward emits the bias grad as a deterministic column sum
        # instead of an eager activation-sized .sum re-read per Linear).
        fold_bias = False
        if x.is_cuda and os.environ.get("UNICORE_FOLD_BIAS", "1") == "1":
            from unicore_amd import ops

            fold_bias = (
                ops.gpu_kernels_available()
                and self.self_attn.out_proj.bias is not None
                and ops.colsum_supported(self.self_attn.out_proj.bias.numel())
                and self.fc1.bias is not None
                and ops.colsum_supported(self.fc1.bias.numel())
                and self.fc2.bias is not None
                and ops.colsum_supported(self.fc2.bias.numel())
                and self._fuse_gelu
            )

        residual = x
        if not self.post_ln:
            x = self.self_attn_layer_norm(x)
        x = self.self_attn(
            query=x,
            key_padding_mask=padding_mask,
            attn_bias=attn_bias,
            return_attn=return_attn,
            skip_out_bias=fold_bias,
        )
        if return_attn:
            x, attn_weights, attn_probs = x
        x = dropout_add(
            x, residual, self.dropout, self.training,
            bias=self.self_attn.out_proj.bias if fold_bias else None,
        )
        if self.post_ln:
            x = self.self_attn_layer_norm(x)

        residual = x
        if not self.post_ln:
            x = self.final_layer_norm(x)
        if fold_bias:
            x = F.linear(x, self.fc1.weight)
            x = gelu_dropout(x, self.activation_dropout, self.training,
                             bias=self.fc1.bias)
            x = F.linear(x, self.fc2.weight)
            x = dropout_add(x, residual, self.dropout, self.training,
                            bias=self.fc2.bias)
        else:
            x = self.fc1(x)
            if self._fuse_gelu and x.is_cuda:
                x = gelu_dropout(x, self.activation_dropout, self.training)
            else:
                x = self.activation_fn(x)
                x = F.dropout(x, p=self.activation_dropout,
                              training=self.training)
            x = self.fc2(x)
            x = dropout_add(x, residual, self.dropout, self.training)
        if self.post_ln:
            x = self.final_layer_norm(x)
        if not return_attn:
            return x
        else:
            return x, attn_weights, attn_probs
