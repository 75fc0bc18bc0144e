"""Transformer encoder layer (post-LN) with fused LayerNorm + softmax_dropout.

Functional parity with reference unicore/modules/transformer_encoder_layer.py:56-98.
"""

from typing import Optional

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
        residual = x
        if not self.post_ln:
            x = self.self_attn_layer_norm(x)
        x = self.self_attn(
            query=x,
            key_padding_mask=padding_mask,
            attn_bias=attn_bias,
            return_attn=return_attn,
        )
        if return_attn:
            x, attn_weights, attn_probs = x
        x = dropout_add(x, residual, self.dropout, self.training)
        if self.post_ln:
            x = self.self_attn_layer_norm(x)

        residual = x
        if not self.post_ln:
            x = self.final_layer_norm(x)
        x = self.fc1(x)
        if self._fuse_gelu and x.is_cuda:
            x = gelu_dropout(x, self.activation_dropout, self.training)
        else:
            x = self.activation_fn(x)
            x = F.dropout(x, p=self.activation_dropout, training=self.training)
        x = self.fc2(x)
        x = dropout_add(x, residual, self.dropout, self.training)
        if self.post_ln:
            x = self.final_layer_norm(x)
        if not return_attn:
            return x
        else:
            return x, attn_weights, attn_probs
