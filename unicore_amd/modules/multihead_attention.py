"""Multi-head attention with fused softmax+bias+dropout inner step.

Functional parity with reference unicore/modules/multihead_attention.py
(SelfMultiheadAttention:16-105, CrossMultiheadAttention). The attention
weights are materialized O(L^2) (bmm -> fused softmax_dropout -> bmm), which
matches the reference's scope; the fused kernel carries the additive pair
bias and key-padding mask into softmax.
"""

from typing import Optional

import torch
from torch import Tensor, nn

from .softmax_dropout import softmax_dropout


class _QKVSplit(torch.autograd.Function):
    """Fused head-split: qkv (B, L, 3E) -> q,k,v each (B*H, L, D), with the
    q-scaling folded in (one HIP permute-copy each way instead of the
    chunk + 3x transpose-contiguous + scale chain and its backward cat)."""

    @staticmethod
    def forward(ctx, qkv, num_heads, scale):
        from unicore_amd import ops

        q, k, v = ops.qkv_split_fwd(qkv, num_heads, scale)
        ctx.num_heads = num_heads
        ctx.scale = scale
        ctx.bsz = qkv.shape[0]
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        from unicore_amd import ops

        dqkv = ops.qkv_split_bwd(
            dq.contiguous(), dk.contiguous(), dv.contiguous(),
            ctx.bsz, ctx.num_heads, ctx.scale,
        )
        return dqkv, None, None


class SelfMultiheadAttention(nn.Module):
    def __init__(
        self,
        embed_dim,
        num_heads,
        dropout=0.1,
        bias=True,
        scaling_factor=1,
    ):
        super().__init__()
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.dropout = dropout

        self.head_dim = embed_dim // num_heads
        assert (
            self.head_dim * num_heads == self.embed_dim
        ), "embed_dim must be divisible by num_heads"
        self.scaling = (self.head_dim * scaling_factor) ** -0.5

        self.in_proj = nn.Linear(embed_dim, embed_dim * 3, bias=bias)
        self.out_proj = nn.Linear(embed_dim, embed_dim, bias=bias)

    def forward(
        self,
        query,
        key_padding_mask: Optional[Tensor] = None,
        attn_bias: Optional[Tensor] = None,
        return_attn: bool = False,
    ) -> Tensor:
        bsz, tgt_len, embed_dim = query.size()
        assert embed_dim == self.embed_dim

        qkv = self.in_proj(query)
        use_fused_split = False
        if qkv.is_cuda and self.head_dim % 8 == 0:
            from unicore_amd import ops

            use_fused_split = ops.gpu_kernels_available()
        if use_fused_split:
            q, k, v = _QKVSplit.apply(qkv.contiguous(), self.num_heads, self.scaling)
        else:
            q, k, v = qkv.chunk(3, dim=-1)
            q = (
                q.view(bsz, tgt_len, self.num_heads, self.head_dim)
                .transpose(1, 2)
                .contiguous()
                .view(bsz * self.num_heads, -1, self.head_dim)
                * self.scaling
            )
            k = (
                k.view(bsz, -1, self.num_heads, self.head_dim)
                .transpose(1, 2)
                .contiguous()
                .view(bsz * self.num_heads, -1, self.head_dim)
            )
            v = (
                v.view(bsz, -1, self.num_heads, self.head_dim)
                .transpose(1, 2)
                .contiguous()
                .view(bsz * self.num_heads, -1, self.head_dim)
            )

        assert k is not None
        src_len = k.size(1)

        attn_weights = torch.bmm(q, k.transpose(1, 2))

        assert list(attn_weights.size()) == [bsz * self.num_heads, tgt_len, src_len]

        mask = None
        if key_padding_mask is not None and key_padding_mask.dim() == 0:
            key_padding_mask = None
        if key_padding_mask is not None:
            # additive float mask, (bsz, src_len) or (bsz, 1, 1, src_len);
            # broadcast over heads + query positions by the fused kernel
            assert key_padding_mask.size(0) == bsz
            assert key_padding_mask.size(-1) == src_len
            mask = key_padding_mask.view(bsz, 1, 1, src_len).to(attn_weights.dtype)

        attn_weights = attn_weights.view(bsz, self.num_heads, tgt_len, src_len)
        if not return_attn:
            attn = softmax_dropout(
                attn_weights,
                self.dropout,
                self.training,
                mask=mask,
                bias=attn_bias,
            )
        else:
            attn_weights = attn_weights + (mask if mask is not None else 0)
            if attn_bias is not None:
                attn_weights = attn_weights + attn_bias
            attn = softmax_dropout(
                attn_weights, self.dropout, self.training, inplace=False
            )

        attn = attn.view(bsz * self.num_heads, tgt_len, src_len)
        o = torch.bmm(attn, v)
        assert list(o.size()) == [bsz * self.num_heads, tgt_len, self.head_dim]

        o = (
            o.view(bsz, self.num_heads, tgt_len, self.head_dim)
            .transpose(1, 2)
            .contiguous()
            .view(bsz, tgt_len, embed_dim)
        )
        o = self.out_proj(o)
        if not return_attn:
            return o
        else:
            return o, attn_weights, attn


class CrossMultiheadAttention(nn.Module):
    def __init__(
        self,
        embed_dim,
        num_heads,
        dropout=0.1,
        bias=True,
        scaling_factor=1,
    ):
        super().__init__()
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.dropout = dropout

        self.head_dim = embed_dim // num_heads
        assert (
            self.head_dim * num_heads == self.embed_dim
        ), "embed_dim must be divisible by num_heads"
        self.scaling = (self.head_dim * scaling_factor) ** -0.5

        self.q_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.k_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.v_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.out_proj = nn.Linear(embed_dim, embed_dim, bias=bias)

    def forward(
        self,
        query,
        key,
        value,
        key_padding_mask: Optional[Tensor] = None,
        attn_bias: Optional[Tensor] = None,
    ) -> Tensor:
        bsz, tgt_len, embed_dim = query.size()
        assert embed_dim == self.embed_dim

        q = self.q_proj(query)
        k = self.k_proj(key)
        v = self.v_proj(value)

        q = (
            q.view(bsz, tgt_len, self.num_heads, self.head_dim)
            .transpose(1, 2)
            .contiguous()
            .view(bsz * self.num_heads, -1, self.head_dim)
            * self.scaling
        )
        if k is not None:
            k = (
                k.view(bsz, -1, self.num_heads, self.head_dim)
                .transpose(1, 2)
                .contiguous()
                .view(bsz * self.num_heads, -1, self.head_dim)
            )
        if v is not None:
            v = (
                v.view(bsz, -1, self.num_heads, self.head_dim)
                .transpose(1, 2)
                .contiguous()
                .view(bsz * self.num_heads, -1, self.head_dim)
            )

        assert k is not None
        src_len = k.size(1)

        attn_weights = torch.bmm(q, k.transpose(1, 2))

        assert list(attn_weights.size()) == [bsz * self.num_heads, tgt_len, src_len]

        mask = None
        if key_padding_mask is not None and key_padding_mask.dim() == 0:
            key_padding_mask = None
        if key_padding_mask is not None:
            assert key_padding_mask.size(0) == bsz
            assert key_padding_mask.size(1) == src_len
            mask = key_padding_mask.view(bsz, 1, 1, src_len).to(attn_weights.dtype)

        attn_weights = attn_weights.view(bsz, self.num_heads, tgt_len, src_len)
        attn = softmax_dropout(
            attn_weights, self.dropout, self.training, mask=mask, bias=attn_bias
        ).view(bsz * self.num_heads, tgt_len, src_len)

        o = torch.bmm(attn, v)
        assert list(o.size()) == [bsz * self.num_heads, tgt_len, self.head_dim]

        o = (
            o.view(bsz, self.num_heads, tgt_len, self.head_dim)
            .transpose(1, 2)
            .contiguous()
            .view(bsz, tgt_len, embed_dim)
        )
        o = self.out_proj(o)
        return o
