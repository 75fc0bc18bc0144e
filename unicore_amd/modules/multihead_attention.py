"""Multi-head attention with fused softmax+bias+dropout inner step.

Functional parity with reference unicore/modules/multihead_attention.py
(SelfMultiheadAttention:16-105, CrossMultiheadAttention). The attention
weights are materialized O(L^2) (bmm -> fused softmax_dropout -> bmm), which
matches the reference's scope; the fused kernel carries the additive pair
bias and key-padding mask into softmax.
"""

from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor, nn

import os

from .softmax_dropout import _broadcast_descr, softmax_dropout


def _flash_enabled():
    return os.environ.get("UNICORE_FLASH_ATTN", "0") == "1"


class _FlashAttn(torch.autograd.Function):
    """Flash attention (bf16, head_dim 64): O = dropout(softmax(QK^T +
    bias + mask)) V computed tile-wise by the MFMA kernel — the score
    matrix never touches HBM.  Dropout keep bits are recomputed from the
    philox seed in backward (no stored mask)."""

    @staticmethod
    def forward(ctx, q, k, v, bias, bias_od, mask, mask_od, p, training):
        from unicore_amd import ops

        o, lse, seed = ops.flash_attn_fwd(
            q, k, v, bias, bias_od, mask, mask_od, p, training
        )
        ctx.save_for_backward(q, k, v, o, lse,
                              bias if bias is not None else q.new_empty(0),
                              mask if mask is not None else q.new_empty(0))
        ctx.bias_od = bias_od
        ctx.mask_od = mask_od
        ctx.p = p
        ctx.dropped = training and p > 0
        ctx.seed = int(seed)
        return o

    @staticmethod
    def backward(ctx, d_out):
        from unicore_amd import ops

        q, k, v, o, lse, bias, mask = ctx.saved_tensors
        bias_t = bias if bias.numel() else None
        mask_t = mask if mask.numel() else None
        want_dbias = bias_t is not None and ctx.needs_input_grad[3]
        grads = ops.flash_attn_bwd(
            d_out.contiguous(), q, k, v, o, lse, bias_t, ctx.bias_od,
            want_dbias, mask_t, ctx.mask_od, ctx.p, ctx.dropped, ctx.seed,
        )
        dq, dk, dv = grads[0], grads[1], grads[2]
        dbias = None
        if len(grads) > 3:
            # grads[3] is the finished (nb, L, L) fp32 bias gradient: the
            # kernel fuses the broadcast-batch reduction into dq's dS pass
            # (deterministic mode reduces a materialized dS host-side)
            dbias = grads[3].to(bias_t.dtype)
        return dq, dk, dv, dbias, None, None, None, None, None


def _prep_flash_src(src, batch_dims, k_len, allowed_sq):
    """Normalize an additive bias/mask to the flash kernel's
    (src_nb, src_q, k) + outer_div form; (None, 1, False) if the broadcast
    pattern or shape is unsupported (caller falls back to the materialized
    path)."""
    if src is None:
        return None, 1, True
    if src.shape[-1] != k_len or src.dim() < 2:
        return None, 1, False
    sq = src.shape[-2]
    if sq not in allowed_sq:
        return None, 1, False
    descr = _broadcast_descr(tuple(src.shape[:-2]), batch_dims)
    if descr is None:
        return None, 1, False
    src_nb, outer_div = descr
    t = src.reshape(src_nb, sq, k_len)
    if not t.is_contiguous():
        t = t.contiguous()
    return t, outer_div, True



def _to_heads(t, bsz, num_heads, head_dim, scale=None):
    """(B, L, E) -> (B*H, L, D), optionally folding the query scale."""
    out = t.view(bsz, -1, num_heads, head_dim).transpose(1, 2)
    out = out.contiguous().view(bsz * num_heads, -1, head_dim)
    return out * scale if scale is not None else out


def _from_heads(t, bsz, num_heads, tgt_len, head_dim):
    """(B*H, L, D) -> (B, L, E)."""
    merged = t.view(bsz, num_heads, tgt_len, head_dim).transpose(1, 2)
    return merged.contiguous().view(bsz, tgt_len, num_heads * head_dim)


def _padding_to_additive(kpm, bsz, src_len, dtype):
    """Normalize a key-padding mask to the additive (B, 1, 1, k) form the
    fused softmax broadcasts; scalar placeholders mean "no mask"."""
    if kpm is not None and kpm.dim() == 0:
        kpm = None
    if kpm is None:
        return None
    assert kpm.size(0) == bsz and kpm.size(-1) == src_len
    return kpm.view(bsz, 1, 1, src_len).to(dtype)


class _QKScores(torch.autograd.Function):
    """scores = q @ k^T with a backward that emits dk CONTIGUOUS.

    Autograd's BmmBackward differentiates through the k.transpose(1, 2)
    view, handing downstream consumers (the fused qkv-split backward) a
    transposed dk view whose .contiguous() copied ~100 MB per layer
    (profiled at ~112 us x 12 on the BERT bench). Computing
    dk = dS^T @ q directly is the same TN strided-batched GEMM hipBLASLt
    would run anyway, but its output is already (BH, L, D) contiguous."""

    @staticmethod
    def forward(ctx, q, k):
        ctx.save_for_backward(q, k)
        return torch.bmm(q, k.transpose(1, 2))

    @staticmethod
    def backward(ctx, ds):
        q, k = ctx.saved_tensors
        dq = torch.bmm(ds, k)
        dk = torch.bmm(ds.transpose(1, 2), q)
        return dq, dk


def qk_scores(q, k):
    if q.requires_grad or k.requires_grad:
        return _QKScores.apply(q, k)
    return torch.bmm(q, k.transpose(1, 2))


class _QKVSplit(torch.autograd.Function):
    """Fused head-split: qkv (B, L, 3E) -> q,k,v each (B*H, L, D), with the
    q-scaling folded in (one HIP permute-copy each way instead of the
    chunk + 3x transpose-contiguous + scale chain and its backward cat)."""

    @staticmethod
    def forward(ctx, qkv, bias, num_heads, scale):
        from unicore_amd import ops

        q, k, v = ops.qkv_split_fwd(qkv, num_heads, scale, bias)
        ctx.num_heads = num_heads
        ctx.scale = scale
        ctx.bsz = qkv.shape[0]
        ctx.bias_dtype = bias.dtype if bias is not None else None
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        from unicore_amd import ops

        dqkv, db = ops.qkv_split_bwd(
            dq.contiguous(), dk.contiguous(), dv.contiguous(),
            ctx.bsz, ctx.num_heads, ctx.scale,
            bias_grad=ctx.bias_dtype is not None,
        )
        dbias = db.to(ctx.bias_dtype) if ctx.bias_dtype is not None else None
        return dqkv, dbias, None, None


class _AttnMerge(torch.autograd.Function):
    """(B*H, L, D) -> (B, L, H*D) with 16 B vectors on both sides (torch's
    strided copy walks the permuted side with 2-byte scalars)."""

    @staticmethod
    def forward(ctx, x, bsz, num_heads):
        from unicore_amd import ops

        ctx.bsz = bsz
        ctx.num_heads = num_heads
        return ops.attn_merge(x.contiguous(), bsz, num_heads)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        return (
            ops.attn_merge(grad.contiguous(), ctx.bsz, ctx.num_heads,
                           inverse=True),
            None,
            None,
        )


class SelfMultiheadAttention(nn.Module):
    def __init__(self, embed_dim, num_heads, dropout=0.1, bias=True,
                 scaling_factor=1):
        super().__init__()
        self.embed_dim, self.num_heads = embed_dim, num_heads
        self.dropout = float(dropout)
        self.head_dim = embed_dim // num_heads
        assert self.head_dim * num_heads == embed_dim, \
            "embed_dim must be divisible by num_heads"
        self.scaling = (self.head_dim * scaling_factor) ** -0.5

        self.in_proj = nn.Linear(embed_dim, embed_dim * 3, bias=bias)
        self.out_proj = nn.Linear(embed_dim, embed_dim, bias=bias)

    def forward(self, query, key_padding_mask: Optional[Tensor] = None,
                attn_bias: Optional[Tensor] = None, return_attn: bool = False,
                skip_out_bias: bool = False) -> Tensor:
        """``skip_out_bias``: run out_proj without its bias — the caller
        folds it into the following fused dropout+residual op (which then
        also produces the bias gradient)."""
        bsz, tgt_len, embed_dim = query.shape
        assert embed_dim == self.embed_dim, "input width mismatch"

        use_fused_split = False
        if query.is_cuda and self.head_dim % 8 == 0:
            from unicore_amd import ops

            use_fused_split = ops.gpu_kernels_available()
        if use_fused_split:
            # bias-free GEMM: the split kernel adds the bias (free) and its
            # backward emits the bias grad as a deterministic column sum
            in_bias = self.in_proj.bias
            fold_bias = in_bias is not None and ops.colsum_supported(
                in_bias.numel()
            )
            qkv = F.linear(query, self.in_proj.weight,
                           None if fold_bias else in_bias)
            q, k, v = _QKVSplit.apply(
                qkv.contiguous(), in_bias if fold_bias else None,
                self.num_heads, self.scaling,
            )
        else:
            qkv = self.in_proj(query)
            q, k, v = qkv.chunk(3, dim=-1)
            q = _to_heads(q, bsz, self.num_heads, self.head_dim, self.scaling)
            k = _to_heads(k, bsz, self.num_heads, self.head_dim)
            v = _to_heads(v, bsz, self.num_heads, self.head_dim)

        src_len = k.shape[1]
        mask = _padding_to_additive(key_padding_mask, bsz, src_len, q.dtype)

        # flash path: bf16, head_dim 64, L % 64 == 0, kernel-expressible
        # bias/mask broadcasts — the L x L score matrix never hits HBM.
        # Opt-in (UNICORE_FLASH_ATTN=1): correct and fully tested, but the
        # v1 tile structure does not yet beat hipBLASLt bmm + the fused
        # softmax on BERT-base shapes (see profiles/README.md).
        o = None
        if (
            use_fused_split
            # opt-in for training at short L; always on under no_grad
            # (forward-only flash is ~1.4-3.8x the materialized chain) and
            # for long sequences, where the measured fwd+bwd crossover sits
            # at L ~= 2048-4096 and the L x L score matrix stops fitting
            # (tools/flash_microbench.py --long-seq)
            and (_flash_enabled() or not torch.is_grad_enabled()
                 or tgt_len >= 4096)
            and not return_attn
            and q.dtype == torch.bfloat16
            and self.head_dim == 64
            and tgt_len == src_len
            and tgt_len % 64 == 0
        ):
            batch_dims = (bsz, self.num_heads)
            bias_allowed = (
                {tgt_len}
                if attn_bias is not None and attn_bias.requires_grad
                else {1, tgt_len}
            )
            bias_k, bias_od, ok_b = _prep_flash_src(
                attn_bias, batch_dims, src_len, bias_allowed
            )
            mask_k, mask_od, ok_m = _prep_flash_src(mask, batch_dims, src_len, {1})
            if ok_b and ok_m:
                o = _FlashAttn.apply(
                    q, k, v, bias_k, bias_od, mask_k, mask_od,
                    self.dropout, self.training,
                )

        if o is None:
            # materialized O(L^2) chain: bmm -> fused softmax(+bias+mask
            # +dropout) -> bmm
            scores = qk_scores(q, k)
            assert scores.shape == (bsz * self.num_heads, tgt_len, src_len)
            scores = scores.view(bsz, self.num_heads, tgt_len, src_len)
            if return_attn:
                attn_weights = scores + (0 if mask is None else mask)
                if attn_bias is not None:
                    attn_weights = attn_weights + attn_bias
                attn = softmax_dropout(
                    attn_weights, self.dropout, self.training, inplace=False
                )
            else:
                attn = softmax_dropout(
                    scores, self.dropout, self.training,
                    mask=mask, bias=attn_bias,
                )
            attn = attn.view(bsz * self.num_heads, tgt_len, src_len)
            attn_out = torch.bmm(attn, v)
        else:
            attn_out = o
        assert attn_out.shape == (bsz * self.num_heads, tgt_len, self.head_dim)

        if use_fused_split:
            attn_out = _AttnMerge.apply(attn_out, bsz, self.num_heads)
        else:
            attn_out = _from_heads(attn_out, bsz, self.num_heads, tgt_len,
                                   self.head_dim)
        if skip_out_bias:
            attn_out = F.linear(attn_out, self.out_proj.weight)
        else:
            attn_out = self.out_proj(attn_out)
        if return_attn:
            return attn_out, attn_weights, attn
        return attn_out


class CrossMultiheadAttention(nn.Module):
    def __init__(self, embed_dim, num_heads, dropout=0.1, bias=True,
                 scaling_factor=1):
        super().__init__()
        self.embed_dim, self.num_heads = embed_dim, num_heads
        self.dropout = float(dropout)
        self.head_dim = embed_dim // num_heads
        assert self.head_dim * num_heads == embed_dim, \
            "embed_dim must be divisible by num_heads"
        self.scaling = (self.head_dim * scaling_factor) ** -0.5

        self.q_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.k_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.v_proj = nn.Linear(embed_dim, embed_dim, bias=bias)
        self.out_proj = nn.Linear(embed_dim, embed_dim, bias=bias)

    def forward(self, query, key, value,
                key_padding_mask: Optional[Tensor] = None,
                attn_bias: Optional[Tensor] = None) -> Tensor:
        bsz, tgt_len, embed_dim = query.shape
        assert embed_dim == self.embed_dim, "input width mismatch"

        q = _to_heads(self.q_proj(query), bsz, self.num_heads, self.head_dim,
                      self.scaling)
        k = _to_heads(self.k_proj(key), bsz, self.num_heads, self.head_dim)
        v = _to_heads(self.v_proj(value), bsz, self.num_heads, self.head_dim)

        src_len = k.shape[1]
        scores = qk_scores(q, k)
        assert scores.shape == (bsz * self.num_heads, tgt_len, src_len)
        mask = _padding_to_additive(key_padding_mask, bsz, src_len,
                                    scores.dtype)

        attn = softmax_dropout(
            scores.view(bsz, self.num_heads, tgt_len, src_len),
            self.dropout, self.training, mask=mask, bias=attn_bias,
        ).view(bsz * self.num_heads, tgt_len, src_len)

        heads_out = torch.bmm(attn, v)
        assert heads_out.shape == (bsz * self.num_heads, tgt_len, self.head_dim)
        return self.out_proj(
            _from_heads(heads_out, bsz, self.num_heads, tgt_len, self.head_dim)
        )
