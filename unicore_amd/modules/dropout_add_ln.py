"""Fused dropout(x [+ bias]) + residual add + LayerNorm (post-LN join).

One kernel computes ``LN(residual + dropout(x + bias))``, writing the
pre-norm sum once (it is both the backward input and, in post-LN stacks,
never needed again) — versus the dropout_add + LN pair, this removes a
full extra read of the hidden tensor and a kernel launch per site.

Backward chains the two existing kernels: layernorm_backward over the
saved sum yields d_sum (+ dgamma/dbeta); dropout_add_backward maps d_sum
through the keep-mask and emits the folded-bias column sum; the residual
gradient IS d_sum. Eager fallback composes F.dropout + add + F.layer_norm.
"""

import os

import torch
import torch.nn.functional as F


class _DropoutAddLN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, bias, gamma, beta, p, is_training, eps):
        from unicore_amd import ops

        normed, summed, dmask, mean, invvar = ops.dropout_add_ln_fwd(
            x.contiguous(), res.contiguous(), bias, gamma.contiguous(),
            beta.contiguous(), p, is_training, eps,
        )
        ctx.save_for_backward(summed, dmask, mean, invvar, gamma)
        ctx.p = p
        ctx.bias_dim = bias.numel() if bias is not None else 0
        ctx.bias_dtype = bias.dtype if bias is not None else None
        return normed

    @staticmethod
    def backward(ctx, d_norm):
        from unicore_amd import ops

        summed, dmask, mean, invvar, gamma = ctx.saved_tensors
        ds, dgamma, dbeta = ops.layernorm_bwd(
            d_norm.contiguous(), summed, mean, invvar, gamma
        )
        dbias = None
        if dmask.numel() == 0 and ctx.bias_dim == 0:
            dx = ds
        else:
            dx, db = ops.dropout_add_bwd(ds, dmask, ctx.p, ctx.bias_dim)
            if ctx.bias_dim:
                dbias = db.to(ctx.bias_dtype)
        return dx, ds, dbias, dgamma, dbeta, None, None, None


class _DropoutAddLNPre(torch.autograd.Function):
    """Pre-LN chain variant: returns BOTH the summed residual stream and
    the normed view (the next sub-module's input). Backward receives
    gradients for both; the summed grad joins the LN input grad before the
    dropout/bias unmap."""

    @staticmethod
    def forward(ctx, x, res, bias, gamma, beta, p, is_training, eps):
        from unicore_amd import ops

        normed, summed, dmask, mean, invvar = ops.dropout_add_ln_fwd(
            x.contiguous(), res.contiguous(), bias, gamma.contiguous(),
            beta.contiguous(), p, is_training, eps,
        )
        ctx.save_for_backward(summed, dmask, mean, invvar, gamma)
        ctx.p = p
        ctx.bias_dim = bias.numel() if bias is not None else 0
        ctx.bias_dtype = bias.dtype if bias is not None else None
        return summed, normed

    @staticmethod
    def backward(ctx, d_sum, d_norm):
        from unicore_amd import ops

        summed, dmask, mean, invvar, gamma = ctx.saved_tensors
        if d_norm is None:
            # normed output unused downstream: only the stream grad flows
            ds = torch.zeros_like(summed)
            dgamma = torch.zeros_like(gamma)
            dbeta = torch.zeros_like(gamma)
        else:
            ds, dgamma, dbeta = ops.layernorm_bwd(
                d_norm.contiguous(), summed, mean, invvar, gamma
            )
        if d_sum is not None:
            ds = ds + d_sum
        dbias = None
        if dmask.numel() == 0 and ctx.bias_dim == 0:
            dx = ds
        else:
            dx, db = ops.dropout_add_bwd(ds, dmask, ctx.p, ctx.bias_dim)
            if ctx.bias_dim:
                dbias = db.to(ctx.bias_dtype)
        return dx, ds, dbias, dgamma, dbeta, None, None, None


def _fuse_ok(x, residual, bias):
    if not (
        x.is_cuda
        and x.shape == residual.shape
        and x.shape[-1] % 8 == 0
        and x.shape[-1] <= 2048
        and os.environ.get("UNICORE_FUSED_LN_JOIN", "1") == "1"
    ):
        return False
    from unicore_amd import ops

    bias_ok = bias is None or (
        ops.colsum_supported(bias.numel()) and x.shape[-1] == bias.numel()
    )
    return bias_ok and (ops.gpu_kernels_available()
                        or not ops.allow_eager_on_gpu())


def dropout_add_ln_pre(x, residual, ln, p, is_training, bias=None):
    """Pre-LN residual chain step: ``s = residual + dropout(x + bias)``;
    returns ``(s, ln(s))`` — *s* continues the residual stream, ``ln(s)``
    feeds the next sub-module. Fused into one kernel on GPU."""
    if _fuse_ok(x, residual, bias):
        return _DropoutAddLNPre.apply(
            x, residual, bias, ln.weight, ln.bias, p, is_training, ln.eps
        )
    if bias is not None:
        x = x + bias
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    s = residual + x
    return s, ln(s)


def dropout_add_ln(x, residual, ln, p, is_training, bias=None):
    """``ln(residual + dropout(x + bias, p))`` — fused on GPU.

    *ln* is the LayerNorm module whose weight/bias/eps apply.
    """
    if _fuse_ok(x, residual, bias):
        return _DropoutAddLN.apply(
            x, residual, bias, ln.weight, ln.bias, p, is_training, ln.eps
        )
    if bias is not None:
        x = x + bias
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    return ln(residual + x)
