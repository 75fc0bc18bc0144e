"""Fused gated multiply: out = (x + bias_x) * sigmoid(g + bias_g).

The Evoformer/AlphaFold gating idiom.  Both projection biases ride the
kernel (their Linears run bias-free) and backward returns the bias grads
as deterministic column sums — replacing torch's sigmoid + mul kernel
pair and two activation-sized bias-grad reductions.  Eager fallback on
CPU / when the extension is absent.
"""

import torch


class _GatedMul(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, g, bias_x, bias_g):
        from unicore_amd import ops

        x = x.contiguous()
        g = g.contiguous()
        ctx.save_for_backward(x, g, bias_x, bias_g)
        return ops.gated_mul_fwd(x, g, bias_x, bias_g)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        x, g, bias_x, bias_g = ctx.saved_tensors
        dx, dg, db = ops.gated_mul_bwd(grad, x, g, bias_x, bias_g)
        dbx = dbg = None
        if bias_x is not None:
            C = bias_x.numel()
            dbx = db[:C].to(bias_x.dtype)
            dbg = db[C:].to(bias_g.dtype)
        return dx, dg, dbx, dbg


def gated_mul(x, g, bias_x=None, bias_g=None):
    """(x + bias_x) * sigmoid(g + bias_g); biases optional (must be both
    present or both absent for the fused path)."""
    if x.is_cuda and x.numel() % 8 == 0 and x.shape == g.shape:
        from unicore_amd import ops

        bias_ok = (bias_x is None) == (bias_g is None) and (
            bias_x is None
            or (ops.colsum_supported(bias_x.numel()) and x.shape[-1] == bias_x.numel())
        )
        if bias_ok and (ops.gpu_kernels_available() or not ops.allow_eager_on_gpu()):
            return _GatedMul.apply(x, g, bias_x, bias_g)
    if bias_x is not None:
        x = x + bias_x
    if bias_g is not None:
        g = g + bias_g
    return x * torch.sigmoid(g)
