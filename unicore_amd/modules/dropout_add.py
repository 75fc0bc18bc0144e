"""Fused dropout(x [+ bias]) + residual add: out = residual + dropout(x + b).

One HIP pass instead of torch's dropout + add pair; used at the two
residual joins of every transformer layer.  With ``bias`` the preceding
Linear runs bias-free and this op both adds the bias (free — the tensor
is already in registers) and produces its gradient as a deterministic
column sum in backward, replacing the eager per-Linear ``grad.sum(0)``
re-read of the full activation-sized grad tensor.  Eager fallback on CPU.
"""

import torch
import torch.nn.functional as F


class _DropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, bias, p, is_training):
        from unicore_amd import ops

        out, dmask = ops.dropout_add_fwd(
            x.contiguous(), res.contiguous(), p, is_training, bias
        )
        ctx.p = p
        ctx.bias_dim = bias.numel() if bias is not None else 0
        ctx.bias_dtype = bias.dtype if bias is not None else None
        ctx.save_for_backward(dmask)
        return out

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        (dmask,) = ctx.saved_tensors
        grad = grad.contiguous()
        dbias = None
        if dmask.numel() == 0 and ctx.bias_dim == 0:
            return grad, grad, None, None, None
        dx, db = ops.dropout_add_bwd(grad, dmask, ctx.p, ctx.bias_dim)
        if ctx.bias_dim:
            dbias = db.to(ctx.bias_dtype)
        return dx, grad, dbias, None, None


def dropout_add(x, residual, p, is_training, bias=None):
    """residual + dropout(x + bias, p), fused on GPU."""
    if x.is_cuda and x.numel() % 8 == 0 and x.shape == residual.shape:
        from unicore_amd import ops

        bias_ok = bias is None or (
            ops.colsum_supported(bias.numel()) and x.shape[-1] == bias.numel()
        )
        if bias_ok and (ops.gpu_kernels_available() or not ops.allow_eager_on_gpu()):
            return _DropoutAdd.apply(x, residual, bias, p, is_training)
    if bias is not None:
        x = x + bias
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    return residual + x
