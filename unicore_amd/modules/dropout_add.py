"""Fused dropout(x) + residual add: out = residual + dropout(x, p).

One HIP pass instead of torch's dropout + add pair; used at the two
residual joins of every transformer layer. Eager fallback on CPU.
"""

import torch
import torch.nn.functional as F


class _DropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, p, is_training):
        from unicore_amd import ops

        out, dmask = ops.dropout_add_fwd(x.contiguous(), res.contiguous(), p,
                                         is_training)
        ctx.p = p
        ctx.save_for_backward(dmask)
        return out

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        (dmask,) = ctx.saved_tensors
        grad = grad.contiguous()
        if dmask.numel() == 0:
            return grad, grad, None, None
        dx = ops.dropout_add_bwd(grad, dmask, ctx.p)
        return dx, grad, None, None


def dropout_add(x, residual, p, is_training):
    """residual + dropout(x, p), fused on GPU."""
    if x.is_cuda and x.numel() % 8 == 0 and x.shape == residual.shape:
        from unicore_amd import ops

        if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
            return _DropoutAdd.apply(x, residual, p, is_training)
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    return residual + x
