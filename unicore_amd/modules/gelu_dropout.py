"""Fused exact-GELU + dropout (the FFN inner activation hot path).

One HIP pass each way instead of torch's gelu->dropout kernel pair; exact
erf GELU matching F.gelu's default. Eager fallback on CPU / when the
extension is absent.
"""

import torch
import torch.nn.functional as F


class _GeluDropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, is_training):
        from unicore_amd import ops

        x = x.contiguous()
        out, dmask = ops.gelu_dropout_fwd(x, p, is_training)
        ctx.p = p
        ctx.save_for_backward(x, dmask)
        return out

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        x, dmask = ctx.saved_tensors
        dx = ops.gelu_dropout_bwd(grad.contiguous(), x, dmask, ctx.p)
        return dx, None, None


def gelu_dropout(x, p, is_training):
    """dropout(gelu(x), p) fused on GPU; eager elsewhere."""
    if x.is_cuda and x.numel() % 8 == 0:
        from unicore_amd import ops

        if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
            return _GeluDropout.apply(x, p, is_training)
    x = F.gelu(x)
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    return x
