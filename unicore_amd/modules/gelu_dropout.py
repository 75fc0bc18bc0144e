"""Fused exact-GELU + dropout (the FFN inner activation hot path).

One HIP pass each way instead of torch's gelu->dropout kernel pair; exact
erf GELU matching F.gelu's default.  With ``bias`` the preceding Linear
runs bias-free: the kernel adds the bias ahead of the GELU and its
backward emits the bias gradient as a deterministic column sum (the
eager path re-read the full activation-sized grad for ``grad.sum(0)``).
Eager fallback on CPU / when the extension is absent.
"""

import torch
import torch.nn.functional as F


class _GeluDropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias, p, is_training):
        from unicore_amd import ops

        x = x.contiguous()
        out, dmask = ops.gelu_dropout_fwd(x, p, is_training, bias)
        ctx.p = p
        ctx.save_for_backward(x, dmask, bias)
        return out

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        x, dmask, bias = ctx.saved_tensors
        dx, db = ops.gelu_dropout_bwd(grad.contiguous(), x, dmask, ctx.p,
                                      bias=bias)
        dbias = db.to(bias.dtype) if bias is not None else None
        return dx, dbias, None, None


class _GeluDropoutNoBias(torch.autograd.Function):
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
        dx, _ = ops.gelu_dropout_bwd(grad.contiguous(), x, dmask, ctx.p)
        return dx, None, None


def gelu_dropout(x, p, is_training, bias=None):
    """dropout(gelu(x + bias), p) fused on GPU; eager elsewhere."""
    if x.is_cuda and x.numel() % 8 == 0:
        from unicore_amd import ops

        bias_ok = bias is None or (
            ops.colsum_supported(bias.numel()) and x.shape[-1] == bias.numel()
        )
        if bias_ok and (ops.gpu_kernels_available() or not ops.allow_eager_on_gpu()):
            if bias is None:
                return _GeluDropoutNoBias.apply(x, p, is_training)
            return _GeluDropout.apply(x, bias, p, is_training)
    if bias is not None:
        x = x + bias
    x = F.gelu(x)
    if is_training and p > 0:
        x = F.dropout(x, p=p)
    return x
