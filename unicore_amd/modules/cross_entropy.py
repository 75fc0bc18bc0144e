"""Fused token cross entropy: sum-reduced NLL over (N, V) logits without
materializing the fp32 log-softmax (online logsumexp forward, softmax-minus-
onehot backward regenerated from the saved per-row LSE)."""

import torch
import torch.nn.functional as F


class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, ignore_index):
        from unicore_amd import ops

        logits = logits.contiguous()
        target = target.contiguous()
        loss, lse = ops.cross_entropy_fwd(logits, target, ignore_index)
        ctx.save_for_backward(logits, target, lse)
        ctx.ignore_index = ignore_index
        return loss.sum()

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        logits, target, lse = ctx.saved_tensors
        gs = grad.detach().to(torch.float32).reshape(1).contiguous()
        dx = ops.cross_entropy_bwd(logits, target, lse, gs, ctx.ignore_index)
        return dx, None, None


def fused_nll_loss(logits, target, ignore_index=-100):
    """sum(-log_softmax(logits)[i, target_i]) over rows with
    target != ignore_index; fp32 math; equals
    F.nll_loss(F.log_softmax(logits, -1, dtype=float32), target,
    ignore_index=..., reduction="sum")."""
    if logits.is_cuda and logits.dim() == 2:
        from unicore_amd import ops

        if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
            return _FusedCE.apply(logits, target, ignore_index)
    return F.nll_loss(
        F.log_softmax(logits, dim=-1, dtype=torch.float32),
        target,
        ignore_index=ignore_index,
        reduction="sum",
    )
