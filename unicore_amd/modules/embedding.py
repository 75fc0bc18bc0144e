"""Embedding with a fast atomic-scatter backward.

torch's sort-based embedding backward is ~6x slower than an fp32 atomic
scatter at BERT-scale token counts.  Atomic accumulation order is
non-deterministic, so the module falls back to F.embedding when
torch.use_deterministic_algorithms(True) is set (or on CPU / without the
extension).
"""

import torch
import torch.nn.functional as F
from torch import nn


class _EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight, indices, padding_idx):
        ctx.save_for_backward(indices)
        ctx.num_embeddings = weight.shape[0]
        ctx.padding_idx = padding_idx
        return F.embedding(indices, weight, padding_idx)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        (indices,) = ctx.saved_tensors
        gw = ops.embedding_bwd(
            grad.contiguous(), indices.contiguous().view(-1),
            ctx.num_embeddings,
            ctx.padding_idx if ctx.padding_idx is not None else -1,
        )
        return gw, None, None


class Embedding(nn.Embedding):
    def forward(self, input):
        use_fast = (
            self.weight.is_cuda
            and not torch.are_deterministic_algorithms_enabled()
            and self.max_norm is None
            and not self.sparse
            and self.scale_grad_by_freq is False
        )
        if use_fast:
            from unicore_amd import ops

            use_fast = ops.gpu_kernels_available()
        if use_fast:
            return _EmbeddingFn.apply(self.weight, input, self.padding_idx)
        return super().forward(input)
