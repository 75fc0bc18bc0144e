"""Fused LayerNorm.

Functional parity with the reference's fused layernorm
(reference unicore/modules/layer_norm.py:22-71 + csrc/layernorm/*): forward
saves per-row fp32 mean/invvar; backward computes grad_input in one wave64
row kernel and (grad_gamma, grad_beta) via a two-stage LDS column reduction.

Unlike the reference (which whitelists 16 hidden sizes,
csrc/layernorm/layernorm.cu:188-245), our CDNA4 kernel handles any hidden
dim; rows are register-resident up to 8192 elems and LDS-staged beyond.
The eager F.layer_norm path is the CPU fallback and numerics oracle.
"""

import numbers

import torch
import torch.nn.functional as F
from torch.nn.parameter import Parameter


class FusedLayerNormFastFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, bias, eps):
        from unicore_amd import ops

        input = input.contiguous()
        weight = weight.contiguous()
        bias = bias.contiguous()
        output, mean, invvar = ops.layernorm_fwd(input, weight, bias, eps)
        ctx.save_for_backward(input, weight, bias, mean, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        from unicore_amd import ops

        input, weight, bias, mean, invvar = ctx.saved_tensors
        grad_output = grad_output.contiguous()
        grad_input, grad_weight, grad_bias = ops.layernorm_bwd(
            grad_output, input, mean, invvar, weight
        )
        return grad_input, grad_weight, grad_bias, None


class LayerNorm(torch.nn.Module):
    def __init__(self, normalized_shape, eps=1e-5, elementwise_affine=True):
        super().__init__()
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = torch.Size(normalized_shape)
        self.eps = eps
        assert elementwise_affine
        self.weight = Parameter(torch.empty(*normalized_shape))
        self.bias = Parameter(torch.empty(*normalized_shape))
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)
        torch.nn.init.zeros_(self.bias)

    def forward(self, input):
        if input.is_cuda:
            from unicore_amd import ops

            if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
                return FusedLayerNormFastFunction.apply(
                    input, self.weight, self.bias, self.eps
                )
        return F.layer_norm(
            input, self.normalized_shape, self.weight, self.bias, self.eps
        )

    def extra_repr(self):
        return "{normalized_shape}, eps={eps}, elementwise_affine=True".format(
            **self.__dict__
        )
