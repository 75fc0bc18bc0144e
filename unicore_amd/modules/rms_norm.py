"""Fused RMSNorm (same structure as LayerNorm minus mean/beta).

Functional parity with reference unicore/modules/rms_norm.py:24-99 +
csrc/rmsnorm/*; any hidden dim supported (no 16-size whitelist).
"""

import numbers

import torch
from torch.nn.parameter import Parameter


def _eager_rms_norm(x, normalized_shape, weight, eps):
    dtype = x.dtype
    xf = x.float()
    variance = xf.pow(2).mean(-1, keepdim=True)
    xf = xf * torch.rsqrt(variance + eps)
    return (weight.float() * xf).to(dtype)


class FusedRMSNormFastFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, eps):
        from unicore_amd import ops

        input = input.contiguous()
        weight = weight.contiguous()
        output, invvar = ops.rmsnorm_fwd(input, weight, eps)
        ctx.save_for_backward(input, weight, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        from unicore_amd import ops

        input, weight, invvar = ctx.saved_tensors
        grad_output = grad_output.contiguous()
        grad_input, grad_weight = ops.rmsnorm_bwd(grad_output, input, invvar, weight)
        return grad_input, grad_weight, None


class RMSNorm(torch.nn.Module):
    def __init__(self, normalized_shape, eps=1e-6, elementwise_affine=True):
        super().__init__()
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = torch.Size(normalized_shape)
        self.eps = eps
        assert elementwise_affine
        self.weight = Parameter(torch.empty(*normalized_shape))
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)

    def forward(self, input):
        if input.is_cuda:
            from unicore_amd import ops

            if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
                return FusedRMSNormFastFunction.apply(input, self.weight, self.eps)
        return _eager_rms_norm(input, self.normalized_shape, self.weight, self.eps)

    def extra_repr(self):
        return "{normalized_shape}, eps={eps}, elementwise_affine=True".format(
            **self.__dict__
        )
