"""Fused gaussian pair-basis: coords (B, L, 3) -> (B, L, L, K) features.

g[b,i,j,k] = exp(-0.5 * ((|c_i - c_j| - mean_k) / (|std_k| + 1e-3))^2)

On GPU this runs one HIP kernel each way (csrc/gaussian.hip) instead of
torch.cdist + five (B, L, L, K) elementwise passes — the dominant
non-GEMM cost of the mol_pairbias model.  Distances are recomputed from
the tiny (B, L, 3) coords in backward rather than saved.  Eager fallback
(also the numerics oracle) computes the same fp32 math then casts once.
"""

import os

import torch


class _GaussianBasis(torch.autograd.Function):
    @staticmethod
    def forward(ctx, coords, means, stds, out_dtype):
        from unicore_amd import ops

        coords = coords.contiguous()
        ctx.save_for_backward(coords, means, stds)
        return ops.gaussian_basis_fwd(coords, means, stds, out_dtype)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        coords, means, stds = ctx.saved_tensors
        d_coords, d_means, d_stds = ops.gaussian_basis_bwd(
            grad.contiguous(), coords, means, stds
        )
        return d_coords, d_means, d_stds, None


def _eager_gaussian_basis(coords, means, stds, out_dtype):
    # exact pairwise distances (cdist's matmul path loses ~1e-3 at fp32);
    # the +eye/-eye shift keeps the diagonal at zero with a zero gradient,
    # matching both cdist's subgradient and the HIP kernel
    diff = coords.unsqueeze(2) - coords.unsqueeze(1)  # (B, L, L, 3)
    ssq = diff.pow(2).sum(-1)
    eye = torch.eye(coords.size(1), device=coords.device, dtype=ssq.dtype)
    dist = (ssq + eye).sqrt() - eye
    x = dist.unsqueeze(-1) - means.view(1, 1, 1, -1)
    inv = 1.0 / (stds.abs() + 1e-3)
    return torch.exp(-0.5 * (x * inv.view(1, 1, 1, -1)) ** 2).to(out_dtype)


def _shape_ok(n_kernels):
    tpp = n_kernels // 8
    return (
        n_kernels >= 8
        and n_kernels % 8 == 0
        and (tpp & (tpp - 1)) == 0
        and tpp <= 64
    )


def gaussian_basis(coords, means, stds, out_dtype=None):
    """coords (B, L, 3) fp32; means/stds (K,) fp32 parameters."""
    if out_dtype is None:
        out_dtype = coords.dtype
    if os.environ.get("UNICORE_GAUSSIAN_EAGER", "0") == "1":  # A/B benchmarking
        return _eager_gaussian_basis(coords, means, stds, out_dtype)
    if coords.is_cuda and _shape_ok(means.numel()):
        from unicore_amd import ops

        if ops.gpu_kernels_available() or not ops.allow_eager_on_gpu():
            return _GaussianBasis.apply(coords, means, stds, out_dtype)
    return _eager_gaussian_basis(coords, means, stds, out_dtype)


class _GaussianPairBias(torch.autograd.Function):
    """Fully-fused pair bias: coords -> (B, H, L, L), with the K->H Linear,
    the permute to head-major and the padding-key masked_fill folded in."""

    @staticmethod
    def forward(ctx, coords, means, stds, weight, bias, pad, fill, out_dtype):
        from unicore_amd import ops

        coords = coords.contiguous()
        ctx.save_for_backward(coords, means, stds, weight)
        ctx.pad = pad
        out = ops.gaussian_pair_bias_fwd(
            coords, means, stds, weight, bias, pad, fill, out_dtype
        )
        return out

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        coords, means, stds, weight = ctx.saved_tensors
        d_coords, d_means, d_stds, d_w, d_b = ops.gaussian_pair_bias_bwd(
            grad.contiguous(), coords, means, stds, weight, ctx.pad
        )
        return (
            d_coords,
            d_means.to(means.dtype),
            d_stds.to(stds.dtype),
            d_w.to(weight.dtype),
            d_b.to(weight.dtype),
            None,
            None,
            None,
        )


def gaussian_pair_bias_fused_ok(n_kernels, n_heads):
    if os.environ.get("UNICORE_GAUSSIAN_EAGER", "0") == "1":  # A/B benchmarking
        return False
    from unicore_amd import ops

    return ops.has_kernels() and ops.gaussian_pair_bias_supported(
        n_kernels, n_heads
    )
