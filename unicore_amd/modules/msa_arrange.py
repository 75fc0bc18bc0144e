"""Evoformer MSA layout moves: (B, S, L, E) <-> head-major batched layouts.

row mode: (B*H*S, L, D) — attention over residues for each MSA row
col mode: (B*L*H, S, D) — attention over rows for each residue column

torch runs these 5-D permutes through its strided copy (2-byte scalar
accesses on one side; ~27k small launches per Evoformer step at the
stress config); the kernel keeps 16 B vectors on both sides and accepts
``chunk()`` views of the fused qkv projection directly (no pre-copy).
"""

import torch


class _MsaArrange(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x4, heads, col):
        from unicore_amd import ops

        B, S, L, C = x4.shape
        ctx.dims = (B, S, L, heads, col)
        return ops.msa_arrange(x4, B, S, L, heads, col)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        B, S, L, H, col = ctx.dims
        g = ops.msa_arrange(grad.contiguous(), B, S, L, H, col, inverse=True)
        return g, None, None


class _MsaMerge(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, B, S, L, heads, col):
        from unicore_amd import ops

        ctx.dims = (B, S, L, heads, col)
        ctx.in_shape = x.shape
        return ops.msa_arrange(x.contiguous(), B, S, L, heads, col,
                               inverse=True)

    @staticmethod
    def backward(ctx, grad):
        from unicore_amd import ops

        B, S, L, H, col = ctx.dims
        g = ops.msa_arrange(grad.contiguous(), B, S, L, H, col)
        return g.reshape(ctx.in_shape), None, None, None, None, None


def _fused_ok(x, heads):
    if not x.is_cuda:
        return False
    from unicore_amd import ops

    D = x.shape[-1] // heads
    return D % 8 == 0 and (ops.gpu_kernels_available()
                           or not ops.allow_eager_on_gpu())


def msa_arrange(x4, heads, col):
    """(B, S, L, E) [any row stride, contiguous E] -> head-major layout."""
    B, S, L, C = x4.shape
    D = C // heads
    if _fused_ok(x4, heads) and x4.stride(-1) == 1 and x4.stride(2) % 8 == 0:
        out = _MsaArrange.apply(x4, heads, col)
        if col:
            return out  # (B*L*H, S, D)
        return out.view(B * heads * S, L, D)
    if col:
        return (x4.reshape(B, S, L, heads, D).permute(0, 2, 3, 1, 4)
                .reshape(B * L * heads, S, D))
    return (x4.reshape(B, S, L, heads, D).permute(0, 3, 1, 2, 4)
            .reshape(B * heads * S, L, D))


def msa_merge(x, B, S, L, heads, col):
    """head-major layout -> (B, S, L, E)."""
    D = x.shape[-1]
    if _fused_ok(x, 1):
        return _MsaMerge.apply(x, B, S, L, heads, col)
    if col:
        return (x.view(B, L, heads, S, D).permute(0, 3, 1, 2, 4)
                .reshape(B, S, L, heads * D))
    return (x.view(B, heads, S, L, D).permute(0, 2, 3, 1, 4)
            .reshape(B, S, L, heads * D))
