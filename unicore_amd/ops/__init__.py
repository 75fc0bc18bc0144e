"""Python surface over the gfx950 HIP kernel extension.

The extension (``unicore_amd._kernels``) is built in-tree by setup.py with
``PYTORCH_ROCM_ARCH=gfx950`` and provides the fused hot ops:
softmax(+bias,+mask)+dropout fwd/bwd, LayerNorm/RMSNorm fwd/bwd, fused AdamW,
multi-tensor L2 norm, and fp32->bf16 stochastic rounding
(CDNA4-native equivalents of the reference's 8 CUDA extensions,
reference csrc/ + setup.py:141-387).

Policy: on a CUDA/ROCm device the HIP path is mandatory — if the extension
failed to import, any op called with GPU tensors raises loudly (no silent
eager fallback on GPU). On CPU the modules use their eager reference path,
which also serves as the numerics oracle in tests.
"""

import logging
import os

logger = logging.getLogger(__name__)

_kernels = None
_import_error = None

try:
    from unicore_amd import _kernels  # type: ignore[attr-defined]
except ImportError as e:  # extension not built
    _kernels = None
    _import_error = e


def has_kernels() -> bool:
    return _kernels is not None


def require_kernels():
    """Raise if the HIP extension is unavailable.

    Called from every op entry point when tensors live on the GPU: running
    eager fallbacks silently on an MI355X would defeat the point of the
    framework, so we fail loudly instead.
    """
    if _kernels is None:
        raise RuntimeError(
            "unicore_amd._kernels HIP extension is not available "
            f"(import error: {_import_error}). Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            "Set UNICORE_AMD_ALLOW_EAGER=1 only for debugging."
        )


def allow_eager_on_gpu() -> bool:
    return os.environ.get("UNICORE_AMD_ALLOW_EAGER", "0") == "1"


def gpu_kernels_available() -> bool:
    """True if ops on CUDA tensors should take the HIP kernel path."""
    return _kernels is not None


# ---------------------------------------------------------------------------
# Thin wrappers (keep all tensor-shape munging in the module-level shims)
# ---------------------------------------------------------------------------


def softmax_dropout_fwd(
    is_training, inputs, mask, mask_outer_div, bias, bias_outer_div, dropout_prob
):
    """inputs: (n_batch, q, k) contiguous; mask (mb, mq, k) / bias (bb, bq, k)
    addressed as ((b / outer_div) % nb, qi % sq, :).
    Returns (out, dropout_mask, softmax_out)."""
    require_kernels()
    return _kernels.softmax_dropout_forward(
        is_training,
        inputs,
        mask,
        int(mask_outer_div),
        bias,
        int(bias_outer_div),
        float(dropout_prob),
    )


def softmax_dropout_bwd(grad_output, softmax_results, dropout_mask, dropout_prob):
    require_kernels()
    return _kernels.softmax_dropout_backward(
        grad_output, softmax_results, dropout_mask, float(dropout_prob)
    )


def layernorm_fwd(input, gamma, beta, eps):
    require_kernels()
    return _kernels.layernorm_forward(input, gamma, beta, float(eps))


def layernorm_bwd(grad_out, input, mean, invvar, gamma):
    require_kernels()
    return _kernels.layernorm_backward(grad_out, input, mean, invvar, gamma)


def rmsnorm_fwd(input, gamma, eps):
    require_kernels()
    return _kernels.rmsnorm_forward(input, gamma, float(eps))


def rmsnorm_bwd(grad_out, input, invvar, gamma):
    require_kernels()
    return _kernels.rmsnorm_backward(grad_out, input, invvar, gamma)


def fused_adam(
    p, m, v, g, lr, beta1, beta2, eps, grad_scale, step, bias_correction, weight_decay
):
    require_kernels()
    _kernels.adam(
        p,
        m,
        v,
        g,
        float(lr),
        float(beta1),
        float(beta2),
        float(eps),
        float(grad_scale),
        int(step),
        bool(bias_correction),
        float(weight_decay),
    )


def fused_l2norm(tensors, chunk_size=2048 * 64):
    require_kernels()
    return _kernels.multi_tensor_l2norm(int(chunk_size), tensors)


def fused_fp32_to_bf16_sr(src_fp32, dst_bf16):
    require_kernels()
    _kernels.fp32_to_bf16_sr(src_fp32, dst_bf16)


def qkv_split_fwd(qkv, num_heads, scale, bias=None):
    require_kernels()
    return _kernels.qkv_split_forward(qkv, bias, int(num_heads), float(scale))


def qkv_split_bwd(dq, dk, dv, bsz, num_heads, scale, bias_grad=False):
    require_kernels()
    return _kernels.qkv_split_backward(
        dq, dk, dv, int(bsz), int(num_heads), float(scale), bool(bias_grad)
    )


def gelu_dropout_fwd(x, p, is_training, bias=None):
    require_kernels()
    return _kernels.gelu_dropout_forward(x, bias, float(p), bool(is_training))


def gelu_dropout_bwd(grad, x, dmask, p, bias=None):
    # bias given: also return the fp32 column-sum bias grad (x is then the
    # bias-free Linear output; the kernel re-adds bias for the gelu grad)
    require_kernels()
    return _kernels.gelu_dropout_backward(grad, x, bias, dmask, float(p))


def flash_attn_fwd(q, k, v, bias, bias_outer_div, mask, mask_outer_div, p, is_training):
    require_kernels()
    return _kernels.flash_attn_forward(
        q, k, v, bias, int(bias_outer_div), mask, int(mask_outer_div),
        float(p), bool(is_training)
    )


def flash_attn_bwd(d_out, q, k, v, o, lse, bias, bias_outer_div, bias_needs_grad,
                   mask, mask_outer_div, p, dropped, seed):
    require_kernels()
    return _kernels.flash_attn_backward(
        d_out, q, k, v, o, lse, bias, int(bias_outer_div), bool(bias_needs_grad),
        mask, int(mask_outer_div), float(p), bool(dropped), int(seed)
    )


def dropout_add_ln_fwd(x, res, bias, gamma, beta, p, is_training, eps):
    require_kernels()
    return _kernels.dropout_add_ln_forward(
        x, res, bias, gamma, beta, float(p), bool(is_training), float(eps)
    )


def dropout_add_fwd(x, res, p, is_training, bias=None):
    require_kernels()
    return _kernels.dropout_add_forward(x, res, bias, float(p), bool(is_training))


def dropout_add_bwd(grad, dmask, p, bias_dim=0):
    # bias_dim > 0: also return the fp32 column-sum bias grad
    require_kernels()
    return _kernels.dropout_add_backward(grad, dmask, float(p), int(bias_dim))


def colsum_supported(C) -> bool:
    import math

    if C <= 0 or C % 8 != 0 or C > 4096:
        return False
    return (C // math.gcd(2048, C)) <= 256


def embedding_bwd(grad, indices, num_embeddings, padding_idx):
    require_kernels()
    return _kernels.embedding_backward(
        grad, indices, int(num_embeddings),
        int(padding_idx) if padding_idx is not None else -1,
    )


def cross_entropy_fwd(logits, target, ignore_index):
    require_kernels()
    return _kernels.cross_entropy_forward(logits, target, int(ignore_index))


def cross_entropy_bwd(logits, target, lse, grad_scale, ignore_index):
    require_kernels()
    return _kernels.cross_entropy_backward(
        logits, target, lse, grad_scale, int(ignore_index)
    )


def gaussian_basis_fwd(coords, means, stds, out_dtype):
    """(B, L, 3) fp32 coords -> (B, L, L, K) gaussian basis in out_dtype."""
    require_kernels()
    return _kernels.gaussian_basis_forward(coords, means, stds, out_dtype)


def gaussian_basis_bwd(dg, coords, means, stds):
    require_kernels()
    return _kernels.gaussian_basis_backward(dg, coords, means, stds)


def gaussian_basis_supported(n_kernels) -> bool:
    return _kernels is not None and _kernels.gaussian_basis_supported(
        int(n_kernels)
    )


def gaussian_pair_bias_fwd(coords, means, stds, weight, bias, pad, fill, out_dtype):
    """coords (B,L,3) fp32 -> (B,H,L,L) attention bias, Linear+permute+mask fused."""
    require_kernels()
    return _kernels.gaussian_pair_bias_forward(
        coords, means, stds, weight, bias, pad, float(fill), out_dtype
    )


def gaussian_pair_bias_bwd(dbias, coords, means, stds, weight, pad):
    require_kernels()
    return _kernels.gaussian_pair_bias_backward(
        dbias, coords, means, stds, weight, pad
    )


def gaussian_pair_bias_supported(n_kernels, n_heads) -> bool:
    return _kernels is not None and _kernels.gaussian_pair_bias_supported(
        int(n_kernels), int(n_heads)
    )


def softmax_dropout_bwd_bias(grad, softmax_out, dmask, p, bb, bq, od):
    """In-place softmax backward that also returns the broadcast bias grad
    (fp32, (bb*bq, k)) without re-reading the grad tensor."""
    require_kernels()
    return _kernels.softmax_dropout_backward_bias(
        grad, softmax_out, dmask, float(p), int(bb), int(bq), int(od)
    )


def softmax_dropout_bwd_bias_supported(n_batch, q, k, bb, bq, od) -> bool:
    return _kernels is not None and _kernels.softmax_dropout_backward_bias_supported(
        int(n_batch), int(q), int(k), int(bb), int(bq), int(od)
    )


def attn_merge(x, bsz, num_heads, inverse=False):
    # (B*H, L, D) -> (B, L, H*D); inverse maps back. 16B on both sides.
    require_kernels()
    return _kernels.attn_merge(x, int(bsz), int(num_heads), bool(inverse))


def gated_mul_fwd(x, g, bias_x=None, bias_g=None):
    # out = (x + bias_x) * sigmoid(g + bias_g)
    require_kernels()
    return _kernels.gated_mul_forward(x, g, bias_x, bias_g)


def gated_mul_bwd(grad, x, g, bias_x=None, bias_g=None):
    require_kernels()
    return _kernels.gated_mul_backward(grad, x, g, bias_x, bias_g)


def msa_arrange(x, B, S, L, heads, col, inverse=False):
    # (B,S,L,E) <-> head-major MSA layouts; see modules/msa_arrange.py
    require_kernels()
    return _kernels.msa_arrange(
        x, int(B), int(S), int(L), int(heads), bool(col), bool(inverse)
    )
