// Fused QKV head-split for attention on gfx950.
//
// Replaces the reference's chunk(3) + 3x(transpose+contiguous) + q*scale
// chain (reference unicore/modules/multihead_attention.py:46-70) — and its
// backward cat — with one permute-copy kernel each way:
//   forward:  qkv (B, L, 3, H, D) -> q,k,v each (B*H, L, D), q pre-scaled
//   backward: dq,dk,dv (B*H, L, D) -> dqkv (B, L, 3, H, D), dq scaled
// Same HBM traffic as one pass (16 B/lane vectors); kills the separate
// q-scale sweep, the torch cat kernel and 5 extra launches per layer.
#include "common.h"
#include "fold.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <optional>
#include <vector>

namespace {

template <typename T, bool HAS_BIAS>
__global__ void qkv_split_fwd_kernel(const T* __restrict__ qkv, T* __restrict__ q,
                                     T* __restrict__ k, T* __restrict__ v,
                                     const T* __restrict__ bias,
                                     int64_t n8, int L, int H, int D8,
                                     float scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int C = 3 * H * D8 * 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    int64_t tmp = i;
    const int d8 = (int)(tmp % D8);
    tmp /= D8;
    const int h = (int)(tmp % H);
    tmp /= H;
    const int t = (int)(tmp % 3);
    tmp /= 3;
    const int l = (int)(tmp % L);
    const int64_t b = tmp / L;

    float f[8];
    load8(qkv + i * 8, f);
    if constexpr (HAS_BIAS) {
      float fb[8];
      load8(bias + (int)((i * 8) % C), fb);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += fb[j];
    }
    T* out = t == 0 ? q : (t == 1 ? k : v);
    if (t == 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] *= scale;
    }
    const int64_t o = (((b * H + h) * (int64_t)L + l) * D8 + d8) * 8;
    store8(out + o, f);
  }
}

template <typename T, bool BGRAD>
__global__ void qkv_split_bwd_kernel(T* __restrict__ dqkv, const T* __restrict__ dq,
                                     const T* __restrict__ dk,
                                     const T* __restrict__ dv,
                                     float* __restrict__ partials, int64_t n8,
                                     int L, int H, int D8, float scale) {
  extern __shared__ float s_col[];
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int C = 3 * H * D8 * 8;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  int c0 = -1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    int64_t tmp = i;
    const int d8 = (int)(tmp % D8);
    tmp /= D8;
    const int h = (int)(tmp % H);
    tmp /= H;
    const int t = (int)(tmp % 3);
    tmp /= 3;
    const int l = (int)(tmp % L);
    const int64_t b = tmp / L;

    const T* g = t == 0 ? dq : (t == 1 ? dk : dv);
    const int64_t o = (((b * H + h) * (int64_t)L + l) * D8 + d8) * 8;
    float f[8];
    load8(g + o, f);
    if (t == 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] *= scale;
    }
    if constexpr (BGRAD) {
      if (c0 < 0) c0 = (int)((i * 8) % C);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += f[j];
    }
    store8(dqkv + i * 8, f);
  }
  if constexpr (BGRAD) {
    colsum_block_fold(acc, c0, C, s_col,
                      partials + (int64_t)blockIdx.x * C);
  }
}

#define DISPATCH_FTYPES(st, NAME, ...)                               \
  switch (st) {                                                      \
    case at::ScalarType::Float: {                                    \
      using scalar_t = float;                                        \
      __VA_ARGS__;                                                   \
      break;                                                         \
    }                                                                \
    case at::ScalarType::Half: {                                     \
      using scalar_t = __half;                                       \
      __VA_ARGS__;                                                   \
      break;                                                         \
    }                                                                \
    case at::ScalarType::BFloat16: {                                 \
      using scalar_t = __hip_bfloat16;                               \
      __VA_ARGS__;                                                   \
      break;                                                         \
    }                                                                \
    default:                                                         \
      TORCH_CHECK(false, NAME, ": unsupported dtype ", st);          \
  }

}  // namespace

std::vector<at::Tensor> qkv_split_forward(at::Tensor qkv,
                                          std::optional<at::Tensor> bias,
                                          int64_t num_heads, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous() && qkv.dim() == 3,
              "qkv_split: expected contiguous (B, L, 3E)");
  const int64_t B = qkv.size(0);
  const int L = (int)qkv.size(1);
  const int64_t E3 = qkv.size(2);
  TORCH_CHECK(E3 % (3 * num_heads) == 0, "qkv_split: bad inner dim");
  const int H = (int)num_heads;
  const int D = (int)(E3 / (3 * num_heads));
  TORCH_CHECK(D % 8 == 0, "qkv_split: head_dim must be a multiple of 8");
  const int D8 = D / 8;
  auto opts = qkv.options();
  auto q = at::empty({B * H, L, D}, opts);
  auto k = at::empty({B * H, L, D}, opts);
  auto v = at::empty({B * H, L, D}, opts);
  const int64_t n8 = qkv.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  const bool has_bias = bias.has_value();
  at::Tensor bc;
  if (has_bias) {
    bc = bias->contiguous();
    TORCH_CHECK(bc.numel() == E3 && bc.scalar_type() == qkv.scalar_type(),
                "qkv_split: bad bias");
  }
  DISPATCH_FTYPES(qkv.scalar_type(), "qkv_split_forward", {
    if (has_bias)
      qkv_split_fwd_kernel<scalar_t, true><<<grid, 256, 0, stream>>>(
          reinterpret_cast<const scalar_t*>(qkv.data_ptr()),
          reinterpret_cast<scalar_t*>(q.data_ptr()),
          reinterpret_cast<scalar_t*>(k.data_ptr()),
          reinterpret_cast<scalar_t*>(v.data_ptr()),
          reinterpret_cast<const scalar_t*>(bc.data_ptr()), n8, L, H, D8,
          (float)scale);
    else
      qkv_split_fwd_kernel<scalar_t, false><<<grid, 256, 0, stream>>>(
          reinterpret_cast<const scalar_t*>(qkv.data_ptr()),
          reinterpret_cast<scalar_t*>(q.data_ptr()),
          reinterpret_cast<scalar_t*>(k.data_ptr()),
          reinterpret_cast<scalar_t*>(v.data_ptr()), nullptr, n8, L, H, D8,
          (float)scale);
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {q, k, v};
}

std::vector<at::Tensor> qkv_split_backward(at::Tensor dq, at::Tensor dk,
                                           at::Tensor dv, int64_t B,
                                           int64_t num_heads, double scale,
                                           bool bias_grad) {
  TORCH_CHECK(dq.is_cuda() && dq.is_contiguous() && dk.is_contiguous() &&
                  dv.is_contiguous(),
              "qkv_split_backward: grads must be contiguous CUDA");
  const int H = (int)num_heads;
  const int L = (int)dq.size(1);
  const int D = (int)dq.size(2);
  TORCH_CHECK(D % 8 == 0, "qkv_split_backward: head_dim % 8");
  const int64_t C = 3LL * H * D;
  const bool bgrad = bias_grad && colsum_supported(C);
  auto dqkv = at::empty({B, L, C}, dq.options());
  auto dbias = at::empty({bgrad ? C : 0}, dq.options().dtype(at::kFloat));
  const int64_t n8 = dqkv.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = bgrad ? colsum_grid(n8, C) : unicore_grid((n8 + 255) / 256);
  at::Tensor partials;
  if (bgrad) partials = at::empty({grid, C}, dbias.options());
  const size_t lds = bgrad ? (size_t)C * sizeof(float) : 0;
  DISPATCH_FTYPES(dq.scalar_type(), "qkv_split_backward", {
    if (bgrad)
      qkv_split_bwd_kernel<scalar_t, true><<<grid, 256, lds, stream>>>(
          reinterpret_cast<scalar_t*>(dqkv.data_ptr()),
          reinterpret_cast<const scalar_t*>(dq.data_ptr()),
          reinterpret_cast<const scalar_t*>(dk.data_ptr()),
          reinterpret_cast<const scalar_t*>(dv.data_ptr()),
          partials.data_ptr<float>(), n8, L, H, D / 8, (float)scale);
    else
      qkv_split_bwd_kernel<scalar_t, false><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(dqkv.data_ptr()),
          reinterpret_cast<const scalar_t*>(dq.data_ptr()),
          reinterpret_cast<const scalar_t*>(dk.data_ptr()),
          reinterpret_cast<const scalar_t*>(dv.data_ptr()), nullptr, n8, L, H,
          D / 8, (float)scale);
  });
  if (bgrad) {
    unicore_fold_columns(partials.data_ptr<float>(), dbias.data_ptr<float>(),
                         grid, (int)C, partials.options(), stream);
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {dqkv, dbias};
}

namespace {

// attention output merge: (B*H, L, D) -> (B, L, H*D) and its inverse.
// torch's copy kernel walks the permuted side with 2-byte scalar
// accesses (~2.9 TB/s measured); decoding from the OUTPUT index keeps
// both sides on 16 B vectors (the inner D stays contiguous either way).
template <typename T, bool INVERSE>
__global__ void attn_merge_kernel(T* __restrict__ out, const T* __restrict__ in,
                                  int64_t n8, int L, int H, int D8) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    int64_t tmp = i;
    const int d8 = (int)(tmp % D8);
    tmp /= D8;
    int h, l;
    if constexpr (!INVERSE) {
      // i indexes the merged output (b, l, h, d8)
      h = (int)(tmp % H);
      tmp /= H;
      l = (int)(tmp % L);
    } else {
      // i indexes the head-major output (b, h, l, d8)
      l = (int)(tmp % L);
      tmp /= L;
      h = (int)(tmp % H);
    }
    const int64_t b = tmp / (INVERSE ? H : L);
    float f[8];
    const int64_t src = INVERSE
                            ? ((b * L + l) * (int64_t)H + h) * D8 + d8
                            : ((b * H + h) * (int64_t)L + l) * D8 + d8;
    load8(in + src * 8, f);
    store8(out + i * 8, f);
  }
}

}  // namespace

at::Tensor attn_merge(at::Tensor x, int64_t B, int64_t num_heads,
                      bool inverse) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "attn_merge: contiguous CUDA");
  const int H = (int)num_heads;
  int L, D;
  at::Tensor out;
  if (!inverse) {
    TORCH_CHECK(x.dim() == 3 && x.size(0) == B * H, "attn_merge: bad shape");
    L = (int)x.size(1);
    D = (int)x.size(2);
    out = at::empty({B, L, (int64_t)H * D}, x.options());
  } else {
    TORCH_CHECK(x.dim() == 3 && x.size(0) == B && x.size(2) % H == 0,
                "attn_merge: bad shape");
    L = (int)x.size(1);
    D = (int)(x.size(2) / H);
    out = at::empty({B * H, L, D}, x.options());
  }
  TORCH_CHECK(D % 8 == 0, "attn_merge: head_dim % 8");
  const int64_t n8 = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  DISPATCH_FTYPES(x.scalar_type(), "attn_merge", {
    if (inverse)
      attn_merge_kernel<scalar_t, true><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()), n8, L, H, D / 8);
    else
      attn_merge_kernel<scalar_t, false><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()), n8, L, H, D / 8);
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return out;
}

namespace {

// Evoformer MSA arranges: (B, S, L, H, D) <-> head-major batched layouts.
//   ROW mode: (B*H*S, L, D)  (attention over L for each msa row)
//   COL mode: (B*L*H, S, D)  (attention over S for each column)
// torch runs each as a 5-D permute through its strided copy (2-byte
// scalars on one side, ~27k small launches per evoformer step); decoding
// from the OUTPUT index keeps both sides on 16 B vectors.
template <typename T, bool COL, bool INVERSE>
__global__ void msa_arrange_kernel(T* __restrict__ out, const T* __restrict__ in,
                                   int64_t n8, int S, int L, int H, int D8,
                                   int64_t in_rstride8) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    // decode the NATURAL (B,S,L,H,D8) coordinates from whichever side is
    // the kernel's output, then compose the other side's flat index
    int64_t b;
    int s, l, h, d8;
    int64_t tmp = i;
    d8 = (int)(tmp % D8);
    tmp /= D8;
    if (INVERSE) {
      // output is (B,S,L,H,D8)
      h = (int)(tmp % H);
      tmp /= H;
      l = (int)(tmp % L);
      tmp /= L;
      s = (int)(tmp % S);
      b = tmp / S;
    } else if (COL) {
      // output is (B,L,H,S,D8)
      s = (int)(tmp % S);
      tmp /= S;
      h = (int)(tmp % H);
      tmp /= H;
      l = (int)(tmp % L);
      b = tmp / L;
    } else {
      // output is (B,H,S,L,D8)
      l = (int)(tmp % L);
      tmp /= L;
      s = (int)(tmp % S);
      tmp /= S;
      h = (int)(tmp % H);
      b = tmp / H;
    }
    int64_t src;
    if (!INVERSE) {
      // input rows are ((b,s,l)) with an arbitrary row stride (chunk()
      // views of the fused qkv projection stay copy-free)
      src = ((b * S + s) * (int64_t)L + l) * in_rstride8 + h * D8 + d8;
    } else if (COL) {
      // input is (B,L,H,S,D8)
      src = ((((b * L + l) * (int64_t)H + h) * S + s) * D8 + d8);
    } else {
      // input is (B,H,S,L,D8)
      src = ((((b * H + h) * (int64_t)S + s) * L + l) * D8 + d8);
    }
    float f[8];
    load8(in + src * 8, f);
    store8(out + i * 8, f);
  }
}

}  // namespace

at::Tensor msa_arrange(at::Tensor x, int64_t B, int64_t S, int64_t L,
                       int64_t H, bool col, bool inverse) {
  TORCH_CHECK(x.is_cuda(), "msa_arrange: CUDA tensor");
  const int64_t total = B * S * L * H;
  TORCH_CHECK(x.numel() % total == 0, "msa_arrange: bad shape");
  const int64_t D = x.numel() / total;
  TORCH_CHECK(D % 8 == 0, "msa_arrange: head_dim % 8");
  int64_t rstride8 = H * D / 8;
  if (!inverse) {
    // accept a last-dim-contiguous 4-D view (B,S,L,C) with any row stride
    TORCH_CHECK(x.dim() == 4 && x.size(3) == H * D && x.stride(3) == 1 &&
                    x.stride(2) % 8 == 0 &&
                    x.stride(1) == L * x.stride(2) &&
                    x.stride(0) == S * x.stride(1),
                "msa_arrange: need (B,S,L,C) with contiguous C");
    rstride8 = x.stride(2) / 8;
  } else {
    TORCH_CHECK(x.is_contiguous(), "msa_arrange: inverse needs contiguous");
  }
  at::Tensor out;
  if (inverse)
    out = at::empty({B, S, L, H * D}, x.options());
  else if (col)
    out = at::empty({B * L * H, S, D}, x.options());
  else
    out = at::empty({B * H, S, L, D}, x.options());
  const int64_t n8 = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  DISPATCH_FTYPES(x.scalar_type(), "msa_arrange", {
    auto launch = [&](auto col_tag, auto inv_tag) {
      msa_arrange_kernel<scalar_t, decltype(col_tag)::value,
                         decltype(inv_tag)::value><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()), n8, (int)S, (int)L,
          (int)H, (int)(D / 8), rstride8);
    };
    if (col) {
      if (inverse) launch(std::true_type{}, std::true_type{});
      else launch(std::true_type{}, std::false_type{});
    } else {
      if (inverse) launch(std::false_type{}, std::true_type{});
      else launch(std::false_type{}, std::false_type{});
    }
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return out;
}
