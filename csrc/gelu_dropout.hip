// Fused GELU(+bias-free)+dropout for the FFN inner activation on gfx950.
//
// Replaces the torch gelu -> dropout kernel pair (each a full read+write of
// the (B, L, 4E) inner tensor) with one pass each way.  Exact (erf) GELU to
// match torch's default; dropout keep-mask stored as a bitfield (one uint8
// per 8-element vector, same contract as the softmax_dropout kernel);
// backward recomputes gelu'(x) from the saved input.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

#include <optional>
#include <vector>

namespace {

__device__ __forceinline__ float gelu_fwd(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
}

__device__ __forceinline__ float gelu_grad(float x) {
  const float cdf = 0.5f * (1.0f + erff(x * 0.70710678118654752f));
  const float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

template <typename T, bool DROP, bool HAS_BIAS>
__global__ void gelu_dropout_fwd_kernel(T* __restrict__ out,
                                        uint8_t* __restrict__ dmask,
                                        const T* __restrict__ x,
                                        const T* __restrict__ bias, int C,
                                        int64_t n8,
                                        float pinv, uint32_t pthresh,
                                        uint64_t seed, uint64_t offset) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i < n8; i += stride) {
    float f[8];
    load8(x + i * 8, f);
    if constexpr (HAS_BIAS) {
      float fb[8];
      load8(bias + (int)((i * 8) % C), fb);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += fb[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = gelu_fwd(f[j]);
    if constexpr (DROP) {
      bool keep[8];
      keep16x8(seed + offset * 0x9E3779B97F4A7C15ull, (uint64_t)i, 0, pthresh,
               keep);
      uint8_t bits = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        bits |= (uint8_t)(keep[j] ? 1u : 0u) << j;
        f[j] = keep[j] ? f[j] * pinv : 0.f;
      }
      dmask[i] = bits;
    }
    store8(out + i * 8, f);
  }
}

template <typename T, bool DROP, bool BGRAD>
__global__ void gelu_dropout_bwd_kernel(T* __restrict__ dx, const T* __restrict__ g,
                                        const T* __restrict__ x,
                                        const T* __restrict__ bias,
                                        const uint8_t* __restrict__ dmask,
                                        float* __restrict__ partials, int C,
                                        int64_t n8, float pinv) {
  extern __shared__ float s_col[];
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  int c0 = -1;
  for (int64_t i = tid; i < n8; i += stride) {
    float gv[8], xv[8];
    load8(g + i * 8, gv);
    load8(x + i * 8, xv);
    if constexpr (BGRAD) {
      // the saved x is the bias-free Linear output: rebuild gelu's input
      if (c0 < 0) c0 = (int)((i * 8) % C);
      float fb[8];
      load8(bias + c0, fb);
#pragma unroll
      for (int j = 0; j < 8; ++j) xv[j] += fb[j];
    }
    uint8_t bits = 0xFF;
    if constexpr (DROP) bits = dmask[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = gv[j];
      if constexpr (DROP) t = (bits >> j) & 1 ? t * pinv : 0.f;
      gv[j] = t * gelu_grad(xv[j]);
      if constexpr (BGRAD) acc[j] += gv[j];
    }
    store8(dx + i * 8, gv);
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

std::vector<at::Tensor> gelu_dropout_forward(at::Tensor x,
                                             std::optional<at::Tensor> bias,
                                             double p, bool is_training) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "gelu_dropout: contiguous CUDA");
  TORCH_CHECK(x.numel() % 8 == 0, "gelu_dropout: numel % 8 == 0");
  const int64_t n8 = x.numel() / 8;
  const bool drop = is_training && p > 0.0;
  auto out = at::empty_like(x);
  at::Tensor dmask;
  float pinv = 1.f;
  uint32_t pthresh = 0;
  uint64_t seed = 0, offset = 0;
  if (drop) {
    dmask = at::empty({n8}, x.options().dtype(at::kByte));
    const double pc = std::min(p, 0.999999);
    pinv = (float)(1.0 / (1.0 - pc));
    pthresh = keep16_threshold(pc);
    auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
        std::nullopt, at::cuda::detail::getDefaultCUDAGenerator());
    at::PhiloxCudaState state;
    {
      std::lock_guard<std::mutex> lock(gen->mutex_);
      state = gen->philox_cuda_state(4 + n8 / (2048LL * 256) * 2);
    }
    seed = state.seed_.val;
    offset = state.offset_.val;
  } else {
    dmask = at::empty({0}, x.options().dtype(at::kByte));
  }
  const bool has_bias = bias.has_value();
  at::Tensor bc;
  int C = 0;
  if (has_bias) {
    bc = bias->contiguous();
    C = (int)bc.numel();
    TORCH_CHECK(C > 0 && C % 8 == 0 && x.size(-1) == C &&
                    bc.scalar_type() == x.scalar_type(),
                "gelu_dropout: bad bias");
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  DISPATCH_FTYPES(x.scalar_type(), "gelu_dropout_forward", {
    auto launch = [&](auto drop_tag, auto bias_tag) {
      constexpr bool DROP = decltype(drop_tag)::value;
      constexpr bool HB = decltype(bias_tag)::value;
      gelu_dropout_fwd_kernel<scalar_t, DROP, HB><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          DROP ? dmask.data_ptr<uint8_t>() : nullptr,
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          HB ? reinterpret_cast<const scalar_t*>(bc.data_ptr()) : nullptr, C,
          n8, pinv, pthresh, seed, offset);
    };
    if (drop) {
      if (has_bias) launch(std::true_type{}, std::true_type{});
      else launch(std::true_type{}, std::false_type{});
    } else {
      if (has_bias) launch(std::false_type{}, std::true_type{});
      else launch(std::false_type{}, std::false_type{});
    }
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {out, dmask};
}

std::vector<at::Tensor> gelu_dropout_backward(at::Tensor grad, at::Tensor x,
                                              std::optional<at::Tensor> bias,
                                              at::Tensor dmask, double p) {
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() && x.is_contiguous(),
              "gelu_dropout_backward: contiguous CUDA");
  const int64_t n8 = x.numel() / 8;
  const bool drop = dmask.defined() && dmask.numel() > 0;
  const bool bgrad = bias.has_value();
  at::Tensor bc;
  int C = 0;
  if (bgrad) {
    bc = bias->contiguous();
    C = (int)bc.numel();
    TORCH_CHECK(colsum_supported(C), "gelu_dropout_backward: bad bias dim");
  }
  const float pinv =
      drop ? (float)(1.0 / (1.0 - std::min(p, 0.999999))) : 1.f;
  auto dx = at::empty_like(x);
  auto dbias = at::empty({bgrad ? (int64_t)C : 0},
                         grad.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = bgrad ? colsum_grid(n8, C) : unicore_grid((n8 + 255) / 256);
  at::Tensor partials;
  if (bgrad) partials = at::empty({grid, (int64_t)C}, dbias.options());
  const size_t lds = bgrad ? (size_t)C * sizeof(float) : 0;
  DISPATCH_FTYPES(x.scalar_type(), "gelu_dropout_backward", {
    auto launch = [&](auto drop_tag, auto bg_tag) {
      constexpr bool DROP = decltype(drop_tag)::value;
      constexpr bool BG = decltype(bg_tag)::value;
      gelu_dropout_bwd_kernel<scalar_t, DROP, BG><<<grid, 256, lds, stream>>>(
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<const scalar_t*>(grad.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          BG ? reinterpret_cast<const scalar_t*>(bc.data_ptr()) : nullptr,
          DROP ? dmask.data_ptr<uint8_t>() : nullptr,
          BG ? partials.data_ptr<float>() : nullptr, C, n8, pinv);
    };
    if (drop) {
      if (bgrad) launch(std::true_type{}, std::true_type{});
      else launch(std::true_type{}, std::false_type{});
    } else {
      if (bgrad) launch(std::false_type{}, std::true_type{});
      else launch(std::false_type{}, std::false_type{});
    }
  });
  if (bgrad) {
    unicore_col_fold_kernel<<<C, 256, 0, stream>>>(
        partials.data_ptr<float>(), dbias.data_ptr<float>(), grid, C);
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {dx, dbias};
}
