// Fused dropout(x) + residual add for gfx950: out = residual + keep*pinv*x.
// Replaces the torch dropout kernel + add kernel pair (saves one full
// read+write of the hidden tensor per site; two sites per transformer
// layer).  Bitfield keep-mask, Philox keyed by PyTorch's generator.
// Backward: dx = g * keep * pinv (one pass); d_residual = g (pass-through,
// no kernel).
#include "common.h"
#include "fold.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

#include <optional>
#include <vector>

namespace {

template <typename T, bool DROP, bool HAS_BIAS>
__global__ void dropout_add_fwd_kernel(T* __restrict__ out,
                                       uint8_t* __restrict__ dmask,
                                       const T* __restrict__ x,
                                       const T* __restrict__ res,
                                       const T* __restrict__ bias, int C,
                                       int64_t n8,
                                       float pinv, uint32_t pthresh,
                                       uint64_t seed, uint64_t offset) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i < n8; i += stride) {
    float fx[8], fr[8];
    load8(x + i * 8, fx);
    load8(res + i * 8, fr);
    if constexpr (HAS_BIAS) {
      float fb[8];
      load8(bias + (int)((i * 8) % C), fb);
#pragma unroll
      for (int j = 0; j < 8; ++j) fx[j] += fb[j];
    }
    if constexpr (DROP) {
      bool keep[8];
      keep16x8(seed + offset * 0x9E3779B97F4A7C15ull, (uint64_t)i, 0, pthresh,
               keep);
      uint8_t bits = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        bits |= (uint8_t)(keep[j] ? 1u : 0u) << j;
        fx[j] = fr[j] + (keep[j] ? fx[j] * pinv : 0.f);
      }
      dmask[i] = bits;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) fx[j] += fr[j];
    }
    store8(out + i * 8, fx);
  }
}

template <typename T, bool DROP, bool BGRAD>
__global__ void dropout_add_bwd_kernel(T* __restrict__ dx, const T* __restrict__ g,
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
    if (BGRAD && c0 < 0) c0 = (int)((i * 8) % C);
    float f[8];
    load8(g + i * 8, f);
    if constexpr (DROP) {
      const uint8_t bits = dmask[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] = (bits >> j) & 1 ? f[j] * pinv : 0.f;
      store8(dx + i * 8, f);
    }
    if constexpr (BGRAD) {
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += f[j];
    }
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

std::vector<at::Tensor> dropout_add_forward(at::Tensor x, at::Tensor res,
                                            std::optional<at::Tensor> bias,
                                            double p, bool is_training) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && res.is_contiguous(),
              "dropout_add: contiguous CUDA");
  TORCH_CHECK(x.sizes() == res.sizes() && x.scalar_type() == res.scalar_type(),
              "dropout_add: x/res mismatch");
  TORCH_CHECK(x.numel() % 8 == 0, "dropout_add: numel % 8 == 0");
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
                "dropout_add: bad bias");
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  DISPATCH_FTYPES(x.scalar_type(), "dropout_add_forward", {
    auto launch = [&](auto drop_tag, auto bias_tag) {
      constexpr bool DROP = decltype(drop_tag)::value;
      constexpr bool HB = decltype(bias_tag)::value;
      dropout_add_fwd_kernel<scalar_t, DROP, HB><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          DROP ? dmask.data_ptr<uint8_t>() : nullptr,
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          reinterpret_cast<const scalar_t*>(res.data_ptr()),
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

std::vector<at::Tensor> dropout_add_backward(at::Tensor grad, at::Tensor dmask,
                                             double p, int64_t bias_dim) {
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous(), "dropout_add_backward");
  const int64_t n8 = grad.numel() / 8;
  const bool drop = dmask.numel() > 0;
  TORCH_CHECK(!drop || dmask.numel() == n8, "dropout_add_backward: mask mismatch");
  const bool bgrad = bias_dim > 0;
  const int C = (int)bias_dim;
  TORCH_CHECK(!bgrad || colsum_supported(C), "dropout_add_backward: bad bias dim");
  const float pinv = (float)(1.0 / (1.0 - std::min(p, 0.999999)));
  // !drop: dx == grad, the caller aliases it; only the colsum runs
  auto dx = drop ? at::empty_like(grad) : grad;
  auto dbias = at::empty({bgrad ? (int64_t)C : 0},
                         grad.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = bgrad ? colsum_grid(n8, C) : unicore_grid((n8 + 255) / 256);
  at::Tensor partials;
  if (bgrad) partials = at::empty({grid, (int64_t)C}, dbias.options());
  const size_t lds = bgrad ? (size_t)C * sizeof(float) : 0;
  DISPATCH_FTYPES(grad.scalar_type(), "dropout_add_backward", {
    auto launch = [&](auto drop_tag, auto bg_tag) {
      constexpr bool DROP = decltype(drop_tag)::value;
      constexpr bool BG = decltype(bg_tag)::value;
      if (!DROP && !BG) return;  // nothing to do
      dropout_add_bwd_kernel<scalar_t, DROP, BG><<<grid, 256, lds, stream>>>(
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<const scalar_t*>(grad.data_ptr()),
          DROP ? dmask.data_ptr<uint8_t>() : nullptr,
          BG ? partials.data_ptr<float>() : nullptr, C, n8, pinv);
    };
    if (drop) {
      if (bgrad) launch(std::true_type{}, std::true_type{});
      else launch(std::true_type{}, std::false_type{});
    } else if (bgrad) {
      launch(std::false_type{}, std::true_type{});
    }
  });
  if (bgrad) {
    unicore_fold_columns(partials.data_ptr<float>(), dbias.data_ptr<float>(),
                         grid, C, partials.options(), stream);
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {dx, dbias};
}
