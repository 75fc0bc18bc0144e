// Fused dropout(x [+ bias]) + residual add + LayerNorm for gfx950.
//
// The post-LN residual join (out = LN(res + dropout(x + b))) is two kernels
// today (dropout_add, then LN) with the summed tensor making a full HBM
// round-trip between them. This kernel computes the sum in registers, writes
// it ONCE (saved for backward / the next residual), reduces mean/var from
// the registers, and writes the normed output — one launch, one read of
// x+res instead of two reads + an extra intermediate read.
//
// Backward reuses the existing kernels unchanged: layernorm_backward on the
// saved sum produces d_sum (+ dgamma/dbeta), and dropout_add_backward maps
// d_sum through the keep-mask (+ folded-bias column sum); d_res = d_sum.
// The dmask layout and Philox keying are IDENTICAL to dropout_add.hip
// (flat 8-element index), so dropout_add_backward consumes the mask as-is.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

#include <optional>
#include <vector>

namespace {

template <typename T, int NV, bool DROP, bool HAS_BIAS>
__global__ void dropout_add_ln_fwd_kernel(
    T* __restrict__ normed, T* __restrict__ summed,
    uint8_t* __restrict__ dmask, float* __restrict__ mean,
    float* __restrict__ invvar, const T* __restrict__ x,
    const T* __restrict__ res, const T* __restrict__ bias,
    const T* __restrict__ gamma, const T* __restrict__ beta, int64_t n1,
    int n2, float eps, float pinv, uint32_t pthresh, uint64_t rngkey) {
  const int lane = threadIdx.x;
  const int wid = threadIdx.y;
  const int row8 = n2 / 8;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.y + wid; row < n1;
       row += (int64_t)gridDim.x * blockDim.y) {
    const int64_t rbase = row * (int64_t)n2;
    float vals[NV][8];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        float fx[8], fr[8];
        load8(x + rbase + e0, fx);
        load8(res + rbase + e0, fr);
        if constexpr (HAS_BIAS) {
          float fb[8];
          load8(bias + e0, fb);
#pragma unroll
          for (int j = 0; j < 8; ++j) fx[j] += fb[j];
        }
        if constexpr (DROP) {
          const int64_t flat8 = row * (int64_t)row8 + lane + i * 64;
          bool keep[8];
          keep16x8(rngkey, (uint64_t)flat8, 0, pthresh, keep);
          uint8_t bits = 0;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            bits |= (uint8_t)(keep[j] ? 1u : 0u) << j;
            fx[j] = keep[j] ? fx[j] * pinv : 0.f;
          }
          dmask[flat8] = bits;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float s = fr[j] + fx[j];
          vals[i][j] = s;
          sum += s;
          sumsq += s * s;
        }
        store8(summed + rbase + e0, vals[i]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[i][j] = 0.f;
      }
    }
    const float mu = wave_sum(sum) / n2;
    const float var = wave_sum(sumsq) / n2 - mu * mu;
    const float iv = rsqrtf(var + eps);
    if (lane == 0) {
      mean[row] = mu;
      invvar[row] = iv;
    }
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        float g[8], b[8], o[8];
        load8(gamma + e0, g);
        load8(beta + e0, b);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = (vals[i][j] - mu) * iv * g[j] + b[j];
        store8(normed + rbase + e0, o);
      }
    }
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

// returns {normed, summed, dmask, mean, invvar}
std::vector<at::Tensor> dropout_add_ln_forward(
    at::Tensor x, at::Tensor res, std::optional<at::Tensor> bias,
    at::Tensor gamma, at::Tensor beta, double p, bool is_training,
    double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && res.is_contiguous(),
              "dropout_add_ln: contiguous CUDA");
  TORCH_CHECK(x.sizes() == res.sizes() && x.scalar_type() == res.scalar_type(),
              "dropout_add_ln: x/res mismatch");
  const int n2 = (int)x.size(-1);
  TORCH_CHECK(n2 % 8 == 0 && n2 <= 2048,
              "dropout_add_ln: hidden must be %%8==0 and <= 2048");
  TORCH_CHECK(gamma.numel() == n2 && beta.numel() == n2 &&
                  gamma.is_contiguous() && beta.is_contiguous(),
              "dropout_add_ln: bad gamma/beta");
  const int64_t n1 = x.numel() / n2;

  const bool drop = is_training && p > 0.0;
  auto normed = at::empty_like(x);
  auto summed = at::empty_like(x);
  auto fopt = x.options().dtype(at::kFloat);
  auto mean = at::empty({n1}, fopt);
  auto invvar = at::empty({n1}, fopt);
  at::Tensor dmask;
  float pinv = 1.f;
  uint32_t pthresh = 0;
  uint64_t rngkey = 0;
  const int64_t n8 = x.numel() / 8;
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
    rngkey = state.seed_.val + state.offset_.val * 0x9E3779B97F4A7C15ull;
  } else {
    dmask = at::empty({0}, x.options().dtype(at::kByte));
  }
  const bool has_bias = bias.has_value();
  at::Tensor bc;
  if (has_bias) {
    bc = bias->contiguous();
    TORCH_CHECK((int)bc.numel() == n2 && bc.scalar_type() == x.scalar_type(),
                "dropout_add_ln: bad bias");
  }

  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 block(64, 4);
  const dim3 grid(unicore_grid((n1 + 3) / 4));
  DISPATCH_FTYPES(x.scalar_type(), "dropout_add_ln_forward", {
    auto launch = [&](auto nv_tag, auto drop_tag, auto bias_tag) {
      constexpr int NV = decltype(nv_tag)::value;
      constexpr bool DROP = decltype(drop_tag)::value;
      constexpr bool HB = decltype(bias_tag)::value;
      dropout_add_ln_fwd_kernel<scalar_t, NV, DROP, HB>
          <<<grid, block, 0, stream>>>(
              reinterpret_cast<scalar_t*>(normed.data_ptr()),
              reinterpret_cast<scalar_t*>(summed.data_ptr()),
              DROP ? dmask.data_ptr<uint8_t>() : nullptr,
              mean.data_ptr<float>(), invvar.data_ptr<float>(),
              reinterpret_cast<const scalar_t*>(x.data_ptr()),
              reinterpret_cast<const scalar_t*>(res.data_ptr()),
              HB ? reinterpret_cast<const scalar_t*>(bc.data_ptr()) : nullptr,
              reinterpret_cast<const scalar_t*>(gamma.data_ptr()),
              reinterpret_cast<const scalar_t*>(beta.data_ptr()), n1, n2,
              (float)eps, pinv, pthresh, rngkey);
    };
    auto pick = [&](auto nv_tag) {
      if (drop) {
        if (has_bias) launch(nv_tag, std::true_type{}, std::true_type{});
        else launch(nv_tag, std::true_type{}, std::false_type{});
      } else {
        if (has_bias) launch(nv_tag, std::false_type{}, std::true_type{});
        else launch(nv_tag, std::false_type{}, std::false_type{});
      }
    };
    if (n2 <= 512)
      pick(std::integral_constant<int, 1>{});
    else if (n2 <= 1024)
      pick(std::integral_constant<int, 2>{});
    else
      pick(std::integral_constant<int, 4>{});
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {normed, summed, dmask, mean, invvar};
}
