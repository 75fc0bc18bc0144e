// Fused gated multiply for gfx950: out = (x + bx) * sigmoid(g + bg).
//
// The Evoformer gating idiom (`proj(p) * sigmoid(gate(p))`, AlphaFold
// style) costs torch a sigmoid kernel + a mul kernel forward and a
// sigmoid-backward chain + two activation-sized bias-grad `.sum` reads
// backward.  Here both projection biases ride the fused kernel (their
// Linears run bias-free) and the backward emits both bias grads as
// deterministic column sums via the common.h colsum machinery.
#include "common.h"
#include "fold.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <optional>
#include <vector>

namespace {

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

__device__ __forceinline__ float sigmoidf_(float v) {
  return 1.0f / (1.0f + __expf(-v));
}

template <typename T, bool HAS_BIAS>
__global__ void gated_mul_fwd_kernel(T* __restrict__ out,
                                     const T* __restrict__ x,
                                     const T* __restrict__ g,
                                     const T* __restrict__ bx,
                                     const T* __restrict__ bg, int C,
                                     int64_t n8) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    float fx[8], fg[8];
    load8(x + i * 8, fx);
    load8(g + i * 8, fg);
    if constexpr (HAS_BIAS) {
      float a[8], b[8];
      const int c0 = (int)((i * 8) % C);
      load8(bx + c0, a);
      load8(bg + c0, b);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        fx[j] += a[j];
        fg[j] += b[j];
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) fx[j] *= sigmoidf_(fg[j]);
    store8(out + i * 8, fx);
  }
}

// dx = grad*s; dg = grad*(x+bx)*s*(1-s); optional column sums of both.
template <typename T, bool BGRAD>
__global__ void gated_mul_bwd_kernel(T* __restrict__ dx, T* __restrict__ dg,
                                     const T* __restrict__ grad,
                                     const T* __restrict__ x,
                                     const T* __restrict__ g,
                                     const T* __restrict__ bx,
                                     const T* __restrict__ bg,
                                     float* __restrict__ partials, int C,
                                     int64_t n8) {
  extern __shared__ float s_col[];  // 2C floats in BGRAD mode
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float accx[8], accg[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) accx[j] = accg[j] = 0.f;
  int c0 = -1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    float fG[8], fx[8], fg[8];
    load8(grad + i * 8, fG);
    load8(x + i * 8, fx);
    load8(g + i * 8, fg);
    if constexpr (BGRAD) {
      if (c0 < 0) c0 = (int)((i * 8) % C);
      float a[8], b[8];
      load8(bx + c0, a);
      load8(bg + c0, b);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        fx[j] += a[j];
        fg[j] += b[j];
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float s = sigmoidf_(fg[j]);
      const float dxv = fG[j] * s;
      const float dgv = fG[j] * fx[j] * s * (1.f - s);
      fx[j] = dxv;
      fg[j] = dgv;
      if constexpr (BGRAD) {
        accx[j] += dxv;
        accg[j] += dgv;
      }
    }
    store8(dx + i * 8, fx);
    store8(dg + i * 8, fg);
  }
  if constexpr (BGRAD) {
    // two independent C-space folds into the two halves of the row
    // (folding both into one 2C space would collide: the data period is
    // C, so threads C/8 apart share a column but can land in one band)
    float* prow = partials + (int64_t)blockIdx.x * 2 * C;
    colsum_block_fold(accx, c0, C, s_col, prow);
    __syncthreads();  // s_col is reused by the second fold
    colsum_block_fold(accg, c0, C, s_col, prow + C);
  }
}

}  // namespace

at::Tensor gated_mul_forward(at::Tensor x, at::Tensor g,
                             std::optional<at::Tensor> bx,
                             std::optional<at::Tensor> bg) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && g.is_contiguous() &&
                  x.sizes() == g.sizes() && x.scalar_type() == g.scalar_type(),
              "gated_mul: bad inputs");
  TORCH_CHECK(x.numel() % 8 == 0, "gated_mul: numel % 8");
  const bool has_bias = bx.has_value();
  TORCH_CHECK(has_bias == bg.has_value(), "gated_mul: need both biases");
  at::Tensor bxc, bgc;
  int C = 0;
  if (has_bias) {
    bxc = bx->contiguous();
    bgc = bg->contiguous();
    C = (int)bxc.numel();
    TORCH_CHECK(C > 0 && C % 8 == 0 && x.size(-1) == C &&
                    bgc.numel() == C && bxc.scalar_type() == x.scalar_type(),
                "gated_mul: bad bias");
  }
  auto out = at::empty_like(x);
  const int64_t n8 = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n8 + 255) / 256);
  DISPATCH_FTYPES(x.scalar_type(), "gated_mul_forward", {
    if (has_bias)
      gated_mul_fwd_kernel<scalar_t, true><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          reinterpret_cast<const scalar_t*>(g.data_ptr()),
          reinterpret_cast<const scalar_t*>(bxc.data_ptr()),
          reinterpret_cast<const scalar_t*>(bgc.data_ptr()), C, n8);
    else
      gated_mul_fwd_kernel<scalar_t, false><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(out.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          reinterpret_cast<const scalar_t*>(g.data_ptr()), nullptr, nullptr,
          C, n8);
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return out;
}

std::vector<at::Tensor> gated_mul_backward(at::Tensor grad, at::Tensor x,
                                           at::Tensor g,
                                           std::optional<at::Tensor> bx,
                                           std::optional<at::Tensor> bg) {
  const bool bgrad = bx.has_value();
  at::Tensor bxc, bgc;
  int C = 0;
  if (bgrad) {
    bxc = bx->contiguous();
    bgc = bg->contiguous();
    C = (int)bxc.numel();
    TORCH_CHECK(colsum_supported(C), "gated_mul_backward: bad bias dim");
  }
  auto gc = grad.contiguous();
  auto dx = at::empty_like(x);
  auto dg = at::empty_like(g);
  auto dbias = at::empty({bgrad ? 2LL * C : 0},
                         grad.options().dtype(at::kFloat));
  const int64_t n8 = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = bgrad ? colsum_grid(n8, C) : unicore_grid((n8 + 255) / 256);
  at::Tensor partials;
  if (bgrad) partials = at::empty({grid, 2LL * C}, dbias.options());
  const size_t lds = bgrad ? (size_t)C * sizeof(float) : 0;
  DISPATCH_FTYPES(x.scalar_type(), "gated_mul_backward", {
    if (bgrad)
      gated_mul_bwd_kernel<scalar_t, true><<<grid, 256, lds, stream>>>(
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<scalar_t*>(dg.data_ptr()),
          reinterpret_cast<const scalar_t*>(gc.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          reinterpret_cast<const scalar_t*>(g.data_ptr()),
          reinterpret_cast<const scalar_t*>(bxc.data_ptr()),
          reinterpret_cast<const scalar_t*>(bgc.data_ptr()),
          partials.data_ptr<float>(), C, n8);
    else
      gated_mul_bwd_kernel<scalar_t, false><<<grid, 256, 0, stream>>>(
          reinterpret_cast<scalar_t*>(dx.data_ptr()),
          reinterpret_cast<scalar_t*>(dg.data_ptr()),
          reinterpret_cast<const scalar_t*>(gc.data_ptr()),
          reinterpret_cast<const scalar_t*>(x.data_ptr()),
          reinterpret_cast<const scalar_t*>(g.data_ptr()), nullptr, nullptr,
          nullptr, C, n8);
  });
  if (bgrad) {
    unicore_fold_columns(partials.data_ptr<float>(), dbias.data_ptr<float>(),
                         grid, 2 * C, partials.options(), stream);
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {dx, dg, dbias};
}
