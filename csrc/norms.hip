// Fused LayerNorm / RMSNorm for gfx950 (CDNA4).
//
// Functional counterpart of the reference's layernorm/rmsnorm extensions
// (reference csrc/layernorm/*, csrc/rmsnorm/*), re-designed for wave64:
//  * forward + grad_input: one wave per row, row register-resident as
//    NV x 8 floats/lane (16 B/lane vector loads), __shfl_xor reductions;
//    any hidden size (no 16-dim whitelist — block kernel covers n2 > 4096
//    or n2 % 8 != 0).  fp32 mean/invvar saved per row.
//  * grad_gamma/grad_beta: deterministic two-stage column reduction —
//    stage 1 tiles rows over gridDim.y blocks accumulating fp32 partials
//    in registers (coalesced column-strided loads), stage 2 reduces the
//    partials.  No atomics, bitwise-reproducible.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

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

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <typename T, int NV, bool RMS>
__global__ void norm_fwd_vec_kernel(T* __restrict__ out, float* __restrict__ mean,
                                    float* __restrict__ invvar,
                                    const T* __restrict__ x,
                                    const T* __restrict__ gamma,
                                    const T* __restrict__ beta, int64_t n1, int n2,
                                    float eps) {
  const int lane = threadIdx.x;
  const int wid = threadIdx.y;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.y + wid; row < n1;
       row += (int64_t)gridDim.x * blockDim.y) {
    const T* xrow = x + row * (int64_t)n2;
    float vals[NV][8];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        load8(xrow + e0, vals[i]);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          sum += vals[i][j];
          sumsq += vals[i][j] * vals[i][j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[i][j] = 0.f;
      }
    }
    float mu = 0.f;
    if constexpr (!RMS) {
      mu = wave_sum(sum) / n2;
    }
    sumsq = wave_sum(sumsq);
    const float var = RMS ? sumsq / n2 : sumsq / n2 - mu * mu;
    const float iv = rsqrtf(var + eps);
    if (lane == 0) {
      invvar[row] = iv;
      if constexpr (!RMS) mean[row] = mu;
    }
    T* orow = out + row * (int64_t)n2;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        float g[8], o[8];
        load8(gamma + e0, g);
        if constexpr (!RMS) {
          float b[8];
          load8(beta + e0, b);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            o[j] = (vals[i][j] - mu) * iv * g[j] + b[j];
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) o[j] = vals[i][j] * iv * g[j];
        }
        store8(orow + e0, o);
      }
    }
  }
}

__device__ __forceinline__ void block_red_sum2(float& a, float& b, float* red) {
  a = wave_sum(a);
  b = wave_sum(b);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    red[wid] = a;
    red[4 + wid] = b;
  }
  __syncthreads();
  a = red[0] + red[1] + red[2] + red[3];
  b = red[4] + red[5] + red[6] + red[7];
  __syncthreads();
}

template <typename T, bool RMS>
__global__ void norm_fwd_block_kernel(T* __restrict__ out, float* __restrict__ mean,
                                      float* __restrict__ invvar,
                                      const T* __restrict__ x,
                                      const T* __restrict__ gamma,
                                      const T* __restrict__ beta, int64_t n1, int n2,
                                      float eps) {
  __shared__ float red[8];
  const int tid = threadIdx.x;
  for (int64_t row = blockIdx.x; row < n1; row += gridDim.x) {
    const T* xrow = x + row * (int64_t)n2;
    float sum = 0.f, sumsq = 0.f;
    for (int e = tid; e < n2; e += 256) {
      const float v = Cvt<T>::to_f(xrow[e]);
      sum += v;
      sumsq += v * v;
    }
    block_red_sum2(sum, sumsq, red);
    const float mu = RMS ? 0.f : sum / n2;
    const float var = RMS ? sumsq / n2 : sumsq / n2 - mu * mu;
    const float iv = rsqrtf(var + eps);
    if (tid == 0) {
      invvar[row] = iv;
      if constexpr (!RMS) mean[row] = mu;
    }
    T* orow = out + row * (int64_t)n2;
    for (int e = tid; e < n2; e += 256) {
      const float v = Cvt<T>::to_f(xrow[e]);
      float o = (v - mu) * iv * Cvt<T>::to_f(gamma[e]);
      if constexpr (!RMS) o += Cvt<T>::to_f(beta[e]);
      orow[e] = Cvt<T>::from_f(o);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// backward: grad_input (row-parallel)
// ---------------------------------------------------------------------------
template <typename T, int NV, bool RMS>
__global__ void norm_bwd_dx_vec_kernel(T* __restrict__ dx, const T* __restrict__ dy,
                                       const T* __restrict__ x,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ invvar,
                                       const T* __restrict__ gamma, int64_t n1,
                                       int n2) {
  const int lane = threadIdx.x;
  const int wid = threadIdx.y;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.y + wid; row < n1;
       row += (int64_t)gridDim.x * blockDim.y) {
    const float iv = invvar[row];
    const float mu = RMS ? 0.f : mean[row];
    const T* dyrow = dy + row * (int64_t)n2;
    const T* xrow = x + row * (int64_t)n2;
    float dyg[NV][8], xhat[NV][8];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        float d[8], xv[8], g[8];
        load8(dyrow + e0, d);
        load8(xrow + e0, xv);
        load8(gamma + e0, g);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dyg[i][j] = d[j] * g[j];
          xhat[i][j] = (xv[j] - mu) * iv;
          s1 += dyg[i][j];
          s2 += dyg[i][j] * xhat[i][j];
        }
      }
    }
    if constexpr (!RMS) s1 = wave_sum(s1);
    s2 = wave_sum(s2);
    const float c1 = RMS ? 0.f : s1 / n2;
    const float c2 = s2 / n2;
    T* dxrow = dx + row * (int64_t)n2;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < n2) {
        float o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = iv * (dyg[i][j] - c1 - xhat[i][j] * c2);
        store8(dxrow + e0, o);
      }
    }
  }
}

template <typename T, bool RMS>
__global__ void norm_bwd_dx_block_kernel(T* __restrict__ dx, const T* __restrict__ dy,
                                         const T* __restrict__ x,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ invvar,
                                         const T* __restrict__ gamma, int64_t n1,
                                         int n2) {
  __shared__ float red[8];
  const int tid = threadIdx.x;
  for (int64_t row = blockIdx.x; row < n1; row += gridDim.x) {
    const float iv = invvar[row];
    const float mu = RMS ? 0.f : mean[row];
    const T* dyrow = dy + row * (int64_t)n2;
    const T* xrow = x + row * (int64_t)n2;
    float s1 = 0.f, s2 = 0.f;
    for (int e = tid; e < n2; e += 256) {
      const float dg = Cvt<T>::to_f(dyrow[e]) * Cvt<T>::to_f(gamma[e]);
      const float xh = (Cvt<T>::to_f(xrow[e]) - mu) * iv;
      s1 += dg;
      s2 += dg * xh;
    }
    block_red_sum2(s1, s2, red);
    const float c1 = RMS ? 0.f : s1 / n2;
    const float c2 = s2 / n2;
    T* dxrow = dx + row * (int64_t)n2;
    for (int e = tid; e < n2; e += 256) {
      const float dg = Cvt<T>::to_f(dyrow[e]) * Cvt<T>::to_f(gamma[e]);
      const float xh = (Cvt<T>::to_f(xrow[e]) - mu) * iv;
      dxrow[e] = Cvt<T>::from_f(iv * (dg - c1 - xh * c2));
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// backward: grad_gamma / grad_beta — two-stage deterministic column reduce
// ---------------------------------------------------------------------------
// stage 1: blockDim = (64, 4); block (bx, by) accumulates rows
// [by*rows_per_cta, (by+1)*rows_per_cta) for 4 interleaved column groups
// col = bx*256 + tx + j*64 (coalesced loads), partials fp32 [gridDim.y][n2].
template <typename T, bool RMS>
__global__ void norm_bwd_gb_partial_kernel(const T* __restrict__ dy,
                                           const T* __restrict__ x,
                                           const float* __restrict__ mean,
                                           const float* __restrict__ invvar,
                                           int64_t n1, int n2, int rows_per_cta,
                                           float* __restrict__ part_g,
                                           float* __restrict__ part_b) {
  __shared__ float lds[4][64][8];
  const int tx = threadIdx.x;
  const int ty = threadIdx.y;
  const int64_t row_start = (int64_t)blockIdx.y * rows_per_cta;
  const int64_t row_end = min(row_start + rows_per_cta, n1);
  float acc_g[4] = {0.f, 0.f, 0.f, 0.f};
  float acc_b[4] = {0.f, 0.f, 0.f, 0.f};
  int cols[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) cols[j] = blockIdx.x * 256 + tx + j * 64;
  for (int64_t r = row_start + ty; r < row_end; r += 4) {
    const float iv = invvar[r];
    const float mu = RMS ? 0.f : mean[r];
    const T* dyrow = dy + r * (int64_t)n2;
    const T* xrow = x + r * (int64_t)n2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (cols[j] < n2) {
        const float d = Cvt<T>::to_f(dyrow[cols[j]]);
        acc_g[j] += d * (Cvt<T>::to_f(xrow[cols[j]]) - mu) * iv;
        acc_b[j] += d;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    lds[ty][tx][j] = acc_g[j];
    lds[ty][tx][4 + j] = acc_b[j];
  }
  __syncthreads();
  if (ty == 0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (cols[j] < n2) {
        const int64_t o = (int64_t)blockIdx.y * n2 + cols[j];
        part_g[o] = lds[0][tx][j] + lds[1][tx][j] + lds[2][tx][j] + lds[3][tx][j];
        if constexpr (!RMS)
          part_b[o] =
              lds[0][tx][4 + j] + lds[1][tx][4 + j] + lds[2][tx][4 + j] + lds[3][tx][4 + j];
      }
    }
  }
}

// stage 1, vectorized (n2 % (64*NV) == 0): lane tx owns NV consecutive
// columns at (bx*64 + tx)*NV, loaded as one 2*NV-byte access — the scalar
// variant's per-element 2-byte loads held this kernel to ~0.85 TB/s fabric
// (PMC, profiles/bert_r2_pmc.txt) while its elementwise peers ran 3.2-3.9.
template <typename T, bool RMS, int NV>
__global__ void norm_bwd_gb_partial_vec_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ invvar,
    int64_t n1, int n2, int rows_per_cta, float* __restrict__ part_g,
    float* __restrict__ part_b) {
  __shared__ float lds[4][64][2 * NV];
  const int tx = threadIdx.x;
  const int ty = threadIdx.y;
  const int c0 = (blockIdx.x * 64 + tx) * NV;
  const int64_t row_start = (int64_t)blockIdx.y * rows_per_cta;
  const int64_t row_end = min(row_start + (int64_t)rows_per_cta, n1);
  float acc_g[NV] = {}, acc_b[NV] = {};
  for (int64_t r = row_start + ty; r < row_end; r += 4) {
    const float iv = invvar[r];
    const float mu = RMS ? 0.f : mean[r];
    float d[NV], xv[NV];
    loadN<T, NV>(dy + r * (int64_t)n2 + c0, d);
    loadN<T, NV>(x + r * (int64_t)n2 + c0, xv);
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      acc_g[j] += d[j] * (xv[j] - mu) * iv;
      acc_b[j] += d[j];
    }
  }
#pragma unroll
  for (int j = 0; j < NV; ++j) {
    lds[ty][tx][j] = acc_g[j];
    lds[ty][tx][NV + j] = acc_b[j];
  }
  __syncthreads();
  if (ty == 0) {
    const int64_t o = (int64_t)blockIdx.y * n2 + c0;
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      part_g[o + j] =
          lds[0][tx][j] + lds[1][tx][j] + lds[2][tx][j] + lds[3][tx][j];
      if constexpr (!RMS)
        part_b[o + j] = lds[0][tx][NV + j] + lds[1][tx][NV + j] +
                        lds[2][tx][NV + j] + lds[3][tx][NV + j];
    }
  }
}

// fold [rb][n2] partials down to [gridDim.y][n2] (keeps the final reduce
// wide; without this the last stage ran on ceil(n2/256) blocks serially over
// hundreds of partial rows — measured 12% of a BERT-base step)
__global__ void norm_gb_fold_kernel(const float* __restrict__ in_g,
                                    const float* __restrict__ in_b, int rb,
                                    int chunk, int n2, float* __restrict__ out_g,
                                    float* __restrict__ out_b) {
  const int col = blockIdx.x * 256 + threadIdx.x;
  if (col >= n2) return;
  const int r0 = blockIdx.y * chunk;
  const int r1 = min(r0 + chunk, rb);
  float sg = 0.f, sb = 0.f;
  for (int r = r0; r < r1; ++r) {
    sg += in_g[(int64_t)r * n2 + col];
    if (in_b) sb += in_b[(int64_t)r * n2 + col];
  }
  out_g[(int64_t)blockIdx.y * n2 + col] = sg;
  if (in_b) out_b[(int64_t)blockIdx.y * n2 + col] = sb;
}

template <typename T, bool RMS>
__global__ void norm_bwd_gb_reduce_kernel(const float* __restrict__ part_g,
                                          const float* __restrict__ part_b, int rb,
                                          int n2, T* __restrict__ dg,
                                          T* __restrict__ db) {
  for (int col = blockIdx.x * 256 + threadIdx.x; col < n2;
       col += gridDim.x * 256) {
    float sg = 0.f, sb = 0.f;
    for (int r = 0; r < rb; ++r) {
      sg += part_g[(int64_t)r * n2 + col];
      if constexpr (!RMS) sb += part_b[(int64_t)r * n2 + col];
    }
    dg[col] = Cvt<T>::from_f(sg);
    if constexpr (!RMS) db[col] = Cvt<T>::from_f(sb);
  }
}

struct NormShape {
  int64_t n1;
  int n2;
};

NormShape norm_shape(const at::Tensor& input, const at::Tensor& gamma,
                     const char* name) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous(), name,
              ": input must be contiguous CUDA");
  TORCH_CHECK(gamma.is_cuda() && gamma.is_contiguous(), name,
              ": gamma must be contiguous CUDA");
  TORCH_CHECK(gamma.scalar_type() == input.scalar_type(), name,
              ": gamma dtype must match input");
  const int64_t n2 = gamma.numel();
  TORCH_CHECK(n2 > 0 && n2 <= INT32_MAX && input.numel() % n2 == 0, name,
              ": bad shapes");
  return {input.numel() / n2, (int)n2};
}

template <bool RMS>
std::vector<at::Tensor> norm_forward_impl(const at::Tensor& input,
                                          const at::Tensor& gamma,
                                          const at::Tensor& beta, double eps) {
  const auto s = norm_shape(input, gamma, RMS ? "rmsnorm" : "layernorm");
  auto out = at::empty_like(input);
  auto fopt = input.options().dtype(at::kFloat);
  auto invvar = at::empty({s.n1}, fopt);
  at::Tensor mean;
  if (!RMS) mean = at::empty({s.n1}, fopt);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool vec_ok = (s.n2 % 8 == 0) && s.n2 <= 4096;
  DISPATCH_FTYPES(input.scalar_type(), "norm_forward", {
    const scalar_t* xp = reinterpret_cast<const scalar_t*>(input.data_ptr());
    const scalar_t* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const scalar_t* bp =
        RMS ? nullptr : reinterpret_cast<const scalar_t*>(beta.data_ptr());
    scalar_t* op = reinterpret_cast<scalar_t*>(out.data_ptr());
    float* mp = RMS ? nullptr : mean.data_ptr<float>();
    float* ivp = invvar.data_ptr<float>();
    if (vec_ok) {
      const dim3 block(64, 4);
      const dim3 grid(unicore_grid((s.n1 + 3) / 4));
      auto launch = [&](auto nv_tag) {
        constexpr int NV = decltype(nv_tag)::value;
        norm_fwd_vec_kernel<scalar_t, NV, RMS><<<grid, block, 0, stream>>>(
            op, mp, ivp, xp, gp, bp, s.n1, s.n2, (float)eps);
      };
      if (s.n2 <= 512)
        launch(std::integral_constant<int, 1>{});
      else if (s.n2 <= 1024)
        launch(std::integral_constant<int, 2>{});
      else if (s.n2 <= 2048)
        launch(std::integral_constant<int, 4>{});
      else
        launch(std::integral_constant<int, 8>{});
    } else {
      norm_fwd_block_kernel<scalar_t, RMS>
          <<<unicore_grid(s.n1), 256, 0, stream>>>(op, mp, ivp, xp, gp, bp, s.n1,
                                                   s.n2, (float)eps);
    }
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  if (RMS) return {out, invvar};
  return {out, mean, invvar};
}

template <bool RMS>
std::vector<at::Tensor> norm_backward_impl(const at::Tensor& grad_out,
                                           const at::Tensor& input,
                                           const at::Tensor& mean,
                                           const at::Tensor& invvar,
                                           const at::Tensor& gamma) {
  const auto s = norm_shape(input, gamma, RMS ? "rmsnorm_bwd" : "layernorm_bwd");
  TORCH_CHECK(grad_out.is_cuda() && grad_out.is_contiguous() &&
                  grad_out.sizes() == input.sizes(),
              "norm_backward: bad grad");
  auto dx = at::empty_like(input);
  auto dg = at::empty_like(gamma);
  at::Tensor db;
  if (!RMS) db = at::empty_like(gamma);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool vec_ok = (s.n2 % 8 == 0) && s.n2 <= 4096;

  // stage-1 geometry for dgamma/dbeta; rb capped at 256 so the fp32 partial
  // tiles stay L2-resident for the fold stage
  const int gx = (s.n2 + 255) / 256;
  int rb = (int)std::min<int64_t>(std::max(2048 / gx, 1), (s.n1 + 3) / 4);
  rb = std::max(std::min(rb, 256), 1);
  const int rows_per_cta = (int)((s.n1 + rb - 1) / rb);
  auto fopt = input.options().dtype(at::kFloat);
  auto part_g = at::empty({rb, (int64_t)s.n2}, fopt);
  at::Tensor part_b;
  if (!RMS) part_b = at::empty({rb, (int64_t)s.n2}, fopt);

  DISPATCH_FTYPES(input.scalar_type(), "norm_backward", {
    const scalar_t* dyp = reinterpret_cast<const scalar_t*>(grad_out.data_ptr());
    const scalar_t* xp = reinterpret_cast<const scalar_t*>(input.data_ptr());
    const scalar_t* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const float* mp = RMS ? nullptr : mean.data_ptr<float>();
    const float* ivp = invvar.data_ptr<float>();
    scalar_t* dxp = reinterpret_cast<scalar_t*>(dx.data_ptr());

    if (vec_ok) {
      const dim3 block(64, 4);
      const dim3 grid(unicore_grid((s.n1 + 3) / 4));
      auto launch = [&](auto nv_tag) {
        constexpr int NV = decltype(nv_tag)::value;
        norm_bwd_dx_vec_kernel<scalar_t, NV, RMS><<<grid, block, 0, stream>>>(
            dxp, dyp, xp, mp, ivp, gp, s.n1, s.n2);
      };
      if (s.n2 <= 512)
        launch(std::integral_constant<int, 1>{});
      else if (s.n2 <= 1024)
        launch(std::integral_constant<int, 2>{});
      else if (s.n2 <= 2048)
        launch(std::integral_constant<int, 4>{});
      else
        launch(std::integral_constant<int, 8>{});
    } else {
      norm_bwd_dx_block_kernel<scalar_t, RMS>
          <<<unicore_grid(s.n1), 256, 0, stream>>>(dxp, dyp, xp, mp, ivp, gp, s.n1,
                                                   s.n2);
    }

    const dim3 gb_block(64, 4);
    // vector stage 1 when a whole number of 64-lane x NV-column blocks
    // tiles the row exactly; scalar fallback otherwise
    const int gb_nv = (s.n2 % 512 == 0) ? 8
                      : (s.n2 % 256 == 0) ? 4
                      : (s.n2 % 128 == 0) ? 2 : 0;
    if (gb_nv > 0) {
      const dim3 vec_grid(s.n2 / (64 * gb_nv), rb);
      auto launch_gb = [&](auto nv_tag) {
        constexpr int NV = decltype(nv_tag)::value;
        norm_bwd_gb_partial_vec_kernel<scalar_t, RMS, NV>
            <<<vec_grid, gb_block, 0, stream>>>(
                dyp, xp, mp, ivp, s.n1, s.n2, rows_per_cta,
                part_g.data_ptr<float>(),
                RMS ? nullptr : part_b.data_ptr<float>());
      };
      if (gb_nv == 8)
        launch_gb(std::integral_constant<int, 8>{});
      else if (gb_nv == 4)
        launch_gb(std::integral_constant<int, 4>{});
      else
        launch_gb(std::integral_constant<int, 2>{});
    } else {
      const dim3 gb_grid(gx, rb);
      norm_bwd_gb_partial_kernel<scalar_t, RMS><<<gb_grid, gb_block, 0, stream>>>(
          dyp, xp, mp, ivp, s.n1, s.n2, rows_per_cta, part_g.data_ptr<float>(),
          RMS ? nullptr : part_b.data_ptr<float>());
    }
    const float* red_g = part_g.data_ptr<float>();
    const float* red_b = RMS ? nullptr : part_b.data_ptr<float>();
    int red_rb = rb;
    at::Tensor fold_g, fold_b;
    if (rb > 16) {
      const int fold_rows = 16;
      fold_g = at::empty({fold_rows, (int64_t)s.n2}, fopt);
      if (!RMS) fold_b = at::empty({fold_rows, (int64_t)s.n2}, fopt);
      const int chunk = (rb + fold_rows - 1) / fold_rows;
      norm_gb_fold_kernel<<<dim3(gx, fold_rows), 256, 0, stream>>>(
          red_g, red_b, rb, chunk, s.n2, fold_g.data_ptr<float>(),
          RMS ? nullptr : fold_b.data_ptr<float>());
      red_g = fold_g.data_ptr<float>();
      red_b = RMS ? nullptr : fold_b.data_ptr<float>();
      red_rb = fold_rows;
    }
    norm_bwd_gb_reduce_kernel<scalar_t, RMS>
        <<<unicore_grid(gx), 256, 0, stream>>>(
            red_g, red_b, red_rb, s.n2,
            reinterpret_cast<scalar_t*>(dg.data_ptr()),
            RMS ? nullptr : reinterpret_cast<scalar_t*>(db.data_ptr()));
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  if (RMS) return {dx, dg};
  return {dx, dg, db};
}

}  // namespace

std::vector<at::Tensor> layernorm_forward(at::Tensor input, at::Tensor gamma,
                                          at::Tensor beta, double eps) {
  return norm_forward_impl<false>(input, gamma, beta, eps);
}

std::vector<at::Tensor> layernorm_backward(at::Tensor grad_out, at::Tensor input,
                                           at::Tensor mean, at::Tensor invvar,
                                           at::Tensor gamma) {
  return norm_backward_impl<false>(grad_out, input, mean, invvar, gamma);
}

std::vector<at::Tensor> rmsnorm_forward(at::Tensor input, at::Tensor gamma,
                                        double eps) {
  return norm_forward_impl<true>(input, gamma, gamma, eps);
}

std::vector<at::Tensor> rmsnorm_backward(at::Tensor grad_out, at::Tensor input,
                                         at::Tensor invvar, at::Tensor gamma) {
  return norm_backward_impl<true>(grad_out, input, invvar, invvar, gamma);
}
