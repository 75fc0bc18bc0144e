// Fused AdamW for gfx950 — one grid-stride pass per flat tensor fusing grad
// unscale (g / grad_scale), both moment updates, bias correction and the
// decoupled weight-decay update (functional counterpart of reference
// csrc/adam/adam_kernel.cu, matching the eager oracle in
// unicore_amd/optim/adam.py: denom = sqrt(v) + eps, step_size folds
// sqrt(bc2)/bc1, p -= wd*lr*p before the update).
// m/v are fp32; p/g may be fp32/fp16/bf16.  4 elements per lane.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

template <typename T>
__global__ void adam_kernel(T* __restrict__ p, float* __restrict__ m,
                            float* __restrict__ v, const T* __restrict__ g,
                            int64_t n, float step_size, float beta1, float beta2,
                            float eps, float inv_scale, float wd_factor) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4; i0 < n;
       i0 += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int64_t i = i0 + j;
      if (i < n) {
        const float grad = Cvt<T>::to_f(g[i]) * inv_scale;
        float mi = m[i] * beta1 + grad * (1.f - beta1);
        float vi = v[i] * beta2 + grad * grad * (1.f - beta2);
        m[i] = mi;
        v[i] = vi;
        float pi = Cvt<T>::to_f(p[i]);
        pi *= wd_factor;  // 1 - lr*wd (decoupled weight decay)
        pi -= step_size * mi / (sqrtf(vi) + eps);
        p[i] = Cvt<T>::from_f(pi);
      }
    }
  }
}

// vectorized variant when n % 4 == 0 (16 B/lane for fp32, 8 B for 16-bit p/g
// plus 16 B fp32 moments)
template <typename T>
__global__ void adam_kernel_vec(T* __restrict__ p, float* __restrict__ m,
                                float* __restrict__ v, const T* __restrict__ g,
                                int64_t n4, float step_size, float beta1,
                                float beta2, float eps, float inv_scale,
                                float wd_factor) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 mv = reinterpret_cast<float4*>(m)[i];
    float4 vv = reinterpret_cast<float4*>(v)[i];
    float gv[4], pv[4];
    if constexpr (sizeof(T) == 2) {
      union {
        uint2 u;
        T t[4];
      } G, P;
      G.u = reinterpret_cast<const uint2*>(g)[i];
      P.u = reinterpret_cast<const uint2*>(p)[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        gv[j] = Cvt<T>::to_f(G.t[j]);
        pv[j] = Cvt<T>::to_f(P.t[j]);
      }
    } else {
      const float4 G = reinterpret_cast<const float4*>(g)[i];
      const float4 P = reinterpret_cast<const float4*>(p)[i];
      gv[0] = G.x; gv[1] = G.y; gv[2] = G.z; gv[3] = G.w;
      pv[0] = P.x; pv[1] = P.y; pv[2] = P.z; pv[3] = P.w;
    }
    float mo[4], vo[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float grad = gv[j] * inv_scale;
      float mi = (&mv.x)[j] * beta1 + grad * (1.f - beta1);
      float vi = (&vv.x)[j] * beta2 + grad * grad * (1.f - beta2);
      mo[j] = mi;
      vo[j] = vi;
      pv[j] = pv[j] * wd_factor - step_size * mi / (sqrtf(vi) + eps);
    }
    reinterpret_cast<float4*>(m)[i] = make_float4(mo[0], mo[1], mo[2], mo[3]);
    reinterpret_cast<float4*>(v)[i] = make_float4(vo[0], vo[1], vo[2], vo[3]);
    if constexpr (sizeof(T) == 2) {
      union {
        uint2 u;
        T t[4];
      } P;
#pragma unroll
      for (int j = 0; j < 4; ++j) P.t[j] = Cvt<T>::from_f(pv[j]);
      reinterpret_cast<uint2*>(p)[i] = P.u;
    } else {
      reinterpret_cast<float4*>(p)[i] = make_float4(pv[0], pv[1], pv[2], pv[3]);
    }
  }
}

}  // namespace

void fused_adam(at::Tensor p, at::Tensor m, at::Tensor v, at::Tensor g, double lr,
                double beta1, double beta2, double eps, double grad_scale,
                int64_t step, bool bias_correction, double weight_decay) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && g.is_contiguous() &&
                  m.is_contiguous() && v.is_contiguous(),
              "fused_adam: tensors must be contiguous CUDA");
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat,
              "fused_adam: moments must be fp32");
  TORCH_CHECK(g.scalar_type() == p.scalar_type(), "fused_adam: p/g dtype mismatch");
  const int64_t n = p.numel();
  TORCH_CHECK(m.numel() == n && v.numel() == n && g.numel() == n,
              "fused_adam: size mismatch");
  if (n == 0) return;

  double step_size = lr;
  if (bias_correction) {
    const double bc1 = 1.0 - std::pow(beta1, (double)step);
    const double bc2 = 1.0 - std::pow(beta2, (double)step);
    step_size = lr * std::sqrt(bc2) / bc1;
  }
  const float wd_factor = (float)(1.0 - lr * weight_decay);
  const float inv_scale = (float)(1.0 / grad_scale);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int64_t n4 = n / 4;
  const int grid = unicore_grid((n4 + 255) / 256);

  switch (p.scalar_type()) {
    case at::ScalarType::Float: {
      if (n % 4 == 0)
        adam_kernel_vec<float><<<grid, 256, 0, stream>>>(
            p.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
            g.data_ptr<float>(), n4, (float)step_size, (float)beta1, (float)beta2,
            (float)eps, inv_scale, wd_factor);
      else
        adam_kernel<float><<<grid, 256, 0, stream>>>(
            p.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
            g.data_ptr<float>(), n, (float)step_size, (float)beta1, (float)beta2,
            (float)eps, inv_scale, wd_factor);
      break;
    }
    case at::ScalarType::Half: {
      auto* pp = reinterpret_cast<__half*>(p.data_ptr());
      auto* gp = reinterpret_cast<const __half*>(g.data_ptr());
      if (n % 4 == 0)
        adam_kernel_vec<__half><<<grid, 256, 0, stream>>>(
            pp, m.data_ptr<float>(), v.data_ptr<float>(), gp, n4, (float)step_size,
            (float)beta1, (float)beta2, (float)eps, inv_scale, wd_factor);
      else
        adam_kernel<__half><<<grid, 256, 0, stream>>>(
            pp, m.data_ptr<float>(), v.data_ptr<float>(), gp, n, (float)step_size,
            (float)beta1, (float)beta2, (float)eps, inv_scale, wd_factor);
      break;
    }
    case at::ScalarType::BFloat16: {
      auto* pp = reinterpret_cast<__hip_bfloat16*>(p.data_ptr());
      auto* gp = reinterpret_cast<const __hip_bfloat16*>(g.data_ptr());
      if (n % 4 == 0)
        adam_kernel_vec<__hip_bfloat16><<<grid, 256, 0, stream>>>(
            pp, m.data_ptr<float>(), v.data_ptr<float>(), gp, n4, (float)step_size,
            (float)beta1, (float)beta2, (float)eps, inv_scale, wd_factor);
      else
        adam_kernel<__hip_bfloat16><<<grid, 256, 0, stream>>>(
            pp, m.data_ptr<float>(), v.data_ptr<float>(), gp, n, (float)step_size,
            (float)beta1, (float)beta2, (float)eps, inv_scale, wd_factor);
      break;
    }
    default:
      TORCH_CHECK(false, "fused_adam: unsupported dtype");
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
}
