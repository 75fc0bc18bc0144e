// Multi-tensor L2 norm (grad clipping) for gfx950.
//
// Functional counterpart of reference csrc/multi_tensor/* — up to 48
// (pointer, numel) pairs are packed into the kernel-argument block per
// launch; each 256-thread block grid-strides over fixed-size chunks,
// accumulates a private fp32 sum of squares and writes ONE partial per
// block (deterministic — no atomics); a second kernel reduces partials
// to sqrt(sum).  Python groups tensors by dtype before calling.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

namespace {

constexpr int kMaxTensors = 48;

struct TensorListMeta {
  const void* ptrs[kMaxTensors];
  int64_t numels[kMaxTensors];
  int64_t chunk_start[kMaxTensors + 1];  // prefix sum of per-tensor chunk counts
  int n_tensors;
};

template <typename T>
__global__ void l2norm_chunk_kernel(TensorListMeta meta, int chunk_size,
                                    int64_t n_chunks, float* __restrict__ partials) {
  float acc = 0.f;
  for (int64_t c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    // find owning tensor (n_tensors <= 48: linear scan on the scalar unit)
    int t = 0;
    while (c >= meta.chunk_start[t + 1]) ++t;
    const T* p = reinterpret_cast<const T*>(meta.ptrs[t]);
    const int64_t base = (c - meta.chunk_start[t]) * (int64_t)chunk_size;
    const int64_t end = min(base + (int64_t)chunk_size, meta.numels[t]);
    // vectorized main body: 4 elems/lane
    int64_t i = base + (int64_t)threadIdx.x * 4;
    for (; i + 3 < end; i += 256 * 4) {
      float f[4];
      if constexpr (sizeof(T) == 2) {
        union {
          uint2 u;
          T t4[4];
        } U;
        U.u = *reinterpret_cast<const uint2*>(p + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) f[j] = Cvt<T>::to_f(U.t4[j]);
      } else {
        const float4 v = *reinterpret_cast<const float4*>(p + i);
        f[0] = v.x; f[1] = v.y; f[2] = v.z; f[3] = v.w;
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) acc += f[j] * f[j];
    }
    // tail (only the last chunk of a tensor can be ragged)
    for (; i < end; ++i) {
      const float f = Cvt<T>::to_f(p[i]);
      acc += f * f;
    }
  }
  __shared__ float red[4];
  acc = wave_sum(acc);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) red[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    partials[blockIdx.x] = red[0] + red[1] + red[2] + red[3];
}

__global__ void l2norm_cleanup_kernel(const float* __restrict__ partials, int n,
                                      float* __restrict__ out, bool accumulate) {
  float acc = 0.f;
  for (int i = threadIdx.x; i < n; i += 256) acc += partials[i];
  __shared__ float red[4];
  acc = wave_sum(acc);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) red[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    const float s = red[0] + red[1] + red[2] + red[3];
    out[0] = accumulate ? out[0] + s : s;
  }
}

__global__ void sqrt_inplace_kernel(float* x) { x[0] = sqrtf(x[0]); }

}  // namespace

// Returns sqrt(sum over all tensors of sum(x^2)) as a 0-dim fp32 CUDA tensor.
at::Tensor multi_tensor_l2norm(int64_t chunk_size, std::vector<at::Tensor> tensors) {
  TORCH_CHECK(!tensors.empty(), "multi_tensor_l2norm: empty tensor list");
  const auto st = tensors[0].scalar_type();
  for (auto& t : tensors) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "multi_tensor_l2norm: tensors must be contiguous CUDA");
    TORCH_CHECK(t.scalar_type() == st, "multi_tensor_l2norm: mixed dtypes");
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  auto out = at::zeros({}, tensors[0].options().dtype(at::kFloat));
  const int grid = 1024;
  auto partials = at::empty({grid}, out.options());

  size_t idx = 0;
  bool first = true;
  while (idx < tensors.size()) {
    TensorListMeta meta;
    meta.n_tensors = 0;
    meta.chunk_start[0] = 0;
    while (idx < tensors.size() && meta.n_tensors < kMaxTensors) {
      const auto& t = tensors[idx];
      const int64_t n = t.numel();
      if (n == 0) {
        ++idx;
        continue;
      }
      const int i = meta.n_tensors;
      meta.ptrs[i] = t.data_ptr();
      meta.numels[i] = n;
      meta.chunk_start[i + 1] =
          meta.chunk_start[i] + (n + chunk_size - 1) / chunk_size;
      ++meta.n_tensors;
      ++idx;
    }
    if (meta.n_tensors == 0) break;
    const int64_t n_chunks = meta.chunk_start[meta.n_tensors];
    const int g = unicore_grid(n_chunks, grid);
    switch (st) {
      case at::ScalarType::Float:
        l2norm_chunk_kernel<float><<<g, 256, 0, stream>>>(
            meta, (int)chunk_size, n_chunks, partials.data_ptr<float>());
        break;
      case at::ScalarType::Half:
        l2norm_chunk_kernel<__half><<<g, 256, 0, stream>>>(
            meta, (int)chunk_size, n_chunks, partials.data_ptr<float>());
        break;
      case at::ScalarType::BFloat16:
        l2norm_chunk_kernel<__hip_bfloat16><<<g, 256, 0, stream>>>(
            meta, (int)chunk_size, n_chunks, partials.data_ptr<float>());
        break;
      default:
        TORCH_CHECK(false, "multi_tensor_l2norm: unsupported dtype");
    }
    l2norm_cleanup_kernel<<<1, 256, 0, stream>>>(
        partials.data_ptr<float>(), g, out.data_ptr<float>(), !first);
    first = false;
  }
  sqrt_inplace_kernel<<<1, 1, 0, stream>>>(out.data_ptr<float>());
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return out;
}
