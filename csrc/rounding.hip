// fp32 -> bf16 stochastic rounding for gfx950 (parameter writeback in the
// bf16 optimizer, functional counterpart of reference csrc/rounding/*).
//
// Per element: add a uniform 16-bit value below the bf16 mantissa boundary
// to the fp32 bit pattern, then truncate to the high 16 bits.  Philox
// keyed by PyTorch's generator; every rank calls this under a
// rank-identical torch_seed so the flat fp32->bf16 writeback is identical
// across data-parallel replicas (reference trainer.py:712-713 contract).
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

namespace {

__global__ void fp32_to_bf16_sr_kernel(const float* __restrict__ src,
                                       uint16_t* __restrict__ dst, int64_t n,
                                       uint64_t seed, uint64_t offset) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  Philox4 ph(seed, (uint64_t)tid, offset);
  for (int64_t i0 = tid * 4; i0 < n; i0 += stride * 4) {
    const uint4 r = ph.next();
    const uint32_t rr[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int64_t i = i0 + j;
      if (i < n) {
        uint32_t bits = __float_as_uint(src[i]);
        // only round finite values; keep NaN/Inf bit patterns intact
        if ((bits & 0x7F800000u) != 0x7F800000u) bits += rr[j] & 0xFFFFu;
        dst[i] = (uint16_t)(bits >> 16);
      }
    }
  }
}

}  // namespace

void fp32_to_bf16_sr(at::Tensor src, at::Tensor dst) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous() && dst.is_contiguous(),
              "fp32_to_bf16_sr: tensors must be contiguous CUDA");
  TORCH_CHECK(src.scalar_type() == at::kFloat && dst.scalar_type() == at::kBFloat16,
              "fp32_to_bf16_sr: expected fp32 src, bf16 dst");
  const int64_t n = src.numel();
  TORCH_CHECK(dst.numel() == n, "fp32_to_bf16_sr: size mismatch");
  if (n == 0) return;
  auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
      std::nullopt, at::cuda::detail::getDefaultCUDAGenerator());
  at::PhiloxCudaState state;
  {
    std::lock_guard<std::mutex> lock(gen->mutex_);
    // each thread draws ceil(n / (4 * total_threads)) counters; bound by
    // the grid-stride trip count
    state = gen->philox_cuda_state(4 + n / (2048LL * 256 * 4));
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = unicore_grid((n / 4 + 255) / 256);
  fp32_to_bf16_sr_kernel<<<grid, 256, 0, stream>>>(
      src.data_ptr<float>(), reinterpret_cast<uint16_t*>(dst.data_ptr()), n,
      state.seed_.val, state.offset_.val);
  C10_CUDA_KERNEL_LAUNCH_CHECK();
}
