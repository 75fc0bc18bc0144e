// MFMA plumbing probe: one-wave 16x16x32 bf16 matrix product.
//
// Verifies the lane<->fragment element mappings this codebase assumes for
// v_mfma_f32_16x16x32_bf16 (used by the flash-attention kernels):
//   A (16x32): lane l holds row i = l & 15, cols k = (l >> 4)*8 + [0..8)
//   B (32x16): lane l holds col j = l & 15, rows k = (l >> 4)*8 + [0..8)
//   C/D (16x16): lane l holds col = l & 15, rows (l >> 4)*4 + [0..4)
// Exposed to Python as a unit-testable primitive (asymmetric-input test
// catches any transpose slip).
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

__global__ void mfma_probe_kernel(const uint16_t* __restrict__ A,
                                  const uint16_t* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x;
  bf16x8 a, b;
  const int g = lane >> 4;
  const int r = lane & 15;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[r * 32 + g * 8 + j];      // A[i=r][k=g*8+j]
    b[j] = (short)B[(g * 8 + j) * 16 + r];    // B[k=g*8+j][j=r]
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) D[(g * 4 + j) * 16 + r] = c[j];
}

}  // namespace

at::Tensor mfma_gemm_16x16x32(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
                  A.is_contiguous() && A.sizes() == at::IntArrayRef({16, 32}),
              "A must be bf16 (16, 32) CUDA");
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == at::kBFloat16 &&
                  B.is_contiguous() && B.sizes() == at::IntArrayRef({32, 16}),
              "B must be bf16 (32, 16) CUDA");
  auto D = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  mfma_probe_kernel<<<1, 64, 0, stream>>>(
      reinterpret_cast<const uint16_t*>(A.data_ptr()),
      reinterpret_cast<const uint16_t*>(B.data_ptr()), D.data_ptr<float>());
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return D;
}
