// Fused Gaussian pair-bias basis for gfx950 (CDNA4).
//
// Computes the Uni-Mol-style pairwise gaussian features directly from
// coordinates:  g[b,i,j,k] = exp(-0.5 * ((|c_i - c_j| - mean_k) * inv_k)^2),
// inv_k = 1 / (|std_k| + 1e-3), in one kernel each way.  The eager chain
// (torch.cdist fwd+bwd + five (B,L,L,K) elementwise passes + fp32
// intermediates) is the dominant non-GEMM cost of the mol_pairbias model;
// here distances are recomputed from the (B,L,3) coords (12 B/row, L2
// resident) instead of ever materialising them, so HBM traffic is one
// (B,L,L,K) write forward and one read backward.
//
// Backward is deterministic (no global atomics):
//   kernel 1: dg -> dd (B,L,L distance grads, fp32) + per-block fp32
//             partials of d_mean/d_std (register accumulation, wave
//             shfl reduction, LDS fold across waves)
//   kernel 2: fold partials -> d_means, d_stds
//   kernel 3: dd -> d_coords, one block per (b,i) row summing the i-th
//             row and column of dd (matches torch.cdist's subgradient:
//             zero contribution where dist == 0).
//
// Parity: behavioural counterpart of the gaussian layer used by the
// reference's flagship downstream (Uni-Mol); the reference itself leaves
// this to eager torch.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

namespace {

#define DISPATCH_OUT_FTYPES(st, NAME, ...)                           \
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

__device__ __forceinline__ float pair_dist(const float* __restrict__ coords,
                                           int64_t b, int i, int j, int L,
                                           float& dx, float& dy, float& dz) {
  const float* ci = coords + (b * L + i) * 3;
  const float* cj = coords + (b * L + j) * 3;
  dx = ci[0] - cj[0];
  dy = ci[1] - cj[1];
  dz = ci[2] - cj[2];
  return sqrtf(dx * dx + dy * dy + dz * dz);
}

// one thread per (pair, group of 8 k); threads of a pair are consecutive
// lanes so the 16 B stores of a pair's row coalesce.
template <typename OT>
__global__ void gaussian_fwd_kernel(const float* __restrict__ coords,
                                    const float* __restrict__ means,
                                    const float* __restrict__ stds,
                                    OT* __restrict__ out, int64_t n_pairs,
                                    int L, int K) {
  extern __shared__ float smem[];  // [K] mean, [K] inv_std
  float* s_mean = smem;
  float* s_inv = smem + K;
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    s_mean[k] = means[k];
    s_inv[k] = 1.0f / (fabsf(stds[k]) + 1e-3f);
  }
  __syncthreads();

  const int tpp = K / 8;  // threads per pair
  const int64_t total = n_pairs * tpp;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const int64_t pair = idx / tpp;
    const int k0 = (int)(idx % tpp) * 8;
    const int64_t b = pair / ((int64_t)L * L);
    const int64_t rem = pair - b * (int64_t)L * L;
    const int i = (int)(rem / L), j = (int)(rem % L);
    float dx, dy, dz;
    const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);
    float v[8];
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      const float e = (dist - s_mean[k0 + t]) * s_inv[k0 + t];
      v[t] = __expf(-0.5f * e * e);
    }
    store8(out + pair * K + k0, v);
  }
}

// dg -> dd + per-block d_mean/d_std partials.  Each thread keeps a fixed
// k-group across the grid-stride loop (stride is a multiple of tpp), so
// d_mean/d_std accumulate in registers and reduce deterministically:
// shfl over same-k lanes within the wave, LDS fold across the block's
// waves.  dd is reduced over the tpp lanes of each pair with shfl.
template <typename OT>
__global__ void gaussian_bwd_dd_kernel(const OT* __restrict__ dg,
                                       const float* __restrict__ coords,
                                       const float* __restrict__ means,
                                       const float* __restrict__ stds,
                                       float* __restrict__ dd,
                                       float* __restrict__ partials,
                                       int64_t n_pairs, int L, int K) {
  extern __shared__ float smem[];
  float* s_mean = smem;            // [K]
  float* s_inv = smem + K;         // [K]
  float* s_dinv_ds = smem + 2 * K; // [K]  d inv / d std_param
  float* s_red = smem + 3 * K;     // [4][2K] per-wave fold
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    const float s = stds[k];
    const float inv = 1.0f / (fabsf(s) + 1e-3f);
    s_mean[k] = means[k];
    s_inv[k] = inv;
    const float sgn = (s > 0.f) ? 1.f : ((s < 0.f) ? -1.f : 0.f);
    s_dinv_ds[k] = -sgn * inv * inv;
  }
  __syncthreads();

  const int tpp = K / 8;
  const int64_t total = n_pairs * tpp;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int k0 = (int)(((int64_t)blockIdx.x * blockDim.x + threadIdx.x) % tpp) * 8;

  float acc_dm[8], acc_ds[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) acc_dm[t] = acc_ds[t] = 0.f;

  // uniform trip count: every lane of a wave reaches the shfl reduction
  // even on the ragged tail (inactive lanes contribute zeros)
  for (int64_t base = (int64_t)blockIdx.x * blockDim.x; base < total;
       base += stride) {
    const int64_t idx = base + threadIdx.x;
    const bool active = idx < total;
    const int64_t pair = (active ? idx : total - 1) / tpp;
    const int64_t b = pair / ((int64_t)L * L);
    const int64_t rem = pair - b * (int64_t)L * L;
    const int i = (int)(rem / L), j = (int)(rem % L);
    float dx, dy, dz;
    const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);

    float go[8];
    if (active) {
      load8(dg + pair * K + k0, go);
    } else {
#pragma unroll
      for (int t = 0; t < 8; ++t) go[t] = 0.f;
    }
    float d_dist = 0.f;
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      const float m = s_mean[k0 + t], inv = s_inv[k0 + t];
      const float e = (dist - m) * inv;
      const float g = __expf(-0.5f * e * e);
      const float d_e = go[t] * g * (-e);
      d_dist += d_e * inv;
      acc_dm[t] -= d_e * inv;                        // de/dm = -inv
      acc_ds[t] += d_e * (dist - m) * s_dinv_ds[k0 + t];
    }
    // sum d_dist over the pair's tpp consecutive (aligned) lanes
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      if (off < tpp) d_dist += __shfl_xor(d_dist, off, 64);
    }
    if (active && (int)(idx % tpp) == 0) dd[pair] = d_dist;
  }

  // deterministic block fold of acc_dm/acc_ds (same-k lanes are tpp apart)
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
#pragma unroll
  for (int t = 0; t < 8; ++t) {
    for (int off = tpp; off < 64; off <<= 1) {
      acc_dm[t] += __shfl_xor(acc_dm[t], off, 64);
      acc_ds[t] += __shfl_xor(acc_ds[t], off, 64);
    }
  }
  if (lane < tpp) {
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      s_red[wave * 2 * K + (k0 + t)] = acc_dm[t];
      s_red[wave * 2 * K + K + (k0 + t)] = acc_ds[t];
    }
  }
  __syncthreads();
  const int nwaves = blockDim.x / 64;
  for (int c = threadIdx.x; c < 2 * K; c += blockDim.x) {
    float v = 0.f;
    for (int w = 0; w < nwaves; ++w) v += s_red[w * 2 * K + c];
    partials[(int64_t)blockIdx.x * 2 * K + c] = v;
  }
}

// partials (nb, C) -> out (C): one block per column, threads stride the
// rows, deterministic wave+LDS tree (nb can be the full 2048-block grid;
// a single serial block here cost 479 us at the mol shape).
__global__ void col_fold_kernel(const float* __restrict__ partials,
                                float* __restrict__ out, int nb, int C) {
  __shared__ float s_red[4];
  const int c = blockIdx.x;
  if (c >= C) return;
  float v = 0.f;
  for (int r = threadIdx.x; r < nb; r += blockDim.x)
    v += partials[(int64_t)r * C + c];
  v = wave_sum(v);
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  if (lane == 0) s_red[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = 0.f;
    const int nwaves = blockDim.x / 64;
    for (int w = 0; w < nwaves; ++w) r += s_red[w];
    out[c] = r;
  }
}

// dd -> d_coords.  One block per (b,i):
//   d_coords[b,i] = sum_j (dd[b,i,j] + dd[b,j,i]) * (c_i - c_j) / dist_ij
// (zero where dist == 0, matching cdist's subgradient).  Row loads
// coalesce; column loads hit L2 (dd is fp32 (B,L,L)).
__global__ void gaussian_dcoords_kernel(const float* __restrict__ dd,
                                        const float* __restrict__ coords,
                                        float* __restrict__ d_coords,
                                        int64_t n_rows, int L) {
  __shared__ float s_red[4][3];
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const int64_t b = row / L;
    const int i = (int)(row - b * L);
    float ax = 0.f, ay = 0.f, az = 0.f;
    for (int j = threadIdx.x; j < L; j += blockDim.x) {
      const float w = dd[(b * L + i) * L + j] + dd[(b * L + j) * L + i];
      float dx, dy, dz;
      const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);
      if (dist > 0.f) {
        const float sc = w / dist;
        ax += sc * dx;
        ay += sc * dy;
        az += sc * dz;
      }
    }
    ax = wave_sum(ax);
    ay = wave_sum(ay);
    az = wave_sum(az);
    const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
    if (lane == 0) {
      s_red[wave][0] = ax;
      s_red[wave][1] = ay;
      s_red[wave][2] = az;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      const int nwaves = blockDim.x / 64;
      float rx = 0.f, ry = 0.f, rz = 0.f;
      for (int w = 0; w < nwaves; ++w) {
        rx += s_red[w][0];
        ry += s_red[w][1];
        rz += s_red[w][2];
      }
      d_coords[row * 3 + 0] = rx;
      d_coords[row * 3 + 1] = ry;
      d_coords[row * 3 + 2] = rz;
    }
    __syncthreads();
  }
}

bool gaussian_k_supported(int64_t K) {
  if (K < 8 || K % 8 != 0) return false;
  const int64_t tpp = K / 8;
  return (tpp & (tpp - 1)) == 0 && tpp <= 64;  // power of two, <= one wave
}

// ---------------------------------------------------------------------------
// Fully-fused gaussian pair bias: coords -> (B, H, L, L) attention bias.
// Folds the K->H output Linear, the (B,L,L,H)->(B,H,L,L) permute and the
// padding-key masked_fill into the basis kernel, so the (B,L,L,K) feature
// tensor never exists and the pathological skinny wgrad GEMM
// ((B*L*L, K)^T @ (B*L*L, H), 2.1 ms via hipBLASLt at the mol shape) is
// replaced by in-kernel register accumulation + the deterministic fold.
// ---------------------------------------------------------------------------

// One thread per pair: the K-loop is serial, so every LDS read is a
// wave-wide broadcast (conflict-free), there is no cross-lane reduction,
// and the H stores of a wave hit 64 consecutive j per head plane
// (coalesced).  ~50 VGPRs -> full occupancy.
template <typename OT, int H>
__global__ void gaussian_pair_bias_fwd_kernel(
    const float* __restrict__ coords, const float* __restrict__ means,
    const float* __restrict__ stds, const float* __restrict__ W,  // (H, K)
    const float* __restrict__ bvec,                               // (H)
    const uint8_t* __restrict__ pad,                              // (B, L) | null
    OT* __restrict__ out, float fill, int64_t n_pairs, int L, int K) {
  extern __shared__ float smem[];
  float* s_mean = smem;          // [K]
  float* s_inv = smem + K;       // [K]
  float* s_w = smem + 2 * K;     // [H*K]
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    s_mean[k] = means[k];
    s_inv[k] = 1.0f / (fabsf(stds[k]) + 1e-3f);
  }
  for (int k = threadIdx.x; k < H * K; k += blockDim.x) s_w[k] = W[k];
  __syncthreads();

  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t pair = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       pair < n_pairs; pair += stride) {
    const int64_t b = pair / ((int64_t)L * L);
    const int64_t rem = pair - b * (int64_t)L * L;
    const int i = (int)(rem / L), j = (int)(rem % L);
    float dx, dy, dz;
    const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);
    float acc[H];
#pragma unroll
    for (int h = 0; h < H; ++h) acc[h] = bvec[h];
    for (int k8 = 0; k8 < K; k8 += 8) {
      float g[8];
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const float e = (dist - s_mean[k8 + t]) * s_inv[k8 + t];
        g[t] = __expf(-0.5f * e * e);
      }
#pragma unroll
      for (int h = 0; h < H; ++h) {
        float a = 0.f;
#pragma unroll
        for (int t = 0; t < 8; ++t) a += g[t] * s_w[h * K + k8 + t];
        acc[h] += a;
      }
    }
    const bool masked = (pad != nullptr && pad[b * L + j]);
#pragma unroll
    for (int h = 0; h < H; ++h) {
      const float v = masked ? fill : acc[h];
      out[((b * H + h) * (int64_t)L + i) * L + j] = Cvt<OT>::from_f(v);
    }
  }
}

// Backward is split so neither kernel spills registers (a merged
// version spilled 173 VGPRs -> 544 B of per-thread scratch and ran 10x
// slow):
//   A: dd (distance grads), one thread per pair, broadcast LDS reads
//   B: dW/d_mean/d_std/db per-block partials, 4-k-slice lanes so the dW
//      accumulator is 32 registers instead of 64.
template <typename OT, int H>
__global__ void gaussian_pair_bias_bwd_dd_kernel(
    const OT* __restrict__ dbias, const float* __restrict__ coords,
    const float* __restrict__ means, const float* __restrict__ stds,
    const float* __restrict__ W, const uint8_t* __restrict__ pad,
    float* __restrict__ dd, int64_t n_pairs, int L, int K) {
  extern __shared__ float smem[];
  float* s_mean = smem;       // [K]
  float* s_inv = smem + K;    // [K]
  float* s_w = smem + 2 * K;  // [H*K]
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    s_mean[k] = means[k];
    s_inv[k] = 1.0f / (fabsf(stds[k]) + 1e-3f);
  }
  for (int k = threadIdx.x; k < H * K; k += blockDim.x) s_w[k] = W[k];
  __syncthreads();

  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t pair = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       pair < n_pairs; pair += stride) {
    const int64_t b = pair / ((int64_t)L * L);
    const int64_t rem = pair - b * (int64_t)L * L;
    const int i = (int)(rem / L), j = (int)(rem % L);
    if (pad != nullptr && pad[b * L + j]) {
      dd[pair] = 0.f;
      continue;
    }
    float db[H];
#pragma unroll
    for (int h = 0; h < H; ++h)
      db[h] = Cvt<OT>::to_f(dbias[((b * H + h) * (int64_t)L + i) * L + j]);
    float dx, dy, dz;
    const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);
    float d_dist = 0.f;
    for (int k8 = 0; k8 < K; k8 += 8) {
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const float m = s_mean[k8 + t], inv = s_inv[k8 + t];
        const float e = (dist - m) * inv;
        const float g = __expf(-0.5f * e * e);
        float dg = 0.f;
#pragma unroll
        for (int h = 0; h < H; ++h) dg += db[h] * s_w[h * K + k8 + t];
        d_dist += dg * g * (-e) * inv;
      }
    }
    dd[pair] = d_dist;
  }
}

template <typename OT, int H>
__global__ void gaussian_pair_bias_bwd_part_kernel(
    const OT* __restrict__ dbias, const float* __restrict__ coords,
    const float* __restrict__ means, const float* __restrict__ stds,
    const float* __restrict__ W, const uint8_t* __restrict__ pad,
    float* __restrict__ partials, int64_t n_pairs, int L, int K) {
  const int C = H * K + 2 * K + H;
  extern __shared__ float smem[];
  float* s_mean = smem;               // [K]
  float* s_inv = smem + K;            // [K]
  float* s_dinv_ds = smem + 2 * K;    // [K]
  float* s_w = smem + 3 * K;          // [H*K]
  float* s_red = smem + 3 * K + H * K;  // [nwaves][C]
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    const float s = stds[k];
    const float inv = 1.0f / (fabsf(s) + 1e-3f);
    s_mean[k] = means[k];
    s_inv[k] = inv;
    const float sgn = (s > 0.f) ? 1.f : ((s < 0.f) ? -1.f : 0.f);
    s_dinv_ds[k] = -sgn * inv * inv;
  }
  for (int k = threadIdx.x; k < H * K; k += blockDim.x) s_w[k] = W[k];
  __syncthreads();

  const int tpp = K / 4;  // 4-k slice keeps acc_dw at 32 registers
  const int64_t total = n_pairs * tpp;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int k0 = (int)(((int64_t)blockIdx.x * blockDim.x + threadIdx.x) % tpp) * 4;

  float acc_dm[4], acc_ds[4], acc_dw[H][4], acc_db[H];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc_dm[t] = acc_ds[t] = 0.f;
#pragma unroll
  for (int h = 0; h < H; ++h) {
    acc_db[h] = 0.f;
#pragma unroll
    for (int t = 0; t < 4; ++t) acc_dw[h][t] = 0.f;
  }

  // uniform loop: the shfl broadcast of db below needs whole-wave
  // participation (pair groups are tpp-aligned within the wave)
  for (int64_t base = (int64_t)blockIdx.x * blockDim.x; base < total;
       base += stride) {
    const int64_t idx = base + threadIdx.x;
    const bool active = idx < total;
    const int64_t pair = (active ? idx : total - 1) / tpp;
    const int64_t b = pair / ((int64_t)L * L);
    const int64_t rem = pair - b * (int64_t)L * L;
    const int i = (int)(rem / L), j = (int)(rem % L);
    const bool masked =
        !active || (pad != nullptr && pad[b * L + j]);
    const int lane_in_pair = (int)((threadIdx.x % 64) % tpp);
    float db[H];
    if (tpp >= H) {
      // lane h of each pair group loads head h once, then broadcasts
      float mine = 0.f;
      if (!masked && lane_in_pair < H)
        mine = Cvt<OT>::to_f(
            dbias[((b * H + lane_in_pair) * (int64_t)L + i) * L + j]);
      const int base_lane = (int)(threadIdx.x % 64) - lane_in_pair;
#pragma unroll
      for (int h = 0; h < H; ++h) db[h] = __shfl(mine, base_lane + h, 64);
      if (masked) {
#pragma unroll
        for (int h = 0; h < H; ++h) db[h] = 0.f;
      }
    } else {
#pragma unroll
      for (int h = 0; h < H; ++h)
        db[h] = masked ? 0.f
                       : Cvt<OT>::to_f(
                             dbias[((b * H + h) * (int64_t)L + i) * L + j]);
    }
    if (active && (int)(idx % tpp) == 0) {
#pragma unroll
      for (int h = 0; h < H; ++h) acc_db[h] += db[h];
    }
    if (masked) continue;
    float dx, dy, dz;
    const float dist = pair_dist(coords, b, i, j, L, dx, dy, dz);
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const float m = s_mean[k0 + t], inv = s_inv[k0 + t];
      const float e = (dist - m) * inv;
      const float g = __expf(-0.5f * e * e);
      float dg = 0.f;
#pragma unroll
      for (int h = 0; h < H; ++h) {
        dg += db[h] * s_w[h * K + k0 + t];
        acc_dw[h][t] += db[h] * g;
      }
      const float d_e = dg * g * (-e);
      acc_dm[t] -= d_e * inv;
      acc_ds[t] += d_e * (dist - m) * s_dinv_ds[k0 + t];
    }
  }

  // deterministic wave reduction (same-k lanes tpp apart; db over all lanes)
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    for (int off = tpp; off < 64; off <<= 1) {
      acc_dm[t] += __shfl_xor(acc_dm[t], off, 64);
      acc_ds[t] += __shfl_xor(acc_ds[t], off, 64);
#pragma unroll
      for (int h = 0; h < H; ++h)
        acc_dw[h][t] += __shfl_xor(acc_dw[h][t], off, 64);
    }
  }
#pragma unroll
  for (int h = 0; h < H; ++h) {
    for (int off = 1; off < 64; off <<= 1)
      acc_db[h] += __shfl_xor(acc_db[h], off, 64);
  }
  float* my = s_red + wave * C;
  if (lane < tpp && lane * 4 < K) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int h = 0; h < H; ++h) my[h * K + k0 + t] = acc_dw[h][t];
      my[H * K + k0 + t] = acc_dm[t];
      my[H * K + K + k0 + t] = acc_ds[t];
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int h = 0; h < H; ++h) my[H * K + 2 * K + h] = acc_db[h];
  }
  __syncthreads();
  const int nwaves = blockDim.x / 64;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float v = 0.f;
    for (int w = 0; w < nwaves; ++w) v += s_red[w * C + c];
    partials[(int64_t)blockIdx.x * C + c] = v;
  }
}

bool gaussian_h_supported(int64_t H) {
  // H=16 would spill ~90 VGPRs in the partials kernel; those shapes take
  // the (still fused-basis) fallback path instead
  return H == 4 || H == 8;
}

}  // namespace

torch::Tensor gaussian_basis_forward(torch::Tensor coords,
                                     torch::Tensor means, torch::Tensor stds,
                                     at::ScalarType out_dtype) {
  TORCH_CHECK(coords.is_cuda() && coords.dim() == 3 && coords.size(2) == 3,
              "gaussian: coords must be CUDA (B, L, 3)");
  TORCH_CHECK(coords.scalar_type() == at::ScalarType::Float &&
                  means.scalar_type() == at::ScalarType::Float &&
                  stds.scalar_type() == at::ScalarType::Float,
              "gaussian: coords/means/stds must be fp32");
  const int64_t B = coords.size(0), L = coords.size(1), K = means.numel();
  TORCH_CHECK(gaussian_k_supported(K), "gaussian: unsupported K ", K);
  auto cc = coords.contiguous();
  auto mc = means.contiguous();
  auto sc = stds.contiguous();
  auto out = torch::empty({B, L, L, K}, coords.options().dtype(out_dtype));
  const int64_t n_pairs = B * L * L;
  const int block = 256;
  const int grid = unicore_grid((n_pairs * (K / 8) + block - 1) / block);
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_OUT_FTYPES(out_dtype, "gaussian_fwd", {
    gaussian_fwd_kernel<scalar_t><<<grid, block, 2 * K * sizeof(float),
                                    stream>>>(
        cc.data_ptr<float>(), mc.data_ptr<float>(), sc.data_ptr<float>(),
        reinterpret_cast<scalar_t*>(out.data_ptr()), n_pairs, (int)L, (int)K);
  });
  return out;
}

std::vector<torch::Tensor> gaussian_basis_backward(torch::Tensor dg,
                                                   torch::Tensor coords,
                                                   torch::Tensor means,
                                                   torch::Tensor stds) {
  const int64_t B = coords.size(0), L = coords.size(1), K = means.numel();
  TORCH_CHECK(dg.is_cuda() && dg.dim() == 4 && dg.size(3) == K,
              "gaussian: bad grad shape");
  auto gc = dg.contiguous();
  auto cc = coords.contiguous();
  auto mc = means.contiguous();
  auto sc = stds.contiguous();
  const int64_t n_pairs = B * L * L;
  auto dd = torch::empty({B, L, L}, coords.options());
  auto d_coords = torch::empty_like(cc);
  const int block = 256;
  const int grid = unicore_grid((n_pairs * (K / 8) + block - 1) / block);
  auto partials = torch::empty({grid, 2 * K}, coords.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const size_t lds = (3 * K + (block / 64) * 2 * K) * sizeof(float);
  DISPATCH_OUT_FTYPES(gc.scalar_type(), "gaussian_bwd", {
    gaussian_bwd_dd_kernel<scalar_t><<<grid, block, lds, stream>>>(
        reinterpret_cast<const scalar_t*>(gc.data_ptr()),
        cc.data_ptr<float>(), mc.data_ptr<float>(), sc.data_ptr<float>(),
        dd.data_ptr<float>(), partials.data_ptr<float>(), n_pairs, (int)L,
        (int)K);
  });
  auto fold = torch::empty({2 * K}, coords.options());
  col_fold_kernel<<<2 * (int)K, block, 0, stream>>>(
      partials.data_ptr<float>(), fold.data_ptr<float>(), grid, 2 * (int)K);
  const int64_t n_rows = B * L;
  gaussian_dcoords_kernel<<<unicore_grid(n_rows), block, 0, stream>>>(
      dd.data_ptr<float>(), cc.data_ptr<float>(), d_coords.data_ptr<float>(),
      n_rows, (int)L);
  return {d_coords, fold.narrow(0, 0, K), fold.narrow(0, K, K)};
}

bool gaussian_basis_supported(int64_t K) { return gaussian_k_supported(K); }

static size_t pair_bias_bwd_lds(int64_t K, int64_t H) {
  const int64_t C = H * K + 2 * K + H;
  return (size_t)(3 * K + H * K + 4 * C) * sizeof(float);
}

bool gaussian_pair_bias_supported(int64_t K, int64_t H) {
  // K%8 for the 8-wide fwd chunks; K/4 a power of two <= 64 for the
  // partials kernel's lane slicing
  if (K < 8 || K % 8 != 0) return false;
  const int64_t tpp = K / 4;
  if ((tpp & (tpp - 1)) != 0 || tpp > 64) return false;
  return gaussian_h_supported(H) && pair_bias_bwd_lds(K, H) <= 64 * 1024;
}

#define DISPATCH_H(H, ...)                         \
  switch (H) {                                     \
    case 4: {                                      \
      constexpr int kH = 4;                        \
      __VA_ARGS__;                                 \
      break;                                       \
    }                                              \
    case 8: {                                      \
      constexpr int kH = 8;                        \
      __VA_ARGS__;                                 \
      break;                                       \
    }                                              \
    default:                                       \
      TORCH_CHECK(false, "gaussian_pair_bias: unsupported H ", H); \
  }

torch::Tensor gaussian_pair_bias_forward(torch::Tensor coords,
                                         torch::Tensor means,
                                         torch::Tensor stds, torch::Tensor W,
                                         torch::Tensor bvec,
                                         std::optional<torch::Tensor> pad,
                                         double fill,
                                         at::ScalarType out_dtype) {
  TORCH_CHECK(coords.is_cuda() && coords.dim() == 3 && coords.size(2) == 3,
              "gaussian_pair_bias: coords must be CUDA (B, L, 3)");
  const int64_t B = coords.size(0), L = coords.size(1), K = means.numel();
  const int64_t H = bvec.numel();
  TORCH_CHECK(W.dim() == 2 && W.size(0) == H && W.size(1) == K,
              "gaussian_pair_bias: W must be (H, K)");
  TORCH_CHECK(gaussian_pair_bias_supported(K, H),
              "gaussian_pair_bias: unsupported K/H ", K, "/", H);
  auto cc = coords.contiguous();
  auto mc = means.contiguous().to(at::kFloat);
  auto sc = stds.contiguous().to(at::kFloat);
  auto wc = W.contiguous().to(at::kFloat);
  auto bc = bvec.contiguous().to(at::kFloat);
  const uint8_t* pad_ptr = nullptr;
  torch::Tensor padc;
  if (pad.has_value()) {
    padc = pad->contiguous().to(at::kBool);
    TORCH_CHECK(padc.numel() == B * L, "gaussian_pair_bias: pad must be (B, L)");
    pad_ptr = (const uint8_t*)padc.data_ptr<bool>();
  }
  auto out = torch::empty({B, H, L, L}, coords.options().dtype(out_dtype));
  const int64_t n_pairs = B * L * L;
  const int block = 256;
  const int grid = unicore_grid((n_pairs * (K / 8) + block - 1) / block);
  const size_t lds = (size_t)(2 * K + H * K) * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_OUT_FTYPES(out_dtype, "gaussian_pair_bias_fwd", {
    DISPATCH_H(H, {
      gaussian_pair_bias_fwd_kernel<scalar_t, kH><<<grid, block, lds, stream>>>(
          cc.data_ptr<float>(), mc.data_ptr<float>(), sc.data_ptr<float>(),
          wc.data_ptr<float>(), bc.data_ptr<float>(), pad_ptr,
          reinterpret_cast<scalar_t*>(out.data_ptr()), (float)fill, n_pairs,
          (int)L, (int)K);
    });
  });
  return out;
}

std::vector<torch::Tensor> gaussian_pair_bias_backward(
    torch::Tensor dbias, torch::Tensor coords, torch::Tensor means,
    torch::Tensor stds, torch::Tensor W,
    std::optional<torch::Tensor> pad) {
  const int64_t B = coords.size(0), L = coords.size(1), K = means.numel();
  const int64_t H = W.size(0);
  TORCH_CHECK(dbias.is_cuda() && dbias.dim() == 4 && dbias.size(1) == H &&
                  dbias.size(2) == L && dbias.size(3) == L,
              "gaussian_pair_bias: bad grad shape");
  auto gc = dbias.contiguous();
  auto cc = coords.contiguous();
  auto mc = means.contiguous().to(at::kFloat);
  auto sc = stds.contiguous().to(at::kFloat);
  auto wc = W.contiguous().to(at::kFloat);
  const uint8_t* pad_ptr = nullptr;
  torch::Tensor padc;
  if (pad.has_value()) {
    padc = pad->contiguous().to(at::kBool);
    pad_ptr = (const uint8_t*)padc.data_ptr<bool>();
  }
  const int64_t n_pairs = B * L * L;
  const int64_t C = H * K + 2 * K + H;
  auto fopts = coords.options().dtype(at::kFloat);
  auto dd = torch::empty({B, L, L}, fopts);
  auto d_coords = torch::empty({B, L, 3}, fopts);
  const int block = 256;
  const int grid_dd = unicore_grid((n_pairs + block - 1) / block);
  const int grid = unicore_grid((n_pairs * (K / 4) + block - 1) / block);
  auto partials = torch::empty({grid, C}, fopts);
  auto fold = torch::empty({C}, fopts);
  auto stream = at::cuda::getCurrentCUDAStream();
  const size_t lds_dd = (size_t)(2 * K + H * K) * sizeof(float);
  const size_t lds = pair_bias_bwd_lds(K, H);
  DISPATCH_OUT_FTYPES(gc.scalar_type(), "gaussian_pair_bias_bwd", {
    DISPATCH_H(H, {
      gaussian_pair_bias_bwd_dd_kernel<scalar_t, kH>
          <<<grid_dd, block, lds_dd, stream>>>(
              reinterpret_cast<const scalar_t*>(gc.data_ptr()),
              cc.data_ptr<float>(), mc.data_ptr<float>(), sc.data_ptr<float>(),
              wc.data_ptr<float>(), pad_ptr, dd.data_ptr<float>(), n_pairs,
              (int)L, (int)K);
      gaussian_pair_bias_bwd_part_kernel<scalar_t, kH>
          <<<grid, block, lds, stream>>>(
              reinterpret_cast<const scalar_t*>(gc.data_ptr()),
              cc.data_ptr<float>(), mc.data_ptr<float>(), sc.data_ptr<float>(),
              wc.data_ptr<float>(), pad_ptr, partials.data_ptr<float>(),
              n_pairs, (int)L, (int)K);
    });
  });
  col_fold_kernel<<<(int)C, block, 0, stream>>>(
      partials.data_ptr<float>(), fold.data_ptr<float>(), grid, (int)C);
  const int64_t n_rows = B * L;
  gaussian_dcoords_kernel<<<unicore_grid(n_rows), block, 0, stream>>>(
      dd.data_ptr<float>(), cc.data_ptr<float>(), d_coords.data_ptr<float>(),
      n_rows, (int)L);
  auto d_w = fold.narrow(0, 0, H * K).view({H, K});
  auto d_means = fold.narrow(0, H * K, K);
  auto d_stds = fold.narrow(0, H * K + K, K);
  auto d_b = fold.narrow(0, H * K + 2 * K, H);
  return {d_coords, d_means, d_stds, d_w, d_b};
}

