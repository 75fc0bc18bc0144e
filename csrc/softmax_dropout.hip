// Fused softmax(+attention-mask,+pair-bias)+dropout for gfx950 (CDNA4).
//
// Functional counterpart of the reference's softmax_dropout extension
// (reference csrc/softmax_dropout/*), re-derived for 64-lane wavefronts:
//
//  * vector path (k % 8 == 0, k <= 4096): one wave64 per row, the whole row
//    register-resident as NV x 8 floats per lane (16 B/lane loads), two
//    6-step __shfl_xor reductions (max, sum).  Dropout is fused: in-kernel
//    Philox4x32-10 keyed by PyTorch's generator (seed, subsequence =
//    row*64+lane, offset), and the keep-mask is stored as a bitfield —
//    one uint8 per 8-element vector, written by the owning lane.
//  * block path (any k, p == 0 on this path — the Python shim routes
//    wide/odd rows with dropout through torch dropout, reference
//    unicore/modules/softmax_dropout.py:131-138): 256-thread block per row,
//    3-pass (max / exp+sum / normalize), no LDS row staging needed.
//
// mask/bias broadcast contract (see unicore_amd/modules/softmax_dropout.py):
// source row = ((b / outer_div) % src_nb) * src_q + (q % src_q).
#include "common.h"

#include <cstdlib>

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

#include <optional>
#include <vector>

namespace {

template <typename T, int NV, int ROWS, bool DROP>
__global__ void softmax_fwd_vec_kernel(
    T* __restrict__ out, uint8_t* __restrict__ dmask, T* __restrict__ x,
    const T* __restrict__ amask, int64_t am_nb, int am_q, int64_t am_od,
    const T* __restrict__ bias, int64_t bs_nb, int bs_q, int64_t bs_od,
    int64_t n_rows, int q_len, int k, float pinv, uint32_t pthresh,
    uint64_t seed, uint64_t rng_offset, int64_t bias_major_outer) {
  // ROWS rows per wave per iteration: the row loads issue back-to-back so
  // several 16 B/lane requests are in flight (one row alone leaves the
  // memory system starved at high occupancy)
  //
  // bias_major_outer > 0 turns on the bias-major iteration order: the flat
  // index decomposes as it = (h*q_len + q) * outer + b, so the `outer`
  // consecutive iterations (and a wave's ROWS rows) all share ONE
  // (h, q) bias row — it stays hot in L1/L2 instead of being re-fetched
  // from HBM for all `outer` broadcast batches (~25-50%% of the kernel's
  // read traffic on the BERT shapes).
  const int lane = threadIdx.x;
  const int wid = threadIdx.y;
  const int mrow_bytes = k / 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.y * ROWS;
  for (int64_t row0 = ((int64_t)blockIdx.x * blockDim.y + wid) * ROWS;
       row0 < n_rows; row0 += stride) {
    float vals[ROWS][NV][8];
    float mx[ROWS], inv[ROWS];
    int64_t rows[ROWS];
    // phase 1: issue every row's loads (keeps several 16 B requests in
    // flight) and fold mask/bias + the per-lane max; cross-lane reductions
    // happen in phase 2 so no shuffle chain sits between two rows' loads
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      int64_t row = row0 + r;
      if (bias_major_outer > 0 && row < n_rows) {
        const int64_t hq = row / bias_major_outer;
        const int64_t b = row - hq * bias_major_outer;
        const int64_t h = hq / q_len;
        row = (b * bs_nb + h) * q_len + (hq - h * q_len);
      }
      rows[r] = row;
      float m = -INFINITY;
      if (row0 + r < n_rows) {
        T* xrow = x + row * (int64_t)k;
        const int64_t b = row / q_len;
        const int qi = (int)(row - b * q_len);
        const T* mrow = amask ? amask + (((b / am_od) % am_nb) * am_q +
                                         (qi % am_q)) * (int64_t)k
                              : nullptr;
        const T* brow = bias ? bias + (((b / bs_od) % bs_nb) * bs_q +
                                       (qi % bs_q)) * (int64_t)k
                             : nullptr;
#pragma unroll
        for (int i = 0; i < NV; ++i) {
          const int e0 = (lane + i * 64) * 8;
          if (e0 < k) {
            load8(xrow + e0, vals[r][i]);
            if (mrow) {
              float t[8];
              load8(mrow + e0, t);
#pragma unroll
              for (int j = 0; j < 8; ++j) vals[r][i][j] += t[j];
            }
            if (brow) {
              float t[8];
              load8(brow + e0, t);
#pragma unroll
              for (int j = 0; j < 8; ++j) vals[r][i][j] += t[j];
            }
#pragma unroll
            for (int j = 0; j < 8; ++j) m = fmaxf(m, vals[r][i][j]);
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j) vals[r][i][j] = -INFINITY;
          }
        }
      }
      mx[r] = m;
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) mx[r] = wave_max(mx[r]);
    float psum[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      float sum = 0.f;
      if (row0 + r < n_rows) {  // rows[] is a permutation; bound on the it
#pragma unroll
        for (int i = 0; i < NV; ++i)
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            vals[r][i][j] = __expf(vals[r][i][j] - mx[r]);
            sum += vals[r][i][j];
          }
      }
      psum[r] = sum;
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) inv[r] = 1.0f / wave_sum(psum[r]);
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      if (row0 + r >= n_rows) continue;
      const int64_t row = rows[r];
      T* xrow = x + row * (int64_t)k;
#pragma unroll
      for (int i = 0; i < NV; ++i) {
        const int e0 = (lane + i * 64) * 8;
        if (e0 < k) {
          float y[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) y[j] = vals[r][i][j] * inv[r];
          store8(xrow + e0, y);  // pre-dropout softmax, in-place over input
          if constexpr (DROP) {
            bool keep[8];
            keep16x8(seed, (uint64_t)row * 64 + lane, (lane + i * 64) * 8,
                     pthresh, keep);
            uint8_t bits = 0;
            float o[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              bits |= (uint8_t)(keep[j] ? 1u : 0u) << j;
              o[j] = keep[j] ? y[j] * pinv : 0.f;
            }
            dmask[row * (int64_t)mrow_bytes + lane + i * 64] = bits;
            store8(out + row * (int64_t)k + e0, o);
          }
        }
      }
    }
  }
}

template <typename T, int NV, int ROWS, bool DROP>
__global__ void softmax_bwd_vec_kernel(T* __restrict__ g, const T* __restrict__ y,
                                       const uint8_t* __restrict__ dmask,
                                       int64_t n_rows, int k, float pinv) {
  const int lane = threadIdx.x;
  const int wid = threadIdx.y;
  const int mrow_bytes = k / 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.y * ROWS;
  for (int64_t row0 = ((int64_t)blockIdx.x * blockDim.y + wid) * ROWS;
       row0 < n_rows; row0 += stride) {
    float tv[ROWS][NV][8], yv[ROWS][NV][8];
    float sr[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      const int64_t row = row0 + r;
      float ssum = 0.f;
      if (row < n_rows) {
        T* grow = g + row * (int64_t)k;
        const T* yrow = y + row * (int64_t)k;
#pragma unroll
        for (int i = 0; i < NV; ++i) {
          const int e0 = (lane + i * 64) * 8;
          if (e0 < k) {
            load8(grow + e0, tv[r][i]);
            load8(yrow + e0, yv[r][i]);
            if constexpr (DROP) {
              const uint8_t bits =
                  dmask[row * (int64_t)mrow_bytes + lane + i * 64];
#pragma unroll
              for (int j = 0; j < 8; ++j)
                tv[r][i][j] = (bits >> j) & 1 ? tv[r][i][j] * pinv : 0.f;
            }
#pragma unroll
            for (int j = 0; j < 8; ++j) ssum += tv[r][i][j] * yv[r][i][j];
          }
        }
      }
      sr[r] = ssum;
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) sr[r] = wave_sum(sr[r]);
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      const int64_t row = row0 + r;
      if (row >= n_rows) continue;
      T* grow = g + row * (int64_t)k;
#pragma unroll
      for (int i = 0; i < NV; ++i) {
        const int e0 = (lane + i * 64) * 8;
        if (e0 < k) {
          float dx[8];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            dx[j] = yv[r][i][j] * (tv[r][i][j] - sr[r]);
          store8(grow + e0, dx);
        }
      }
    }
  }
}

// Backward with the broadcast-bias gradient fused in: one block (4 waves)
// per bias row; the waves split the rows that reduce onto it (for BERT's
// (1,H,L,L) rel-pos bias that is the batch dimension), each wave computes
// those rows' grad_input in place (identical math to the vec kernel) while
// accumulating the bias-row sum in registers; a deterministic LDS fold
// writes the fp32 bias grad.  Replaces the eager `.sum(dim=(0,2))` that
// re-read the full (B,H,L,L) grad tensor (604 MB/layer at BERT-base).
template <typename T, int NV, bool DROP>
__global__ void softmax_bwd_biasgrad_kernel(
    T* __restrict__ g, const T* __restrict__ y,
    const uint8_t* __restrict__ dmask, float* __restrict__ dbias,
    int k, float pinv, int q, int bb, int bq, int od, int a) {
  extern __shared__ float s_red[];  // [4][k]
  const int64_t brow = blockIdx.x;  // 0 .. bb*bq-1
  const int jb = (int)(brow / bq);
  const int qb = (int)(brow % bq);
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int mrow_bytes = k / 8;
  const int qrep = q / bq;
  const int64_t group = (int64_t)a * od * qrep;

  float acc[NV][8];
#pragma unroll
  for (int i = 0; i < NV; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = 0.f;

  for (int64_t gidx = wid; gidx < group; gidx += 4) {
    const int64_t t = gidx / ((int64_t)od * qrep);
    const int64_t rem = gidx - t * od * qrep;
    const int e = (int)(rem / qrep);
    const int m = (int)(rem - (int64_t)e * qrep);
    const int64_t rb = ((int64_t)t * bb + jb) * od + e;
    const int64_t row = rb * q + qb + (int64_t)m * bq;

    T* grow = g + row * (int64_t)k;
    const T* yrow = y + row * (int64_t)k;
    float tv[NV][8], yv[NV][8];
    float ssum = 0.f;
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < k) {
        load8(grow + e0, tv[i]);
        load8(yrow + e0, yv[i]);
        if constexpr (DROP) {
          const uint8_t bits = dmask[row * (int64_t)mrow_bytes + lane + i * 64];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            tv[i][j] = (bits >> j) & 1 ? tv[i][j] * pinv : 0.f;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) ssum += tv[i][j] * yv[i][j];
      }
    }
    ssum = wave_sum(ssum);
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int e0 = (lane + i * 64) * 8;
      if (e0 < k) {
        float dx[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dx[j] = yv[i][j] * (tv[i][j] - ssum);
          acc[i][j] += dx[j];
        }
        store8(grow + e0, dx);
      }
    }
  }

  float* my = s_red + (int64_t)wid * k;
#pragma unroll
  for (int i = 0; i < NV; ++i) {
    const int e0 = (lane + i * 64) * 8;
    if (e0 < k) {
#pragma unroll
      for (int j = 0; j < 8; ++j) my[e0 + j] = acc[i][j];
    }
  }
  __syncthreads();
  for (int c = threadIdx.x; c < k; c += blockDim.x) {
    dbias[brow * (int64_t)k + c] =
        s_red[c] + s_red[k + c] + s_red[2 * k + c] + s_red[3 * k + c];
  }
}

__device__ __forceinline__ float block_red_max(float v, float* red) {
  v = wave_max(v);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) red[wid] = v;
  __syncthreads();
  v = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  __syncthreads();
  return v;
}

__device__ __forceinline__ float block_red_sum(float v, float* red) {
  v = wave_sum(v);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) red[wid] = v;
  __syncthreads();
  v = red[0] + red[1] + red[2] + red[3];
  __syncthreads();
  return v;
}

// generic-width row softmax (no dropout on this path), 256 threads per row
template <typename T>
__global__ void softmax_fwd_block_kernel(T* __restrict__ x,
                                         const T* __restrict__ amask, int64_t am_nb,
                                         int am_q, int64_t am_od,
                                         const T* __restrict__ bias, int64_t bs_nb,
                                         int bs_q, int64_t bs_od, int64_t n_rows,
                                         int q_len, int k) {
  __shared__ float red[4];
  const int tid = threadIdx.x;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    T* xrow = x + row * (int64_t)k;
    const int64_t b = row / q_len;
    const int qi = (int)(row - b * q_len);
    const T* mrow = amask ? amask + (((b / am_od) % am_nb) * am_q + (qi % am_q)) * (int64_t)k
                          : nullptr;
    const T* brow = bias ? bias + (((b / bs_od) % bs_nb) * bs_q + (qi % bs_q)) * (int64_t)k
                         : nullptr;
    float mx = -INFINITY;
    for (int e = tid; e < k; e += 256) {
      float v = Cvt<T>::to_f(xrow[e]);
      if (mrow) v += Cvt<T>::to_f(mrow[e]);
      if (brow) v += Cvt<T>::to_f(brow[e]);
      mx = fmaxf(mx, v);
    }
    mx = block_red_max(mx, red);
    float sum = 0.f;
    for (int e = tid; e < k; e += 256) {
      float v = Cvt<T>::to_f(xrow[e]);
      if (mrow) v += Cvt<T>::to_f(mrow[e]);
      if (brow) v += Cvt<T>::to_f(brow[e]);
      const float t = __expf(v - mx);
      xrow[e] = Cvt<T>::from_f(t);
      sum += t;
    }
    sum = block_red_sum(sum, red);
    const float inv = 1.0f / sum;
    for (int e = tid; e < k; e += 256)
      xrow[e] = Cvt<T>::from_f(Cvt<T>::to_f(xrow[e]) * inv);
    __syncthreads();
  }
}

template <typename T>
__global__ void softmax_bwd_block_kernel(T* __restrict__ g, const T* __restrict__ y,
                                         int64_t n_rows, int k) {
  __shared__ float red[4];
  const int tid = threadIdx.x;
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    T* grow = g + row * (int64_t)k;
    const T* yrow = y + row * (int64_t)k;
    float s = 0.f;
    for (int e = tid; e < k; e += 256)
      s += Cvt<T>::to_f(grow[e]) * Cvt<T>::to_f(yrow[e]);
    s = block_red_sum(s, red);
    for (int e = tid; e < k; e += 256) {
      const float yv = Cvt<T>::to_f(yrow[e]);
      grow[e] = Cvt<T>::from_f(yv * (Cvt<T>::to_f(grow[e]) - s));
    }
    __syncthreads();
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

struct SrcDesc {
  const void* ptr = nullptr;
  int64_t nb = 1;
  int q = 1;
  int64_t od = 1;
};

SrcDesc describe_src(const std::optional<at::Tensor>& t, int64_t outer_div, int k,
                     const char* what) {
  SrcDesc d;
  if (t.has_value() && t->defined()) {
    TORCH_CHECK(t->is_cuda() && t->is_contiguous(), what, " must be contiguous CUDA");
    TORCH_CHECK(t->dim() == 3 && t->size(2) == k, what, " must be (nb, q, k)");
    d.ptr = t->data_ptr();
    d.nb = t->size(0);
    d.q = (int)t->size(1);
    d.od = outer_div > 0 ? outer_div : 1;
  }
  return d;
}

}  // namespace

std::vector<at::Tensor> softmax_dropout_forward(
    bool is_training, at::Tensor input, std::optional<at::Tensor> mask,
    int64_t mask_outer_div, std::optional<at::Tensor> bias, int64_t bias_outer_div,
    double dropout_prob) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous(),
              "softmax_dropout: input must be contiguous CUDA");
  TORCH_CHECK(input.dim() == 3, "softmax_dropout: input must be (n, q, k)");
  const int64_t n_batch = input.size(0);
  const int q_len = (int)input.size(1);
  const int k = (int)input.size(2);
  const int64_t n_rows = n_batch * q_len;
  const bool drop = is_training && dropout_prob > 0.0;
  const bool vec_ok = (k % 8 == 0) && k <= 4096;
  TORCH_CHECK(!drop || vec_ok,
              "softmax_dropout: fused dropout requires k % 8 == 0 and k <= 4096 "
              "(the Python shim routes other shapes through torch dropout)");

  const SrcDesc m = describe_src(mask, mask_outer_div, k, "mask");
  const SrcDesc bsrc = describe_src(bias, bias_outer_div, k, "bias");
  if (m.ptr) TORCH_CHECK(mask->scalar_type() == input.scalar_type(), "mask dtype mismatch");
  if (bsrc.ptr) TORCH_CHECK(bias->scalar_type() == input.scalar_type(), "bias dtype mismatch");

  auto stream = at::cuda::getCurrentCUDAStream();

  at::Tensor out = input;  // p == 0: softmax written in place, out aliases it
  at::Tensor dmask;
  float pinv = 1.f;
  uint32_t pthresh = 0;
  uint64_t seed = 0, rng_offset = 0;
  if (drop) {
    out = at::empty_like(input);
    dmask = at::empty({n_batch, q_len, k / 8},
                      input.options().dtype(at::kByte));
    const double p = std::min(dropout_prob, 0.999999);
    pinv = (float)(1.0 / (1.0 - p));
    pthresh = keep16_threshold(p);
    auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
        std::nullopt, at::cuda::detail::getDefaultCUDAGenerator());
    const int nv = (k + 511) / 512;
    at::PhiloxCudaState state;
    {
      std::lock_guard<std::mutex> lock(gen->mutex_);
      state = gen->philox_cuda_state(2 * nv);
    }
    // fold the generator offset into the key so consecutive calls draw
    // independent streams while fwd remains seed-deterministic
    seed = state.seed_.val + state.offset_.val * 0x9E3779B97F4A7C15ull;
    rng_offset = 0;
  } else {
    dmask = at::empty({0}, input.options().dtype(at::kByte));
  }

  if (vec_ok) {
    const dim3 block(64, 4);
    const int rows_mult = k <= 512 ? 16 : (k <= 1024 ? 8 : 4);
    const dim3 grid(unicore_grid((n_rows + rows_mult - 1) / rows_mult));
    // bias-major iteration order applies when a bias broadcasts over an
    // outer batch (od == 1, e.g. the (1, H, L, L) rel-pos bias): the
    // per-(h, q) bias row then stays cache-hot across all `outer` batches
    // measured on the BERT shapes: the permutation's x/out scatter costs
    // more than the bias reread saves (0.996 vs 0.919 ms) — opt-in only
    static const bool bias_major_enabled = []() {
      const char* e = std::getenv("UNICORE_SM_BIASMAJOR");
      return e != nullptr && e[0] == '1';
    }();
    int64_t bias_major_outer = 0;
    if (bias_major_enabled && bsrc.ptr && bsrc.od == 1 && bsrc.q == q_len &&
        n_rows % (bsrc.nb * (int64_t)q_len) == 0) {
      const int64_t outer = n_rows / (bsrc.nb * (int64_t)q_len);
      // only worth permuting when the broadcast actually repeats, and only
      // when the attention mask (if any) does not key off the row's batch
      // in a way the permutation would scatter (am_q == 1 rows are tiny
      // and L2-resident, so those stay fine)
      if (outer > 1) bias_major_outer = outer;
    }
    DISPATCH_FTYPES(input.scalar_type(), "softmax_dropout_forward", {
      auto launch = [&](auto nv_tag, auto drop_tag) {
        constexpr int NV = decltype(nv_tag)::value;
        constexpr bool DROP = decltype(drop_tag)::value;
        constexpr int ROWS = NV == 1 ? 4 : (NV == 2 ? 2 : 1);
        softmax_fwd_vec_kernel<scalar_t, NV, ROWS, DROP>
            <<<grid, block, 0, stream>>>(
                reinterpret_cast<scalar_t*>(out.data_ptr()),
                drop ? dmask.data_ptr<uint8_t>() : nullptr,
                reinterpret_cast<scalar_t*>(input.data_ptr()),
                reinterpret_cast<const scalar_t*>(m.ptr), m.nb, m.q, m.od,
                reinterpret_cast<const scalar_t*>(bsrc.ptr), bsrc.nb, bsrc.q,
                bsrc.od, n_rows, q_len, k, pinv, pthresh, seed, rng_offset,
                bias_major_outer);
      };
      auto pick_nv = [&](auto drop_tag) {
        if (k <= 512)
          launch(std::integral_constant<int, 1>{}, drop_tag);
        else if (k <= 1024)
          launch(std::integral_constant<int, 2>{}, drop_tag);
        else if (k <= 2048)
          launch(std::integral_constant<int, 4>{}, drop_tag);
        else
          launch(std::integral_constant<int, 8>{}, drop_tag);
      };
      if (drop)
        pick_nv(std::true_type{});
      else
        pick_nv(std::false_type{});
    });
  } else {
    const dim3 block(256);
    const dim3 grid(unicore_grid(n_rows));
    DISPATCH_FTYPES(input.scalar_type(), "softmax_dropout_forward", {
      softmax_fwd_block_kernel<scalar_t><<<grid, block, 0, stream>>>(
          reinterpret_cast<scalar_t*>(input.data_ptr()),
          reinterpret_cast<const scalar_t*>(m.ptr), m.nb, m.q, m.od,
          reinterpret_cast<const scalar_t*>(bsrc.ptr), bsrc.nb, bsrc.q, bsrc.od,
          n_rows, q_len, k);
    });
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  // (out, dropout_mask, softmax_results); softmax is in-place over the input
  return {out, dmask, input};
}

at::Tensor softmax_dropout_backward(at::Tensor grad_output, at::Tensor softmax_results,
                                    at::Tensor dropout_mask, double dropout_prob) {
  TORCH_CHECK(grad_output.is_cuda() && grad_output.is_contiguous(),
              "softmax_dropout_backward: grad must be contiguous CUDA");
  TORCH_CHECK(grad_output.sizes() == softmax_results.sizes(), "shape mismatch");
  TORCH_CHECK(grad_output.dim() == 3, "grad must be (n, q, k)");
  const int64_t n_rows = grad_output.size(0) * grad_output.size(1);
  const int k = (int)grad_output.size(2);
  const bool drop = dropout_mask.defined() && dropout_mask.numel() > 0;
  const bool vec_ok = (k % 8 == 0) && k <= 4096;
  TORCH_CHECK(!drop || vec_ok, "softmax_dropout_backward: bad dropout shape");
  const float pinv = drop ? (float)(1.0 / (1.0 - std::min(dropout_prob, 0.999999))) : 1.f;

  auto stream = at::cuda::getCurrentCUDAStream();
  if (vec_ok) {
    const dim3 block(64, 4);
    const int rows_mult = k <= 512 ? 16 : (k <= 1024 ? 8 : 4);
    const dim3 grid(unicore_grid((n_rows + rows_mult - 1) / rows_mult));
    DISPATCH_FTYPES(grad_output.scalar_type(), "softmax_dropout_backward", {
      auto launch = [&](auto nv_tag, auto drop_tag) {
        constexpr int NV = decltype(nv_tag)::value;
        constexpr bool DROP = decltype(drop_tag)::value;
        constexpr int ROWS = NV == 1 ? 4 : (NV == 2 ? 2 : 1);
        softmax_bwd_vec_kernel<scalar_t, NV, ROWS, DROP>
            <<<grid, block, 0, stream>>>(
                reinterpret_cast<scalar_t*>(grad_output.data_ptr()),
                reinterpret_cast<const scalar_t*>(softmax_results.data_ptr()),
                drop ? dropout_mask.data_ptr<uint8_t>() : nullptr, n_rows, k,
                pinv);
      };
      auto pick_nv = [&](auto drop_tag) {
        if (k <= 512)
          launch(std::integral_constant<int, 1>{}, drop_tag);
        else if (k <= 1024)
          launch(std::integral_constant<int, 2>{}, drop_tag);
        else if (k <= 2048)
          launch(std::integral_constant<int, 4>{}, drop_tag);
        else
          launch(std::integral_constant<int, 8>{}, drop_tag);
      };
      if (drop)
        pick_nv(std::true_type{});
      else
        pick_nv(std::false_type{});
    });
  } else {
    const dim3 block(256);
    const dim3 grid(unicore_grid(n_rows));
    DISPATCH_FTYPES(grad_output.scalar_type(), "softmax_dropout_backward", {
      softmax_bwd_block_kernel<scalar_t><<<grid, block, 0, stream>>>(
          reinterpret_cast<scalar_t*>(grad_output.data_ptr()),
          reinterpret_cast<const scalar_t*>(softmax_results.data_ptr()), n_rows, k);
    });
  }
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return grad_output;
}

bool softmax_dropout_backward_bias_supported(int64_t n_batch, int64_t q,
                                             int64_t k, int64_t bb, int64_t bq,
                                             int64_t od) {
  if (k % 8 != 0 || k > 2048) return false;          // LDS fold <= 32 KB
  if (bq <= 0 || q % bq != 0) return false;
  if (bb <= 0 || od <= 0 || n_batch % (bb * od) != 0) return false;
  const int64_t group = (n_batch / (bb * od)) * od * (q / bq);
  return group > 1 && bb * bq >= 512;                // enough blocks to fill
}

std::vector<at::Tensor> softmax_dropout_backward_bias(
    at::Tensor grad_output, at::Tensor softmax_results, at::Tensor dropout_mask,
    double dropout_prob, int64_t bb, int64_t bq, int64_t od) {
  TORCH_CHECK(grad_output.is_cuda() && grad_output.is_contiguous(),
              "softmax_dropout_backward_bias: grad must be contiguous CUDA");
  TORCH_CHECK(grad_output.dim() == 3, "grad must be (n, q, k)");
  const int64_t n_batch = grad_output.size(0);
  const int q = (int)grad_output.size(1);
  const int k = (int)grad_output.size(2);
  TORCH_CHECK(
      softmax_dropout_backward_bias_supported(n_batch, q, k, bb, bq, od),
      "softmax_dropout_backward_bias: unsupported shape");
  const int a = (int)(n_batch / (bb * od));
  const bool drop = dropout_mask.defined() && dropout_mask.numel() > 0;
  const float pinv =
      drop ? (float)(1.0 / (1.0 - std::min(dropout_prob, 0.999999))) : 1.f;
  auto dbias = torch::empty({bb * bq, k},
                            grad_output.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const size_t lds = 4 * (size_t)k * sizeof(float);
  DISPATCH_FTYPES(grad_output.scalar_type(), "softmax_dropout_backward_bias", {
    auto launch = [&](auto nv_tag, auto drop_tag) {
      constexpr int NV = decltype(nv_tag)::value;
      constexpr bool DROP = decltype(drop_tag)::value;
      softmax_bwd_biasgrad_kernel<scalar_t, NV, DROP>
          <<<(int)(bb * bq), 256, lds, stream>>>(
              reinterpret_cast<scalar_t*>(grad_output.data_ptr()),
              reinterpret_cast<const scalar_t*>(softmax_results.data_ptr()),
              drop ? dropout_mask.data_ptr<uint8_t>() : nullptr,
              dbias.data_ptr<float>(), k, pinv, q, (int)bb, (int)bq, (int)od,
              a);
    };
    auto pick_nv = [&](auto drop_tag) {
      if (k <= 512)
        launch(std::integral_constant<int, 1>{}, drop_tag);
      else if (k <= 1024)
        launch(std::integral_constant<int, 2>{}, drop_tag);
      else
        launch(std::integral_constant<int, 4>{}, drop_tag);
    };
    if (drop)
      pick_nv(std::true_type{});
    else
      pick_nv(std::false_type{});
  });
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  return {grad_output, dbias};
}
