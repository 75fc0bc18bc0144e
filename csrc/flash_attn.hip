// Flash-style fused attention for gfx950 (CDNA4 MFMA), bf16, head_dim 64.
//
// Beyond-reference optimization: the reference materializes the O(L^2)
// score matrix (bmm -> fused softmax -> bmm, reference
// unicore/modules/multihead_attention.py:83-105); this kernel computes
// O = dropout(softmax(Q K^T + bias + mask)) V tile-by-tile with online
// softmax, never touching HBM with the score matrix.
//
// Geometry: one 256-thread block (4 wave64) per 64 query rows of one
// (batch*head); each wave owns 16 rows.  S tiles are built from
// v_mfma_f32_16x16x32_bf16 (fragment mappings verified by
// csrc/mfma_probe.hip / tests/test_kernels_gpu.py):
//   A (16x32): lane l -> row l&15,  k (l>>4)*8+[0..8)
//   B (32x16): lane l -> col l&15,  k (l>>4)*8+[0..8)
//   C (16x16): lane l -> col l&15,  rows (l>>4)*4+[0..4)
// P is redistributed C-layout -> A-layout through a per-wave LDS tile.
// Dropout keep bits are a pure function of (seed, bh*L+q, kv):
//   philox(seed, bh*L+q, kv/4) component kv%4 — recomputable by the
// backward kernels with no stored mask.
//
// Per-row LSE (= m + log l) is saved for the backward recomputation.
#include "common.h"

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGeneratorImpl.h>

#include <optional>
#include <vector>

namespace {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

constexpr int BM = 64;   // query rows per block
constexpr int BN = 64;   // kv rows per tile
constexpr int HD = 64;   // head dim

__device__ __forceinline__ bf16x8 load_frag(const uint16_t* p) {
  union {
    uint4 u;
    bf16x8 v;
  } U;
  U.u = *reinterpret_cast<const uint4*>(p);
  return U.v;
}

// reduce a per-lane value across the 16 lanes that share a C-tile row
__device__ __forceinline__ float rowgroup_max(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__device__ __forceinline__ float rowgroup_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// keep-decision for dropout: element (row_global = bh*L + q, col kv) via
// the shared 16-bit keep definition: keep16x8(seed, row_global, kv_block).
template <bool DROP>
__device__ __forceinline__ void keep_bits8(uint64_t seed, uint64_t subseq,
                                           int kv0, uint32_t pthresh,
                                           bool (&keep)[8]) {
  if constexpr (!DROP) {
#pragma unroll
    for (int j = 0; j < 8; ++j) keep[j] = true;
    return;
  }
  keep16x8(seed, subseq, kv0, pthresh, keep);
}

// V^T-resident forward (L <= 512): whole V^T staged once, kv loop runs
// barrier-free (same PMC-driven rationale as the dq K-resident variant).
template <bool HAS_BIAS, bool HAS_MASK, bool DROP, int LMAX>
__global__ __launch_bounds__(256) void flash_fwd_vres_kernel(
    uint16_t* __restrict__ out, float* __restrict__ lse,
    const uint16_t* __restrict__ qp, const uint16_t* __restrict__ kp,
    const uint16_t* __restrict__ vp,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int qt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int q0 = qt * BM + wid * 16;
  const int64_t qbase = (bh * L + q0) * HD;

  __shared__ __attribute__((aligned(16))) uint16_t lds_p[4][16][BN];
  __shared__ __attribute__((aligned(16))) uint16_t lds_vt[HD][LMAX];

  {
    const int st_kv0 = (int)threadIdx.x >> 2;
    const int st_d0 = ((int)threadIdx.x & 3) * 16;
    for (int c = 0; c < L / 64; ++c) {
      const int kv = st_kv0 + c * 64;
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(vp) +
                (bh * L + kv) * (int64_t)HD + st_d0,
            f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(vp) +
                (bh * L + kv) * (int64_t)HD + st_d0 + 8,
            f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_vt[d0][kv ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_vt[d1][kv ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
  }
  __syncthreads();

  bf16x8 aq[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
    aq[ks] = load_frag(qp + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);

  f32x4 o_acc[4] = {};
  float m_i[4], l_i[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_i[r] = -INFINITY;
    l_i[r] = 0.f;
  }
  const uint16_t* bias_rows[4];
  const uint16_t* mask_row = nullptr;
  if (HAS_BIAS) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + lg * 4 + r;
      bias_rows[r] =
          bias + (((bh / bias_od) % bias_nb) * bias_q + (q % bias_q)) * (int64_t)L;
    }
  }
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;

  const int n_tiles = L / BN;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * BN;
    f32x4 s[4];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      f32x4 acc = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bk = load_frag(
            kp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, acc, 0, 0, 0);
      }
      s[cb] = acc;
    }
    if (HAS_BIAS || HAS_MASK) {
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int kv = kv0 + cb * 16 + lr;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float add = 0.f;
          if (HAS_BIAS)
            add += __bfloat162float(
                reinterpret_cast<const __hip_bfloat16*>(bias_rows[r])[kv]);
          if (HAS_MASK)
            add += __bfloat162float(
                reinterpret_cast<const __hip_bfloat16*>(mask_row)[kv]);
          s[cb][r] += add;
        }
      }
    }
    float mnew[4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(s[0][r], s[1][r]), fmaxf(s[2][r], s[3][r]));
      mx = rowgroup_max(mx);
      mnew[r] = fmaxf(m_i[r], mx);
      alpha[r] = __expf(m_i[r] - mnew[r]);
      m_i[r] = mnew[r];
    }
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        s[cb][r] = __expf(s[cb][r] - mnew[r]);
        psum[r] += s[cb][r];
      }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_i[r] = l_i[r] * alpha[r] + rowgroup_sum(psum[r]);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) o_acc[cb][r] *= alpha[r];
    }
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lds_p[wid][lg * 4 + r][cb * 16 + lr] = f32_to_bf16_bits(s[cb][r]);
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      bf16x8 ap = load_frag(&lds_p[wid][lr][ks2 * 32 + lg * 8]);
      if constexpr (DROP) {
        bool keep[8];
        keep_bits8<DROP>(seed, (uint64_t)(bh * L + q0 + lr),
                         kv0 + ks2 * 32 + lg * 8, pthresh, keep);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float pv = bf16_bits_to_f32((uint16_t)(unsigned short)ap[j]);
          pv = keep[j] ? pv * pinv : 0.f;
          ap[j] = (short)f32_to_bf16_bits(pv);
        }
      }
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const bf16x8 bv = load_frag(
            &lds_vt[d][(kv0 + ks2 * 32 + lg * 8) ^ ((d & 7) << 3)]);
        o_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, o_acc[cb], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = 1.0f / l_i[r];
    const int q = q0 + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
      out[(bh * L + q) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(o_acc[cb][r] * inv);
    if (lr == 0) lse[bh * L + q] = m_i[r] + __logf(l_i[r]);
  }
}

template <bool HAS_BIAS, bool HAS_MASK, bool DROP>
__global__ __launch_bounds__(256) void flash_fwd_kernel(
    uint16_t* __restrict__ out, float* __restrict__ lse,
    const uint16_t* __restrict__ qp, const uint16_t* __restrict__ kp,
    const uint16_t* __restrict__ vp,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  // grid: (L/BM, B*H)
  const int qt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;   // wave 0..3
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;           // fragment k-group / row group
  const int lr = lane & 15;           // fragment row (A) / col (B, C)

  const int q0 = qt * BM + wid * 16;  // this wave's first q row
  const int64_t qbase = (bh * L + q0) * HD;

  __shared__ __attribute__((aligned(16))) uint16_t lds_p[4][16][BN];
  // V tile transposed [d][kv], kv XOR-swizzled in 8-element blocks keyed on
  // d&7 so the PV B-fragment reads (different d rows, same kv range) are
  // bank-conflict-free (guide T2 pattern)
  __shared__ __attribute__((aligned(16))) uint16_t lds_vt[HD][BN];

  // Q A-fragments (row lr, d = lg*8 + ks*32 + [0..8))
  bf16x8 aq[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
    aq[ks] = load_frag(qp + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);

  f32x4 o_acc[4] = {};
  float m_i[4], l_i[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_i[r] = -INFINITY;
    l_i[r] = 0.f;
  }

  // bias/mask source rows for this wave's 4 C rows (rows lg*4 + r)
  // bias element addressing (same contract as softmax_dropout):
  //   src_row = ((bh / od) % nb) * src_q + (q % src_q)
  const uint16_t* bias_rows[4];
  const uint16_t* mask_row = nullptr;
  if (HAS_BIAS) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + lg * 4 + r;
      bias_rows[r] =
          bias + (((bh / bias_od) % bias_nb) * bias_q + (q % bias_q)) * (int64_t)L;
    }
  }
  if (HAS_MASK) {
    // mask is (nb, 1, L): one row per (bh / od) batch, shared by all q
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;
  }

  // staging role: thread handles V row kv = tid/4, d block (tid%4)*16
  const int st_kv = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  const int n_tiles = L / BN;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * BN;
    // ---- stage V^T tile ----------------------------------------------
    {
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(vp) +
                (bh * L + kv0 + st_kv) * (int64_t)HD + st_d0,
            f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(vp) +
                (bh * L + kv0 + st_kv) * (int64_t)HD + st_d0 + 8,
            f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_vt[d0][st_kv ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_vt[d1][st_kv ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
    __syncthreads();
    // ---- S = Q K^T (per wave: 16 x 64 as 4 C tiles) -------------------
    f32x4 s[4];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      f32x4 acc = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        // B frag: col = kv0 + cb*16 + lr, k = d = ks*32 + lg*8 + [0..8)
        const bf16x8 bk = load_frag(
            kp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, acc, 0, 0, 0);
      }
      s[cb] = acc;
    }
    // ---- add bias / mask ---------------------------------------------
    if (HAS_BIAS || HAS_MASK) {
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int kv = kv0 + cb * 16 + lr;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float add = 0.f;
          if (HAS_BIAS)
            add += __bfloat162float(
                reinterpret_cast<const __hip_bfloat16*>(bias_rows[r])[kv]);
          if (HAS_MASK)
            add += __bfloat162float(
                reinterpret_cast<const __hip_bfloat16*>(mask_row)[kv]);
          s[cb][r] += add;
        }
      }
    }
    // ---- online softmax ----------------------------------------------
    float mnew[4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(s[0][r], s[1][r]), fmaxf(s[2][r], s[3][r]));
      mx = rowgroup_max(mx);
      mnew[r] = fmaxf(m_i[r], mx);
      alpha[r] = __expf(m_i[r] - mnew[r]);
      m_i[r] = mnew[r];
    }
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        s[cb][r] = __expf(s[cb][r] - mnew[r]);
        psum[r] += s[cb][r];
      }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_i[r] = l_i[r] * alpha[r] + rowgroup_sum(psum[r]);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) o_acc[cb][r] *= alpha[r];
    }
    // ---- redistribute P (C layout) -> A layout via wave-local LDS ----
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lds_p[wid][lg * 4 + r][cb * 16 + lr] =
            f32_to_bf16_bits(s[cb][r]);
    // wave-local LDS: the compiler's lgkmcnt waits order write->read within
    // the wave; no cross-wave sharing of lds_p[wid].
    // ---- O += dropout(P) V -------------------------------------------
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      // A frag of P: row = lr, cols kv = ks2*32 + lg*8 + [0..8)
      bf16x8 ap = load_frag(&lds_p[wid][lr][ks2 * 32 + lg * 8]);
      if constexpr (DROP) {
        bool keep[8];
        keep_bits8<DROP>(seed, (uint64_t)(bh * L + q0 + lr),
                         kv0 + ks2 * 32 + lg * 8, pthresh, keep);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float pv = bf16_bits_to_f32((uint16_t)(unsigned short)ap[j]);
          pv = keep[j] ? pv * pinv : 0.f;
          ap[j] = (short)f32_to_bf16_bits(pv);
        }
      }
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        // B frag of V from the staged transpose: row d = cb*16+lr,
        // kv block (ks2*32 + lg*8) ^ swizzle — one 16 B LDS read
        const int d = cb * 16 + lr;
        const bf16x8 bv =
            load_frag(&lds_vt[d][(ks2 * 32 + lg * 8) ^ ((d & 7) << 3)]);
        o_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, o_acc[cb], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize, store O (bf16) and LSE (fp32) -------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = 1.0f / l_i[r];
    const int q = q0 + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
      out[(bh * L + q) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(o_acc[cb][r] * inv);
    if (lr == 0) lse[bh * L + q] = m_i[r] + __logf(l_i[r]);
  }
}

struct SrcDesc {
  const void* ptr = nullptr;
  int64_t nb = 1;
  int q = 1;
  int64_t od = 1;
};

SrcDesc describe(const std::optional<at::Tensor>& t, int64_t outer_div, int L,
                 const char* what) {
  SrcDesc d;
  if (t.has_value() && t->defined()) {
    TORCH_CHECK(t->is_cuda() && t->is_contiguous() && t->dim() == 3 &&
                    t->size(2) == L && t->scalar_type() == at::kBFloat16,
                what, " must be contiguous bf16 (nb, q, L)");
    d.ptr = t->data_ptr();
    d.nb = t->size(0);
    d.q = (int)t->size(1);
    d.od = outer_div > 0 ? outer_div : 1;
  }
  return d;
}

}  // namespace

// q, k, v: (B*H, L, 64) contiguous bf16 (q pre-scaled).  Returns (o, lse).
std::vector<at::Tensor> flash_attn_forward(at::Tensor q, at::Tensor k, at::Tensor v,
                                           std::optional<at::Tensor> bias,
                                           int64_t bias_outer_div,
                                           std::optional<at::Tensor> mask,
                                           int64_t mask_outer_div,
                                           double dropout_p, bool is_training) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                  v.is_contiguous(),
              "flash_attn: q/k/v must be contiguous CUDA");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "flash_attn: bf16 only");
  TORCH_CHECK(q.dim() == 3 && q.size(2) == HD, "flash_attn: (BH, L, 64) only");
  const int64_t BH = q.size(0);
  const int L = (int)q.size(1);
  TORCH_CHECK(L % BN == 0, "flash_attn: L must be a multiple of 64");
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes(), "shape mismatch");

  const SrcDesc bd = describe(bias, bias_outer_div, L, "bias");
  const SrcDesc md = describe(mask, mask_outer_div, L, "mask");
  if (md.ptr) TORCH_CHECK(md.q == 1, "flash_attn: mask must broadcast over q");

  const bool drop = is_training && dropout_p > 0.0;
  float pinv = 1.f;
  uint32_t pthresh = 0;
  uint64_t seed = 0;
  if (drop) {
    const double pc = std::min(dropout_p, 0.999999);
    pinv = (float)(1.0 / (1.0 - pc));
    pthresh = keep16_threshold(pc);
    auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
        std::nullopt, at::cuda::detail::getDefaultCUDAGenerator());
    at::PhiloxCudaState state;
    {
      std::lock_guard<std::mutex> lock(gen->mutex_);
      // counters consumed per (row) subsequence: L/4 starting at 0; the
      // backward recomputes the same stream, so advance by the full row.
      state = gen->philox_cuda_state((L + 3) / 4 + 4);
    }
    seed = state.seed_.val;
    // NOTE: offset is intentionally NOT used — keep bits must be identical
    // in forward and backward, and both derive them from (seed, row, kv).
    // The generator advance still makes the NEXT op see fresh randomness.
    seed += state.offset_.val * 0x9E3779B97F4A7C15ull;
  }

  auto o = at::empty_like(q);
  auto lse_t = at::empty({BH, (int64_t)L}, q.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 grid(L / BM, BH);

  auto launch = [&](auto has_bias, auto has_mask, auto dropt) {
    constexpr bool HB = decltype(has_bias)::value;
    constexpr bool HM = decltype(has_mask)::value;
    constexpr bool DR = decltype(dropt)::value;
    if (L <= 512)
      flash_fwd_vres_kernel<HB, HM, DR, 512><<<grid, 256, 0, stream>>>(
          reinterpret_cast<uint16_t*>(o.data_ptr()), lse_t.data_ptr<float>(),
          reinterpret_cast<const uint16_t*>(q.data_ptr()),
          reinterpret_cast<const uint16_t*>(k.data_ptr()),
          reinterpret_cast<const uint16_t*>(v.data_ptr()),
          reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb, bd.q, bd.od,
          reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q, md.od, L,
          pinv, pthresh, seed);
    else
      flash_fwd_kernel<HB, HM, DR><<<grid, 256, 0, stream>>>(
          reinterpret_cast<uint16_t*>(o.data_ptr()), lse_t.data_ptr<float>(),
          reinterpret_cast<const uint16_t*>(q.data_ptr()),
          reinterpret_cast<const uint16_t*>(k.data_ptr()),
          reinterpret_cast<const uint16_t*>(v.data_ptr()),
          reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb, bd.q, bd.od,
          reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q, md.od, L,
          pinv, pthresh, seed);
  };
  auto pick = [&](auto has_bias, auto has_mask) {
    if (drop)
      launch(has_bias, has_mask, std::true_type{});
    else
      launch(has_bias, has_mask, std::false_type{});
  };
  if (bd.ptr && md.ptr)
    pick(std::true_type{}, std::true_type{});
  else if (bd.ptr)
    pick(std::true_type{}, std::false_type{});
  else if (md.ptr)
    pick(std::false_type{}, std::true_type{});
  else
    pick(std::false_type{}, std::false_type{});
  C10_CUDA_KERNEL_LAUNCH_CHECK();
  // seed is host-side state: keep it on CPU so `int(seed)` in the autograd
  // wrapper is free (a CUDA scalar here cost a D2H sync per forward and
  // broke hipGraph capture).
  return {o, lse_t,
          at::scalar_tensor((int64_t)seed,
                            at::TensorOptions().dtype(at::kLong))};
}

// ===========================================================================
// backward
// ===========================================================================
//
// Standard flash recomputation: P = exp(S - lse); dP = dO V^T;
// dS = P o (dropmask*pinv*dP - Di),  Di = rowsum(dO o O);
// dQ = dS K;  dK = dS^T Q;  dV = (dropmask*pinv*P)^T dO;
// dBias = sum over broadcast batches of dS (separate grouped pass).
// Dropout keep bits are recomputed from (seed, bh*L+q, kv) — no stored mask.

namespace {

// Di = rowsum(dO o O), fp32; one 8-lane group per row
__global__ void flash_dot_do_o_kernel(float* __restrict__ di,
                                      const uint16_t* __restrict__ dop,
                                      const uint16_t* __restrict__ op,
                                      int64_t n_rows) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int sub = lane >> 3;       // 8 rows per wave
  const int el = lane & 7;         // 8 elems x 8 bytes each
  for (int64_t row = ((int64_t)blockIdx.x * 4 + wid) * 8 + sub;
       row < n_rows; row += (int64_t)gridDim.x * 32) {
    float a[8], b[8];
    load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row * HD + el * 8, a);
    load8(reinterpret_cast<const __hip_bfloat16*>(op) + row * HD + el * 8, b);
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) s += a[j] * b[j];
#pragma unroll
    for (int off = 1; off < 8; off <<= 1) s += __shfl_xor(s, off, 64);
    if (el == 0) di[row] = s;
  }
}

// K^T-resident dq variant (L <= 1024): the whole K^T tile is staged once
// (L*HD bf16 <= 128 KB LDS) so the kv loop runs with NO per-tile barriers —
// the per-tile barrier+staging serialization is what keeps the tiled dq
// kernel at ~2%% MFMA utilization (PMC evidence in profiles/).
template <bool HAS_BIAS, bool HAS_MASK, bool DROP, int LMAX>
__global__ __launch_bounds__(256) void flash_bwd_dq_kres_kernel(
    uint16_t* __restrict__ dq, uint16_t* __restrict__ ds_out,
    const uint16_t* __restrict__ dop, const uint16_t* __restrict__ qp,
    const uint16_t* __restrict__ kp, const uint16_t* __restrict__ vp,
    const float* __restrict__ lse, const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int qt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int q0 = qt * BM + wid * 16;
  const int64_t qbase = (bh * L + q0) * HD;

  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][2][16][BN];
  __shared__ __attribute__((aligned(16))) uint16_t lds_kt[HD][LMAX];

  // stage the WHOLE K^T once: thread covers kv rows tid/4 + 64*c
  {
    const int st_kv0 = (int)threadIdx.x >> 2;
    const int st_d0 = ((int)threadIdx.x & 3) * 16;
    for (int c = 0; c < L / 64; ++c) {
      const int kv = st_kv0 + c * 64;
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
                (bh * L + kv) * (int64_t)HD + st_d0,
            f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
                (bh * L + kv) * (int64_t)HD + st_d0 + 8,
            f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_kt[d0][kv ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_kt[d1][kv ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
  }
  __syncthreads();

  bf16x8 aq[2], ado[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    aq[ks] = load_frag(qp + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    ado[ks] = load_frag(dop + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
  }
  float lse_r[4], di_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    lse_r[r] = lse[bh * L + q0 + lg * 4 + r];
    di_r[r] = di[bh * L + q0 + lg * 4 + r];
  }
  const float di_row = di[bh * L + q0 + lr];
  const uint16_t* bias_rows[4];
  const uint16_t* mask_row = nullptr;
  if (HAS_BIAS) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + lg * 4 + r;
      bias_rows[r] =
          bias + (((bh / bias_od) % bias_nb) * bias_q + (q % bias_q)) * (int64_t)L;
    }
  }
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;

  f32x4 dq_acc[4] = {};
  const int n_tiles = L / BN;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * BN;
    f32x4 s[4], dp[4];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      f32x4 acc = {}, accd = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bk = load_frag(
            kp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, acc, 0, 0, 0);
        const bf16x8 bvt = load_frag(
            vp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        accd = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[ks], bvt, accd, 0, 0, 0);
      }
      s[cb] = acc;
      dp[cb] = accd;
    }
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      const int kv = kv0 + cb * 16 + lr;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s[cb][r];
        if (HAS_BIAS)
          sv += __bfloat162float(
              reinterpret_cast<const __hip_bfloat16*>(bias_rows[r])[kv]);
        if (HAS_MASK)
          sv += __bfloat162float(
              reinterpret_cast<const __hip_bfloat16*>(mask_row)[kv]);
        s[cb][r] = __expf(sv - lse_r[r]);
      }
    }
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        lds_t[wid][0][lg * 4 + r][cb * 16 + lr] = f32_to_bf16_bits(s[cb][r]);
        lds_t[wid][1][lg * 4 + r][cb * 16 + lr] = f32_to_bf16_bits(dp[cb][r]);
      }
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      const bf16x8 pa = load_frag(&lds_t[wid][0][lr][ks2 * 32 + lg * 8]);
      const bf16x8 dpa = load_frag(&lds_t[wid][1][lr][ks2 * 32 + lg * 8]);
      bool keep[8];
      keep_bits8<DROP>(seed, (uint64_t)(bh * L + q0 + lr),
                       kv0 + ks2 * 32 + lg * 8, pthresh, keep);
      bf16x8 dsa;
      float dsf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float pv = bf16_bits_to_f32((uint16_t)(unsigned short)pa[j]);
        float dpv = bf16_bits_to_f32((uint16_t)(unsigned short)dpa[j]);
        if (DROP) dpv = keep[j] ? dpv * pinv : 0.f;
        dsf[j] = pv * (dpv - di_row);
        dsa[j] = (short)f32_to_bf16_bits(dsf[j]);
      }
      if (ds_out != nullptr) {
        union {
          bf16x8 v;
          uint4 u;
        } U;
        U.v = dsa;
        *reinterpret_cast<uint4*>(ds_out + (bh * L + q0 + lr) * (int64_t)L +
                                  kv0 + ks2 * 32 + lg * 8) = U.u;
      }
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const bf16x8 bkf = load_frag(
            &lds_kt[d][(kv0 + ks2 * 32 + lg * 8) ^ ((d & 7) << 3)]);
        dq_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, bkf, dq_acc[cb], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int q = q0 + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
      dq[(bh * L + q) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dq_acc[cb][r]);
  }
}

// 32-row-per-wave K-resident dq (L % 128 == 0, L <= 1024): each wave owns
// TWO 16-row M-tiles, so every K/V B-fragment load feeds two MFMAs and the
// per-tile scalar overheads (lse/di/bias addressing, philox) amortize.
// P/dS redistribute shares one LDS buffer sequentially (wave-local order).
template <bool HAS_BIAS, bool HAS_MASK, bool DROP, int LMAX>
__global__ __launch_bounds__(256) void flash_bwd_dq_k32_kernel(
    uint16_t* __restrict__ dq, uint16_t* __restrict__ ds_out,
    const uint16_t* __restrict__ dop, const uint16_t* __restrict__ qp,
    const uint16_t* __restrict__ kp, const uint16_t* __restrict__ vp,
    const float* __restrict__ lse, const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int qt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int q0 = qt * 128 + wid * 32;   // wave's first q row (2 M-tiles)

  // K^T is staged per kv-tile (8 KB) instead of full-L resident (64 KB):
  // the resident version capped the block at 80 KB LDS -> 2 blocks/CU ->
  // 2 waves/SIMD, and PMC showed 41% of wave cycles parked on waits —
  // occupancy, not staging traffic, was the binding constraint.
  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][2][16][BN];
  __shared__ __attribute__((aligned(16))) uint16_t lds_kt[2][HD][BN];
  const int st_kv0 = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  bf16x8 aq[2][2], ado[2][2];
  float lse_r[2][4], di_r[2][4], di_row[2];
  const uint16_t* bias_rows[2][4];
  const uint16_t* mask_row = nullptr;
#pragma unroll
  for (int mtile = 0; mtile < 2; ++mtile) {
    const int64_t qbase = (bh * L + q0 + mtile * 16) * (int64_t)HD;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      aq[mtile][ks] = load_frag(qp + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
      ado[mtile][ks] = load_frag(dop + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + mtile * 16 + lg * 4 + r;
      lse_r[mtile][r] = lse[bh * L + q];
      di_r[mtile][r] = di[bh * L + q];
      if (HAS_BIAS)
        bias_rows[mtile][r] =
            bias +
            (((bh / bias_od) % bias_nb) * bias_q + (q % bias_q)) * (int64_t)L;
    }
    di_row[mtile] = di[bh * L + q0 + mtile * 16 + lr];
  }
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;

  f32x4 dq_acc[2][4] = {};
  const int n_tiles = L / BN;
  // NOTE (measured): a T14 register prefetch of the next tile's K/V
  // fragments was tried here (issue after the bias loads, single register
  // set) — dq went 630 -> 832-1099 us. At 264+ VGPR / 1 wave/SIMD the
  // extra 96 registers hurt scheduling more than the ~500-cycle load
  // latency costs; the FIFO vm-queue also forces later short loads to
  // drain any outstanding prefetch. Reverted; kept as a record.
  // ping-pong K^T staging: stage tile t+1 into the other buffer while
  // computing tile t, one barrier per iteration
  auto stage_kt = [&](int tile) {
    const int kv = tile * BN + st_kv0;
    uint16_t* buf = &lds_kt[tile & 1][0][0];
    float f0[8], f1[8];
    load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
              (bh * L + kv) * (int64_t)HD + st_d0,
          f0);
    load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
              (bh * L + kv) * (int64_t)HD + st_d0 + 8,
          f1);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d0 = st_d0 + j;
      const int d1 = st_d0 + 8 + j;
      buf[d0 * BN + (st_kv0 ^ ((d0 & 7) << 3))] = f32_to_bf16_bits(f0[j]);
      buf[d1 * BN + (st_kv0 ^ ((d1 & 7) << 3))] = f32_to_bf16_bits(f1[j]);
    }
  };
  stage_kt(0);
  __syncthreads();
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * BN;
    if (t + 1 < n_tiles) stage_kt(t + 1);
    f32x4 s[2][4], dp[2][4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      f32x4 acc0 = {}, acc1 = {}, accd0 = {}, accd1 = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bk = load_frag(
            kp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        const bf16x8 bvt = load_frag(
            vp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[0][ks], bk, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[1][ks], bk, acc1, 0, 0, 0);
        accd0 =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[0][ks], bvt, accd0, 0, 0, 0);
        accd1 =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[1][ks], bvt, accd1, 0, 0, 0);
      }
      s[0][cb] = acc0;
      s[1][cb] = acc1;
      dp[0][cb] = accd0;
      dp[1][cb] = accd1;
    }
    __builtin_amdgcn_s_setprio(0);
#pragma unroll
    for (int mtile = 0; mtile < 2; ++mtile)
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int kv = kv0 + cb * 16 + lr;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = s[mtile][cb][r];
          if (HAS_BIAS)
            sv += __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
                bias_rows[mtile][r])[kv]);
          if (HAS_MASK)
            sv += __bfloat162float(
                reinterpret_cast<const __hip_bfloat16*>(mask_row)[kv]);
          s[mtile][cb][r] = __expf(sv - lse_r[mtile][r]);
        }
      }
    // redistribute P (both M-tiles), read A-fragments, then reuse for dS
#pragma unroll
    for (int mtile = 0; mtile < 2; ++mtile)
#pragma unroll
      for (int cb = 0; cb < 4; ++cb)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          lds_t[wid][mtile][lg * 4 + r][cb * 16 + lr] =
              f32_to_bf16_bits(s[mtile][cb][r]);
    bf16x8 pa[2][2];
#pragma unroll
    for (int mtile = 0; mtile < 2; ++mtile)
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2)
        pa[mtile][ks2] = load_frag(&lds_t[wid][mtile][lr][ks2 * 32 + lg * 8]);
#pragma unroll
    for (int mtile = 0; mtile < 2; ++mtile)
#pragma unroll
      for (int cb = 0; cb < 4; ++cb)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          lds_t[wid][mtile][lg * 4 + r][cb * 16 + lr] =
              f32_to_bf16_bits(dp[mtile][cb][r]);
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      bool keep[2][8];
      bf16x8 dsa[2];
#pragma unroll
      for (int mtile = 0; mtile < 2; ++mtile) {
        const bf16x8 dpa =
            load_frag(&lds_t[wid][mtile][lr][ks2 * 32 + lg * 8]);
        keep_bits8<DROP>(seed, (uint64_t)(bh * L + q0 + mtile * 16 + lr),
                         kv0 + ks2 * 32 + lg * 8, pthresh, keep[mtile]);
        float dsf[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float pv =
              bf16_bits_to_f32((uint16_t)(unsigned short)pa[mtile][ks2][j]);
          float dpv = bf16_bits_to_f32((uint16_t)(unsigned short)dpa[j]);
          if (DROP) dpv = keep[mtile][j] ? dpv * pinv : 0.f;
          dsf[j] = pv * (dpv - di_row[mtile]);
          dsa[mtile][j] = (short)f32_to_bf16_bits(dsf[j]);
        }
        if (ds_out != nullptr) {
          union {
            bf16x8 v;
            uint4 u;
          } U;
          U.v = dsa[mtile];
          *reinterpret_cast<uint4*>(
              ds_out + (bh * L + q0 + mtile * 16 + lr) * (int64_t)L + kv0 +
              ks2 * 32 + lg * 8) = U.u;
        }
      }
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const bf16x8 bkf = load_frag(
            &lds_kt[t & 1][d][(ks2 * 32 + lg * 8) ^ ((d & 7) << 3)]);
        dq_acc[0][cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa[0], bkf, dq_acc[0][cb], 0, 0, 0);
        dq_acc[1][cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa[1], bkf, dq_acc[1][cb], 0, 0, 0);
      }
    }
    // publish tile t+1's staging; reads of buffer t&1 are also done, so
    // its restage at t+2 (after the next barrier) is safe
    __syncthreads();
  }
#pragma unroll
  for (int mtile = 0; mtile < 2; ++mtile)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + mtile * 16 + lg * 4 + r;
#pragma unroll
      for (int cb = 0; cb < 4; ++cb)
        dq[(bh * L + q) * (int64_t)HD + cb * 16 + lr] =
            f32_to_bf16_bits(dq_acc[mtile][cb][r]);
    }
}

template <bool HAS_BIAS, bool HAS_MASK, bool DROP>
__global__ __launch_bounds__(256) void flash_bwd_dq_kernel(
    uint16_t* __restrict__ dq, uint16_t* __restrict__ ds_out,
    float* __restrict__ dbias_acc,
    const uint16_t* __restrict__ dop,
    const uint16_t* __restrict__ qp, const uint16_t* __restrict__ kp,
    const uint16_t* __restrict__ vp, const float* __restrict__ lse,
    const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int qt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int q0 = qt * BM + wid * 16;
  const int64_t qbase = (bh * L + q0) * HD;

  // [wave][0]=P tile, [wave][1]=dS tile (C-layout write, A-layout read)
  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][2][16][BN];
  // K tile transposed [d][kv] (swizzled) for the dQ = dS K product
  __shared__ __attribute__((aligned(16))) uint16_t lds_kt[HD][BN];
  const int st_kv = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  bf16x8 aq[2], ado[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    aq[ks] = load_frag(qp + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    ado[ks] = load_frag(dop + qbase + (int64_t)lr * HD + ks * 32 + lg * 8);
  }
  float lse_r[4], di_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    lse_r[r] = lse[bh * L + q0 + lg * 4 + r];
    di_r[r] = di[bh * L + q0 + lg * 4 + r];
  }
  const float di_row = di[bh * L + q0 + lr];   // Di for this lane's A row
  const uint16_t* bias_rows[4];
  const uint16_t* mask_row = nullptr;
  if (HAS_BIAS) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q = q0 + lg * 4 + r;
      bias_rows[r] =
          bias + (((bh / bias_od) % bias_nb) * bias_q + (q % bias_q)) * (int64_t)L;
    }
  }
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;

  f32x4 dq_acc[4] = {};
  const int n_tiles = L / BN;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * BN;
    {
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
                (bh * L + kv0 + st_kv) * (int64_t)HD + st_d0,
            f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(kp) +
                (bh * L + kv0 + st_kv) * (int64_t)HD + st_d0 + 8,
            f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_kt[d0][st_kv ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_kt[d1][st_kv ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
    __syncthreads();
    f32x4 s[4], dp[4];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      f32x4 acc = {}, accd = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bk = load_frag(
            kp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, acc, 0, 0, 0);
        const bf16x8 bvt = load_frag(
            vp + (bh * L + kv0 + cb * 16 + lr) * (int64_t)HD + ks * 32 + lg * 8);
        accd = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[ks], bvt, accd, 0, 0, 0);
      }
      s[cb] = acc;
      dp[cb] = accd;
    }
    // P = exp(S + bias + mask - lse)
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      const int kv = kv0 + cb * 16 + lr;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s[cb][r];
        if (HAS_BIAS)
          sv += __bfloat162float(
              reinterpret_cast<const __hip_bfloat16*>(bias_rows[r])[kv]);
        if (HAS_MASK)
          sv += __bfloat162float(
              reinterpret_cast<const __hip_bfloat16*>(mask_row)[kv]);
        s[cb][r] = __expf(sv - lse_r[r]);
      }
    }
    // redistribute P and dP to A layout
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        lds_t[wid][0][lg * 4 + r][cb * 16 + lr] = f32_to_bf16_bits(s[cb][r]);
        lds_t[wid][1][lg * 4 + r][cb * 16 + lr] = f32_to_bf16_bits(dp[cb][r]);
      }
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      const bf16x8 pa = load_frag(&lds_t[wid][0][lr][ks2 * 32 + lg * 8]);
      const bf16x8 dpa = load_frag(&lds_t[wid][1][lr][ks2 * 32 + lg * 8]);
      bool keep[8];
      keep_bits8<DROP>(seed, (uint64_t)(bh * L + q0 + lr),
                       kv0 + ks2 * 32 + lg * 8, pthresh, keep);
      bf16x8 dsa;
      float dsf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float pv = bf16_bits_to_f32((uint16_t)(unsigned short)pa[j]);
        float dpv = bf16_bits_to_f32((uint16_t)(unsigned short)dpa[j]);
        if (DROP) dpv = keep[j] ? dpv * pinv : 0.f;
        dsf[j] = pv * (dpv - di_row);
        dsa[j] = (short)f32_to_bf16_bits(dsf[j]);
      }
      if (HAS_BIAS && dbias_acc != nullptr) {
        // fused dbias accumulation (see kres variant)
        const int64_t brow =
            ((bh / bias_od) % bias_nb) * (int64_t)bias_q + ((q0 + lr) % bias_q);
        float* bdst = dbias_acc + brow * (int64_t)L + kv0 + ks2 * 32 + lg * 8;
#pragma unroll
        for (int j = 0; j < 8; ++j) atomicAdd(bdst + j, dsf[j]);
      }
      if (ds_out != nullptr) {
        // materialize dS for the bias gradient (deterministic fallback;
        // the broadcast-batch sum happens as one torch reduction)
        union {
          bf16x8 v;
          uint4 u;
        } U;
        U.v = dsa;
        *reinterpret_cast<uint4*>(ds_out + (bh * L + q0 + lr) * (int64_t)L +
                                  kv0 + ks2 * 32 + lg * 8) = U.u;
      }
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const bf16x8 bkf =
            load_frag(&lds_kt[d][(ks2 * 32 + lg * 8) ^ ((d & 7) << 3)]);
        dq_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, bkf, dq_acc[cb], 0, 0, 0);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int q = q0 + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
      dq[(bh * L + q) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dq_acc[cb][r]);
  }
}

// Q^T/dO^T-hybrid dkv variant (L <= 512): dO^T staged fully (the q loop
// runs without per-tile dO staging barriers); the P/dS redistribute buffer
// is shared sequentially (wave-local ordering), keeping the block at
// <= 80 KB LDS for 2 blocks/CU.
// Q^T/dO^T-hybrid dkv variant (L <= 512): dO^T staged fully (the q loop
// runs without per-tile dO staging barriers); the P/dS redistribute buffer
// is shared sequentially (wave-local ordering), keeping the block at
// <= 80 KB LDS for 2 blocks/CU.
// 32-kv-row-per-wave dkv (L % 128 == 0, L <= 512): two kv M-tiles per
// wave — every Q/dO B-fragment and the staged lds_qt/lds_dot reads feed
// two MFMAs; per-q-tile staging barriers amortize over 2x the math.
template <bool HAS_BIAS, bool HAS_MASK, bool DROP, int LMAX>
__global__ __launch_bounds__(256) void flash_bwd_dkv_q32_kernel(
    uint16_t* __restrict__ dk, uint16_t* __restrict__ dv,
    const uint16_t* __restrict__ dop, const uint16_t* __restrict__ qp,
    const uint16_t* __restrict__ kp, const uint16_t* __restrict__ vp,
    const float* __restrict__ lse, const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int kt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int kv0w = kt * 128 + wid * 32;   // wave's first kv row (2 M-tiles)

  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][2][16][BN];
  __shared__ __attribute__((aligned(16))) uint16_t lds_dot[HD][LMAX];
  __shared__ __attribute__((aligned(16))) uint16_t lds_qt[HD][BM];
  const int st_q0 = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  {
    for (int c = 0; c < L / 64; ++c) {
      const int qq = st_q0 + c * 64;
      const int64_t row = (bh * L + qq) * (int64_t)HD;
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0, f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0 + 8, f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_dot[d0][qq ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_dot[d1][qq ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
  }
  __syncthreads();

  bf16x8 ak[2][2], av[2][2];
  float maskv[2][4];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    const int64_t kvbase = (bh * L + kv0w + mt * 16) * (int64_t)HD;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      ak[mt][ks] = load_frag(kp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
      av[mt][ks] = load_frag(vp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    }
  }
  const uint16_t* mask_row = nullptr;
  if (HAS_MASK) {
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        maskv[mt][r] = __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
            mask_row)[kv0w + mt * 16 + lg * 4 + r]);
  } else {
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int r = 0; r < 4; ++r) maskv[mt][r] = 0.f;
  }

  f32x4 dk_acc[2][4] = {}, dv_acc[2][4] = {};
  const int n_tiles = L / BM;
  for (int tq = 0; tq < n_tiles; ++tq) {
    const int q0 = tq * BM;
    {
      const int qq = q0 + st_q0;
      const int64_t row = (bh * L + qq) * (int64_t)HD;
      float f0[8], f1[8];
      load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0, f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0 + 8, f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_qt[d0][st_q0 ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_qt[d1][st_q0 ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
    __syncthreads();
    f32x4 st[2][4], dpt[2][4];
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      f32x4 a0 = {}, a1 = {}, d0 = {}, d1 = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bq =
            load_frag(qp + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        const bf16x8 bdo =
            load_frag(dop + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        a0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[0][ks], bq, a0, 0, 0, 0);
        a1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[1][ks], bq, a1, 0, 0, 0);
        d0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[0][ks], bdo, d0, 0, 0, 0);
        d1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[1][ks], bdo, d1, 0, 0, 0);
      }
      st[0][cq] = a0;
      st[1][cq] = a1;
      dpt[0][cq] = d0;
      dpt[1][cq] = d1;
    }
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      const float lse_c = lse[bh * L + qcol];
      const float di_c = di[bh * L + qcol];
      const uint16_t* brow =
          HAS_BIAS ? bias + (((bh / bias_od) % bias_nb) * bias_q +
                             (qcol % bias_q)) * (int64_t)L
                   : nullptr;
#pragma unroll
      for (int mt = 0; mt < 2; ++mt) {
        bool k8[8];
        const int kvrow0 = kv0w + mt * 16 + lg * 4;
        keep_bits8<DROP>(seed, (uint64_t)(bh * L + qcol), kvrow0 & ~7, pthresh,
                         k8);
        const int koff = kvrow0 & 7;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = st[mt][cq][r];
          if (HAS_BIAS)
            sv += __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
                brow)[kv0w + mt * 16 + lg * 4 + r]);
          sv += maskv[mt][r];
          const float pv = __expf(sv - lse_c);
          float dpv = dpt[mt][cq][r];
          const bool kp_ = !DROP || k8[koff + r];
          if (DROP) dpv = kp_ ? dpv * pinv : 0.f;
          st[mt][cq][r] = DROP ? (kp_ ? pv * pinv : 0.f) : pv;
          dpt[mt][cq][r] = pv * (dpv - di_c);
        }
      }
    }
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int cq = 0; cq < 4; ++cq)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          lds_t[wid][mt][lg * 4 + r][cq * 16 + lr] = f32_to_bf16_bits(st[mt][cq][r]);
    bf16x8 pta[2][2];
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2)
        pta[mt][ks2] = load_frag(&lds_t[wid][mt][lr][ks2 * 32 + lg * 8]);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int cq = 0; cq < 4; ++cq)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          lds_t[wid][mt][lg * 4 + r][cq * 16 + lr] =
              f32_to_bf16_bits(dpt[mt][cq][r]);
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      bf16x8 dsta[2];
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        dsta[mt] = load_frag(&lds_t[wid][mt][lr][ks2 * 32 + lg * 8]);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const int qx_dot = (q0 + ks2 * 32 + lg * 8) ^ ((d & 7) << 3);
        const int qx_q = (ks2 * 32 + lg * 8) ^ ((d & 7) << 3);
        const bf16x8 bdo = load_frag(&lds_dot[d][qx_dot]);
        const bf16x8 bqf = load_frag(&lds_qt[d][qx_q]);
        dv_acc[0][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pta[0][ks2], bdo,
                                                               dv_acc[0][cb], 0, 0, 0);
        dv_acc[1][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pta[1][ks2], bdo,
                                                               dv_acc[1][cb], 0, 0, 0);
        dk_acc[0][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsta[0], bqf,
                                                               dk_acc[0][cb], 0, 0, 0);
        dk_acc[1][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsta[1], bqf,
                                                               dk_acc[1][cb], 0, 0, 0);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int kv = kv0w + mt * 16 + lg * 4 + r;
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        dk[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
            f32_to_bf16_bits(dk_acc[mt][cb][r]);
        dv[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
            f32_to_bf16_bits(dv_acc[mt][cb][r]);
      }
    }
}

template <bool HAS_BIAS, bool HAS_MASK, bool DROP, int LMAX>
__global__ __launch_bounds__(256, 3) void flash_bwd_dkv_qres_kernel(
    uint16_t* __restrict__ dk, uint16_t* __restrict__ dv,
    const uint16_t* __restrict__ dop, const uint16_t* __restrict__ qp,
    const uint16_t* __restrict__ kp, const uint16_t* __restrict__ vp,
    const float* __restrict__ lse, const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int kt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int kv0w = kt * BN + wid * 16;
  const int64_t kvbase = (bh * L + kv0w) * HD;

  // Both transposed operands (Q^T, dO^T) are staged per q-tile (8 KB
  // each) rather than keeping dO^T L-resident (64 KB): the resident
  // version held the block at 80 KB LDS -> 2 blocks/CU, and PMC showed
  // the kernel parked on waits — occupancy was the binding constraint.
  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][16][BN];
  __shared__ __attribute__((aligned(16))) uint16_t lds_dot[2][HD][BM];
  __shared__ __attribute__((aligned(16))) uint16_t lds_qt[2][HD][BM];
  const int st_q0 = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  bf16x8 ak[2], av[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    ak[ks] = load_frag(kp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    av[ks] = load_frag(vp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
  }
  const uint16_t* mask_row = nullptr;
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;
  float maskv[4];
#pragma unroll
  for (int r = 0; r < 4; ++r)
    maskv[r] = HAS_MASK
                   ? __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
                         mask_row)[kv0w + lg * 4 + r])
                   : 0.f;

  f32x4 dk_acc[4] = {}, dv_acc[4] = {};
  const int n_tiles = L / BM;
  // (T14 prefetch tried and reverted here too: 845 -> 918 us at the
  // 3->2 blocks/CU occupancy cost; see the dq kernel's note.)
  // ping-pong staging of Q^T/dO^T: tile tq+1 stages into the other
  // buffer while tile tq computes, one barrier per iteration
  auto stage_q = [&](int tile) {
    const int qq = tile * BM + st_q0;
    const int64_t row = (bh * L + qq) * (int64_t)HD;
    uint16_t* bq = &lds_qt[tile & 1][0][0];
    uint16_t* bd = &lds_dot[tile & 1][0][0];
    float f0[8], f1[8], g0[8], g1[8];
    load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0, f0);
    load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0 + 8, f1);
    load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0, g0);
    load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0 + 8, g1);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d0 = st_d0 + j;
      const int d1 = st_d0 + 8 + j;
      bq[d0 * BM + (st_q0 ^ ((d0 & 7) << 3))] = f32_to_bf16_bits(f0[j]);
      bq[d1 * BM + (st_q0 ^ ((d1 & 7) << 3))] = f32_to_bf16_bits(f1[j]);
      bd[d0 * BM + (st_q0 ^ ((d0 & 7) << 3))] = f32_to_bf16_bits(g0[j]);
      bd[d1 * BM + (st_q0 ^ ((d1 & 7) << 3))] = f32_to_bf16_bits(g1[j]);
    }
  };
  stage_q(0);
  __syncthreads();
  for (int tq = 0; tq < n_tiles; ++tq) {
    const int q0 = tq * BM;
    if (tq + 1 < n_tiles) stage_q(tq + 1);
    f32x4 st[4], dpt[4];
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      f32x4 acc = {}, accd = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bq =
            load_frag(qp + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[ks], bq, acc, 0, 0, 0);
        const bf16x8 bdo =
            load_frag(dop + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        accd = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[ks], bdo, accd, 0, 0, 0);
      }
      st[cq] = acc;
      dpt[cq] = accd;
    }
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      const float lse_c = lse[bh * L + qcol];
      const float di_c = di[bh * L + qcol];
      const uint16_t* brow =
          HAS_BIAS ? bias + (((bh / bias_od) % bias_nb) * bias_q +
                             (qcol % bias_q)) * (int64_t)L
                   : nullptr;
      bool keep[4] = {true, true, true, true};
      if (DROP) {
        bool k8[8];
        const int blk = (kv0w + lg * 4) & ~7;
        keep16x8(seed, (uint64_t)(bh * L + qcol), blk, pthresh, k8);
        const int off = (kv0w + lg * 4) - blk;
#pragma unroll
        for (int r = 0; r < 4; ++r) keep[r] = k8[off + r];
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = st[cq][r];
        if (HAS_BIAS)
          sv += __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
              brow)[kv0w + lg * 4 + r]);
        sv += maskv[r];
        const float pv = __expf(sv - lse_c);
        float dpv = dpt[cq][r];
        if (DROP) dpv = keep[r] ? dpv * pinv : 0.f;
        st[cq][r] = DROP ? (keep[r] ? pv * pinv : 0.f) : pv;
        dpt[cq][r] = pv * (dpv - di_c);
      }
    }
    // redistribute P^T first (shared buffer, wave-local ordering), read
    // both A-fragments into registers, then reuse the buffer for dS^T
#pragma unroll
    for (int cq = 0; cq < 4; ++cq)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lds_t[wid][lg * 4 + r][cq * 16 + lr] = f32_to_bf16_bits(st[cq][r]);
    bf16x8 pta[2];
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2)
      pta[ks2] = load_frag(&lds_t[wid][lr][ks2 * 32 + lg * 8]);
#pragma unroll
    for (int cq = 0; cq < 4; ++cq)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lds_t[wid][lg * 4 + r][cq * 16 + lr] = f32_to_bf16_bits(dpt[cq][r]);
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      const bf16x8 dsta = load_frag(&lds_t[wid][lr][ks2 * 32 + lg * 8]);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const int qx = (ks2 * 32 + lg * 8) ^ ((d & 7) << 3);  // tile-local
        const bf16x8 bdo = load_frag(&lds_dot[tq & 1][d][qx]);
        const bf16x8 bqf = load_frag(&lds_qt[tq & 1][d][qx]);
        dv_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pta[ks2], bdo,
                                                            dv_acc[cb], 0, 0, 0);
        dk_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsta, bqf, dk_acc[cb], 0, 0, 0);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kv = kv0w + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      dk[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dk_acc[cb][r]);
      dv[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dv_acc[cb][r]);
    }
  }
}

template <bool HAS_BIAS, bool HAS_MASK, bool DROP>
__global__ __launch_bounds__(256) void flash_bwd_dkv_kernel(
    uint16_t* __restrict__ dk, uint16_t* __restrict__ dv,
    const uint16_t* __restrict__ dop, const uint16_t* __restrict__ qp,
    const uint16_t* __restrict__ kp, const uint16_t* __restrict__ vp,
    const float* __restrict__ lse, const float* __restrict__ di,
    const uint16_t* __restrict__ bias, int64_t bias_nb, int bias_q, int64_t bias_od,
    const uint16_t* __restrict__ mask, int64_t mask_nb, int mask_q, int64_t mask_od,
    int L, float pinv, uint32_t pthresh, uint64_t seed) {
  const int kt = blockIdx.x;
  const int64_t bh = blockIdx.y;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
  const int kv0w = kt * BN + wid * 16;   // this wave's first kv row
  const int64_t kvbase = (bh * L + kv0w) * HD;

  __shared__ __attribute__((aligned(16))) uint16_t lds_t[4][2][16][BN];
  // dO and Q tiles transposed [d][q] (swizzled) for the dV / dK products
  __shared__ __attribute__((aligned(16))) uint16_t lds_dot[HD][BM];
  __shared__ __attribute__((aligned(16))) uint16_t lds_qt[HD][BM];
  const int st_q = (int)threadIdx.x >> 2;
  const int st_d0 = ((int)threadIdx.x & 3) * 16;

  bf16x8 ak[2], av[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    ak[ks] = load_frag(kp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
    av[ks] = load_frag(vp + kvbase + (int64_t)lr * HD + ks * 32 + lg * 8);
  }
  const uint16_t* mask_row = nullptr;
  if (HAS_MASK)
    mask_row = mask + (((bh / mask_od) % mask_nb) * mask_q) * (int64_t)L;
  float maskv[4];   // additive mask for this wave's kv rows (constant over q)
#pragma unroll
  for (int r = 0; r < 4; ++r)
    maskv[r] = HAS_MASK
                   ? __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
                         mask_row)[kv0w + lg * 4 + r])
                   : 0.f;

  f32x4 dk_acc[4] = {}, dv_acc[4] = {};
  const int n_tiles = L / BM;
  for (int tq = 0; tq < n_tiles; ++tq) {
    const int q0 = tq * BM;
    {
      float f0[8], f1[8];
      const int64_t row = (bh * L + q0 + st_q) * (int64_t)HD;
      load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0, f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(dop) + row + st_d0 + 8, f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_dot[d0][st_q ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_dot[d1][st_q ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
      load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0, f0);
      load8(reinterpret_cast<const __hip_bfloat16*>(qp) + row + st_d0 + 8, f1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d0 = st_d0 + j;
        const int d1 = st_d0 + 8 + j;
        lds_qt[d0][st_q ^ ((d0 & 7) << 3)] = f32_to_bf16_bits(f0[j]);
        lds_qt[d1][st_q ^ ((d1 & 7) << 3)] = f32_to_bf16_bits(f1[j]);
      }
    }
    __syncthreads();
    f32x4 st[4], dpt[4];
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      f32x4 acc = {}, accd = {};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bq =
            load_frag(qp + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[ks], bq, acc, 0, 0, 0);
        const bf16x8 bdo =
            load_frag(dop + (bh * L + qcol) * (int64_t)HD + ks * 32 + lg * 8);
        accd = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[ks], bdo, accd, 0, 0, 0);
      }
      st[cq] = acc;
      dpt[cq] = accd;
    }
    // P^T = exp(S^T + bias + mask - lse[q]); dS^T = P^T o (drop(dP^T) - Di[q])
#pragma unroll
    for (int cq = 0; cq < 4; ++cq) {
      const int qcol = q0 + cq * 16 + lr;
      const float lse_c = lse[bh * L + qcol];
      const float di_c = di[bh * L + qcol];
      const uint16_t* brow =
          HAS_BIAS ? bias + (((bh / bias_od) % bias_nb) * bias_q +
                             (qcol % bias_q)) * (int64_t)L
                   : nullptr;
      bool keep[4] = {true, true, true, true};
      if (DROP) {
        // this lane's kv rows kv0w + lg*4 + [0..4) live in one 8-block
        bool k8[8];
        const int blk = (kv0w + lg * 4) & ~7;
        keep16x8(seed, (uint64_t)(bh * L + qcol), blk, pthresh, k8);
        const int off = (kv0w + lg * 4) - blk;
#pragma unroll
        for (int r = 0; r < 4; ++r) keep[r] = k8[off + r];
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = st[cq][r];
        if (HAS_BIAS)
          sv += __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
              brow)[kv0w + lg * 4 + r]);
        sv += maskv[r];
        const float pv = __expf(sv - lse_c);
        float dpv = dpt[cq][r];
        if (DROP) dpv = keep[r] ? dpv * pinv : 0.f;
        st[cq][r] = DROP ? (keep[r] ? pv * pinv : 0.f) : pv;  // dropped P^T for dV
        dpt[cq][r] = pv * (dpv - di_c);                       // dS^T
      }
    }
    // redistribute P^T(dropped) and dS^T to A layout
#pragma unroll
    for (int cq = 0; cq < 4; ++cq)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        lds_t[wid][0][lg * 4 + r][cq * 16 + lr] = f32_to_bf16_bits(st[cq][r]);
        lds_t[wid][1][lg * 4 + r][cq * 16 + lr] = f32_to_bf16_bits(dpt[cq][r]);
      }
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      const bf16x8 pta = load_frag(&lds_t[wid][0][lr][ks2 * 32 + lg * 8]);
      const bf16x8 dsta = load_frag(&lds_t[wid][1][lr][ks2 * 32 + lg * 8]);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const int d = cb * 16 + lr;
        const int qx = (ks2 * 32 + lg * 8) ^ ((d & 7) << 3);
        const bf16x8 bdo = load_frag(&lds_dot[d][qx]);
        const bf16x8 bqf = load_frag(&lds_qt[d][qx]);
        dv_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(pta, bdo, dv_acc[cb], 0, 0, 0);
        dk_acc[cb] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsta, bqf, dk_acc[cb], 0, 0, 0);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kv = kv0w + lg * 4 + r;
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      dk[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dk_acc[cb][r]);
      dv[(bh * L + kv) * (int64_t)HD + cb * 16 + lr] =
          f32_to_bf16_bits(dv_acc[cb][r]);
    }
  }
}

}  // namespace

std::vector<at::Tensor> flash_attn_backward(
    at::Tensor d_out, at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
    at::Tensor lse, std::optional<at::Tensor> bias, int64_t bias_outer_div,
    bool bias_needs_grad, std::optional<at::Tensor> mask, int64_t mask_outer_div,
    double dropout_p, bool dropped, int64_t seed_in) {
  TORCH_CHECK(d_out.is_cuda() && d_out.is_contiguous() && q.is_contiguous() &&
                  k.is_contiguous() && v.is_contiguous() && o.is_contiguous(),
              "flash_attn_backward: tensors must be contiguous CUDA");
  const int64_t BH = q.size(0);
  const int L = (int)q.size(1);
  const SrcDesc bd = describe(bias, bias_outer_div, L, "bias");
  const SrcDesc md = describe(mask, mask_outer_div, L, "mask");

  const bool drop = dropped && dropout_p > 0.0;
  float pinv = 1.f;
  uint32_t pthresh = 0;
  if (drop) {
    const double pc = std::min(dropout_p, 0.999999);
    pinv = (float)(1.0 / (1.0 - pc));
    pthresh = keep16_threshold(pc);
  }
  const uint64_t seed = (uint64_t)seed_in;

  auto stream = at::cuda::getCurrentCUDAStream();
  auto di = at::empty({BH, (int64_t)L}, q.options().dtype(at::kFloat));
  const int64_t n_rows = BH * L;
  flash_dot_do_o_kernel<<<unicore_grid((n_rows + 31) / 32), 256, 0, stream>>>(
      di.data_ptr<float>(), reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
      reinterpret_cast<const uint16_t*>(o.data_ptr()), n_rows);

  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  // bias gradient, two strategies:
  //  * short L: materialize dS (BH, L, L) and reduce it over the broadcast
  //    batches — deterministic, and faster than atomics when the broadcast
  //    factor is large (measured: fp32 atomicAdd contention at L=512/B=96
  //    serializes the dq kernel ~4x);
  //  * long L (>= 2048): accumulate dS straight into the (nb, L, L) fp32
  //    buffer with atomics inside the dq kernel — the dS materialization
  //    stops fitting (BH*L*L*2 bytes) while the broadcast factor (and so
  //    the atomic contention) is small. This is what makes a TRAINABLE
  //    pair bias affordable at Uni-Fold-scale sequence lengths.
  // torch.use_deterministic_algorithms() forces the materialized path.
  at::Tensor ds, dbias;
  if (bd.ptr && bias_needs_grad) {
    TORCH_CHECK(bd.q == L, "flash_attn: bias grad requires bias_q == L");
    const bool fuse = L >= 2048 &&
                      !at::globalContext().deterministicAlgorithms();
    if (fuse)
      dbias = at::zeros({bd.nb, (int64_t)bd.q, (int64_t)L},
                        q.options().dtype(at::kFloat));
    else
      ds = at::empty({BH, (int64_t)L, (int64_t)L}, q.options());
  }
  float* dbias_ptr = dbias.defined() ? dbias.data_ptr<float>() : nullptr;
  const dim3 grid(L / BM, BH);

  auto launch_all = [&](auto hb, auto hm, auto dr) {
    constexpr bool HB = decltype(hb)::value;
    constexpr bool HM = decltype(hm)::value;
    constexpr bool DR = decltype(dr)::value;
    if (L <= 512 && L % 128 == 0)
      flash_bwd_dq_k32_kernel<HB, HM, DR, 512>
          <<<dim3(L / 128, BH), 256, 0, stream>>>(
              reinterpret_cast<uint16_t*>(dq.data_ptr()),
              ds.defined() ? reinterpret_cast<uint16_t*>(ds.data_ptr())
                           : nullptr,
              reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
              reinterpret_cast<const uint16_t*>(q.data_ptr()),
              reinterpret_cast<const uint16_t*>(k.data_ptr()),
              reinterpret_cast<const uint16_t*>(v.data_ptr()),
              lse.data_ptr<float>(), di.data_ptr<float>(),
              reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb, bd.q, bd.od,
              reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q, md.od, L,
              pinv, pthresh, seed);
    else if (L <= 512)
      flash_bwd_dq_kres_kernel<HB, HM, DR, 512><<<grid, 256, 0, stream>>>(
          reinterpret_cast<uint16_t*>(dq.data_ptr()),
          ds.defined() ? reinterpret_cast<uint16_t*>(ds.data_ptr()) : nullptr,
          reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
          reinterpret_cast<const uint16_t*>(q.data_ptr()),
          reinterpret_cast<const uint16_t*>(k.data_ptr()),
          reinterpret_cast<const uint16_t*>(v.data_ptr()), lse.data_ptr<float>(),
          di.data_ptr<float>(), reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb,
          bd.q, bd.od, reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q,
          md.od, L, pinv, pthresh, seed);
    else
      flash_bwd_dq_kernel<HB, HM, DR><<<grid, 256, 0, stream>>>(
        reinterpret_cast<uint16_t*>(dq.data_ptr()),
        ds.defined() ? reinterpret_cast<uint16_t*>(ds.data_ptr()) : nullptr,
        dbias_ptr,
        reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
        reinterpret_cast<const uint16_t*>(q.data_ptr()),
        reinterpret_cast<const uint16_t*>(k.data_ptr()),
        reinterpret_cast<const uint16_t*>(v.data_ptr()), lse.data_ptr<float>(),
        di.data_ptr<float>(), reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb,
        bd.q, bd.od, reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q,
        md.od, L, pinv, pthresh, seed);
    if (L <= 512)
      flash_bwd_dkv_qres_kernel<HB, HM, DR, 512><<<grid, 256, 0, stream>>>(
          reinterpret_cast<uint16_t*>(dk.data_ptr()),
          reinterpret_cast<uint16_t*>(dv.data_ptr()),
          reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
          reinterpret_cast<const uint16_t*>(q.data_ptr()),
          reinterpret_cast<const uint16_t*>(k.data_ptr()),
          reinterpret_cast<const uint16_t*>(v.data_ptr()), lse.data_ptr<float>(),
          di.data_ptr<float>(), reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb,
          bd.q, bd.od, reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q,
          md.od, L, pinv, pthresh, seed);
    else
      flash_bwd_dkv_kernel<HB, HM, DR><<<grid, 256, 0, stream>>>(
        reinterpret_cast<uint16_t*>(dk.data_ptr()),
        reinterpret_cast<uint16_t*>(dv.data_ptr()),
        reinterpret_cast<const uint16_t*>(d_out.data_ptr()),
        reinterpret_cast<const uint16_t*>(q.data_ptr()),
        reinterpret_cast<const uint16_t*>(k.data_ptr()),
        reinterpret_cast<const uint16_t*>(v.data_ptr()), lse.data_ptr<float>(),
        di.data_ptr<float>(), reinterpret_cast<const uint16_t*>(bd.ptr), bd.nb,
        bd.q, bd.od, reinterpret_cast<const uint16_t*>(md.ptr), md.nb, md.q,
        md.od, L, pinv, pthresh, seed);
  };
  auto pick = [&](auto hb, auto hm) {
    if (drop)
      launch_all(hb, hm, std::true_type{});
    else
      launch_all(hb, hm, std::false_type{});
  };
  if (bd.ptr && md.ptr)
    pick(std::true_type{}, std::true_type{});
  else if (bd.ptr)
    pick(std::true_type{}, std::false_type{});
  else if (md.ptr)
    pick(std::false_type{}, std::true_type{});
  else
    pick(std::false_type{}, std::false_type{});

  C10_CUDA_KERNEL_LAUNCH_CHECK();
  if (dbias.defined()) return {dq, dk, dv, dbias};
  if (ds.defined()) {
    // deterministic fallback: reduce the materialized dS over the
    // broadcast axes here (BH rows decompose as (outer, nb, od))
    const int64_t outer = BH / (bd.nb * bd.od);
    auto db = ds.view({outer, bd.nb, bd.od, (int64_t)L, (int64_t)L})
                  .sum(at::IntArrayRef{0, 2}, /*keepdim=*/false, at::kFloat);
    return {dq, dk, dv, db};
  }
  return {dq, dk, dv};
}
