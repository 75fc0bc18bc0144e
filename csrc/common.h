// Common device helpers for the unicore_amd gfx950 (CDNA4) kernels.
//
// Design notes (MI355X):
//  - wavefront = 64 lanes; all cross-lane reductions are 6-step __shfl_xor
//    trees over the full wave.
//  - every memory-bound kernel loads 16 B/lane (8 bf16/fp16 elems or two
//    float4 for fp32) — scalar 2-byte loads are ~2x slower on this chip.
//  - Philox4x32-10 is implemented here directly (no cuRAND/hipRAND device
//    lib); seed/offset come from PyTorch's CUDA generator so the
//    seed-determinism contract of the reference framework
//    (unicore/utils.torch_seed) is preserved at the API level.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define UNICORE_WAVE 64

// ---------------------------------------------------------------------------
// dtype conversion traits
// ---------------------------------------------------------------------------
template <typename T>
struct Cvt;

template <>
struct Cvt<float> {
  static __device__ __forceinline__ float to_f(float x) { return x; }
  static __device__ __forceinline__ float from_f(float x) { return x; }
};

template <>
struct Cvt<__half> {
  static __device__ __forceinline__ float to_f(__half x) { return __half2float(x); }
  static __device__ __forceinline__ __half from_f(float x) { return __float2half(x); }
};

template <>
struct Cvt<__hip_bfloat16> {
  static __device__ __forceinline__ float to_f(__hip_bfloat16 x) {
    return __bfloat162float(x);
  }
  static __device__ __forceinline__ __hip_bfloat16 from_f(float x) {
    return __float2bfloat16(x);
  }
};

// ---------------------------------------------------------------------------
// vectorized 8-element load/store (16 B/lane for 2-byte dtypes, 32 B for f32)
// pointers must be 16-byte aligned (guaranteed when the row length is a
// multiple of 8 elements and the base comes from the caching allocator).
// ---------------------------------------------------------------------------
template <typename T>
__device__ __forceinline__ void load8(const T* p, float (&f)[8]) {
  if constexpr (sizeof(T) == 2) {
    union {
      uint4 u;
      T t[8];
    } U;
    U.u = *reinterpret_cast<const uint4*>(p);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = Cvt<T>::to_f(U.t[j]);
  } else {
    const float4 a = reinterpret_cast<const float4*>(p)[0];
    const float4 b = reinterpret_cast<const float4*>(p)[1];
    f[0] = a.x; f[1] = a.y; f[2] = a.z; f[3] = a.w;
    f[4] = b.x; f[5] = b.y; f[6] = b.z; f[7] = b.w;
  }
}

// NV-element variant (NV in {2,4,8}); same alignment contract as load8
// scaled down: the byte width of the load (NV * sizeof(T)) must divide the
// pointer's offset from a 16-byte-aligned base.
template <typename T, int NV>
__device__ __forceinline__ void loadN(const T* p, float (&f)[NV]) {
  static_assert(NV == 2 || NV == 4 || NV == 8, "loadN: NV must be 2/4/8");
  if constexpr (NV == 8) {
    load8(p, f);
  } else if constexpr (sizeof(T) == 2) {
    union {
      uint2 u2;
      unsigned u1;
      T t[NV];
    } U;
    if constexpr (NV == 4)
      U.u2 = *reinterpret_cast<const uint2*>(p);
    else
      U.u1 = *reinterpret_cast<const unsigned*>(p);
#pragma unroll
    for (int j = 0; j < NV; ++j) f[j] = Cvt<T>::to_f(U.t[j]);
  } else if constexpr (NV == 4) {
    const float4 a = *reinterpret_cast<const float4*>(p);
    f[0] = a.x; f[1] = a.y; f[2] = a.z; f[3] = a.w;
  } else {
    const float2 a = *reinterpret_cast<const float2*>(p);
    f[0] = a.x; f[1] = a.y;
  }
}

template <typename T>
__device__ __forceinline__ void store8(T* p, const float (&f)[8]) {
  if constexpr (sizeof(T) == 2) {
    union {
      uint4 u;
      T t[8];
    } U;
#pragma unroll
    for (int j = 0; j < 8; ++j) U.t[j] = Cvt<T>::from_f(f[j]);
    *reinterpret_cast<uint4*>(p) = U.u;
  } else {
    float4 a, b;
    a.x = f[0]; a.y = f[1]; a.z = f[2]; a.w = f[3];
    b.x = f[4]; b.y = f[5]; b.z = f[6]; b.w = f[7];
    reinterpret_cast<float4*>(p)[0] = a;
    reinterpret_cast<float4*>(p)[1] = b;
  }
}

// ---------------------------------------------------------------------------
// wave64 reductions
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---------------------------------------------------------------------------
// Philox4x32-10 counter-based RNG (own implementation; not bitwise-identical
// to cuRAND but honours PyTorch's (seed, subsequence, offset) contract).
// ---------------------------------------------------------------------------
struct Philox4 {
  uint2 key;
  uint4 ctr;

  __device__ Philox4(uint64_t seed, uint64_t subseq, uint64_t offset) {
    key.x = (uint32_t)seed;
    key.y = (uint32_t)(seed >> 32);
    ctr.x = (uint32_t)offset;
    ctr.y = (uint32_t)(offset >> 32);
    ctr.z = (uint32_t)subseq;
    ctr.w = (uint32_t)(subseq >> 32);
  }

  static __device__ __forceinline__ uint32_t mulhilo(uint32_t a, uint32_t b,
                                                     uint32_t* hi) {
    const uint64_t p = (uint64_t)a * (uint64_t)b;
    *hi = (uint32_t)(p >> 32);
    return (uint32_t)p;
  }

  // one 4x32 draw; advances the counter for the next call
  __device__ __forceinline__ uint4 next() {
    uint4 c = ctr;
    uint2 k = key;
#pragma unroll
    for (int i = 0; i < 10; ++i) {
      uint32_t hi0, hi1;
      const uint32_t lo0 = mulhilo(0xD2511F53u, c.x, &hi0);
      const uint32_t lo1 = mulhilo(0xCD9E8D57u, c.z, &hi1);
      c.x = hi1 ^ c.y ^ k.x;
      c.y = lo1;
      c.z = hi0 ^ c.w ^ k.y;
      c.w = lo0;
      k.x += 0x9E3779B9u;
      k.y += 0xBB67AE85u;
    }
    if (++ctr.x == 0) ++ctr.y;
    return c;
  }
};

// 8 dropout keep-decisions from ONE philox draw (16-bit threshold
// resolution — plenty for dropout probabilities).  Definition shared by
// every dropout producer AND the flash-attention backward recompute:
//   keep(subseq, idx) = u16 #(idx%8) of philox(seed, subseq, idx/8)
__device__ __forceinline__ void keep16x8(uint64_t seed, uint64_t subseq,
                                         int idx0 /* multiple of 8 */,
                                         uint32_t pthresh16, bool (&keep)[8]) {
  Philox4 ph(seed, subseq, (uint64_t)(idx0 >> 3));
  const uint4 r = ph.next();
  const uint32_t rr[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
  for (int j = 0; j < 8; ++j)
    keep[j] = ((rr[j >> 1] >> ((j & 1) * 16)) & 0xFFFFu) >= pthresh16;
}

static inline uint32_t keep16_threshold(double p) {
  const double pc = p < 0.999999 ? p : 0.999999;
  const double t = pc * 65536.0;
  return t > 65535.0 ? 65535u : (uint32_t)t;
}

// bf16 <-> bits helpers (avoid relying on __bfloat16_as_ushort availability)
__device__ __forceinline__ uint16_t f32_to_bf16_bits(float x) {
  union {
    __hip_bfloat16 b;
    uint16_t u;
  } U;
  U.b = __float2bfloat16(x);
  return U.u;
}

__device__ __forceinline__ float bf16_bits_to_f32(uint16_t u) {
  union {
    __hip_bfloat16 b;
    uint16_t u16;
  } U;
  U.u16 = u;
  return __bfloat162float(U.b);
}

// grid sizing: cap at 8 blocks per CU (256 CUs) and grid-stride the rest
static inline int unicore_grid(int64_t want, int cap = 2048) {
  if (want < 1) return 1;
  return (int)(want < cap ? want : cap);
}

// ---------------------------------------------------------------------------
// Deterministic per-block column sums for flat elementwise kernels over a
// (rows, C) tensor.  The host picks a grid with stride*8 % C == 0 so each
// thread owns 8 fixed columns; the block fold serializes the (at most
// ceil(2048/C)) contributor bands so no two threads ever add to the same
// LDS slot in the same round.
// ---------------------------------------------------------------------------
static inline long long unicore_gcd_ll(long long a, long long b) {
  while (b) {
    long long t = a % b;
    a = b;
    b = t;
  }
  return a;
}

static inline bool colsum_supported(int64_t C) {
  if (C <= 0 || C % 8 != 0 || C > 4096) return false;
  return (C / unicore_gcd_ll(2048, C)) <= 256;
}

static inline int colsum_grid(int64_t n8, int64_t C) {
  const int64_t mult = C / unicore_gcd_ll(2048, C);
  const int64_t want = (n8 + 255) / 256;
  int64_t blocks = ((want + mult - 1) / mult) * mult;
  int64_t cap = (2048 / mult) * mult;
  if (cap < mult) cap = mult;
  if (blocks > cap) blocks = cap;
  return (int)blocks;
}

// acc[8] holds this thread's partial sums for columns c0..c0+7 (c0 < 0 =>
// thread had no elements); s_col is C floats of LDS.
__device__ __forceinline__ void colsum_block_fold(const float (&acc)[8], int c0,
                                                  int C, float* s_col,
                                                  float* partials_row) {
  for (int c = (int)threadIdx.x; c < C; c += (int)blockDim.x) s_col[c] = 0.f;
  __syncthreads();
  const int band = ((int)threadIdx.x * 8) / C;
  const int nbands = ((int)blockDim.x * 8 + C - 1) / C;
  for (int m = 0; m < nbands; ++m) {
    if (band == m && c0 >= 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) s_col[c0 + j] += acc[j];
    }
    __syncthreads();
  }
  for (int c = (int)threadIdx.x; c < C; c += (int)blockDim.x)
    partials_row[c] = s_col[c];
}

// stage A of the two-stage column fold (see fold.h): coalesced chunk sums
// (nb, C) -> (gridDim.y, C). Lanes own consecutive columns, so every wave
// load is one contiguous 256 B segment — the one-block-per-column final
// kernel below reads columns as strided 4 B gathers instead, which held the
// single-stage fold to 1.5 TB/s on 12k+ partial rows (PMC evidence).
static __global__ void unicore_col_fold_stage_kernel(
    const float* __restrict__ in, float* __restrict__ out, int nb, int chunk,
    int C) {
  const int c = blockIdx.x * 256 + (int)threadIdx.x;
  if (c >= C) return;
  const int r0 = blockIdx.y * chunk;
  const int r1 = min(r0 + chunk, nb);
  float s = 0.f;
  int r = r0;
  for (; r + 4 <= r1; r += 4)
    s += (in[(int64_t)r * C + c] + in[(int64_t)(r + 1) * C + c]) +
         (in[(int64_t)(r + 2) * C + c] + in[(int64_t)(r + 3) * C + c]);
  for (; r < r1; ++r) s += in[(int64_t)r * C + c];
  out[(int64_t)blockIdx.y * C + c] = s;
}

// partials (nb, C) -> out (C); one block per column, deterministic.
static __global__ void unicore_col_fold_kernel(const float* __restrict__ partials,
                                               float* __restrict__ out, int nb,
                                               int C) {
  __shared__ float s_red[4];
  const int c = blockIdx.x;
  if (c >= C) return;
  float v = 0.f;
  for (int r = (int)threadIdx.x; r < nb; r += (int)blockDim.x)
    v += partials[(int64_t)r * C + c];
  v = wave_sum(v);
  const int wave = (int)threadIdx.x / 64, lane = (int)threadIdx.x % 64;
  if (lane == 0) s_red[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = 0.f;
    for (int w = 0; w < (int)blockDim.x / 64; ++w) r += s_red[w];
    out[c] = r;
  }
}
