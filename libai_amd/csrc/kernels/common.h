// Common device helpers for libai_amd CDNA4 (gfx950) kernels.
// Pure HIP — no torch headers; compiled by hipcc --offload-arch=gfx950 only.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64  // CDNA wavefront width (hard-coded per gfx950 guide)

// ---------------------------------------------------------------------------
// bf16 <-> f32 bit helpers (we traffic bf16 as uint16_t to keep loads
// vectorizable as ushortN; hipcc does not auto-vectorize __hip_bfloat16).
// ---------------------------------------------------------------------------
__device__ __forceinline__ float bf2f(uint16_t h) {
  union {
    uint32_t u;
    float f;
  } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t f2bf(float f) {
  union {
    float f;
    uint32_t u;
  } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.u >> 16) & 1u;
  uint32_t rounded = v.u + 0x7fffu + lsb;
  if ((v.u & 0x7f800000u) == 0x7f800000u) rounded = v.u;  // inf/nan passthrough
  return (uint16_t)(rounded >> 16);
}

// 8 x bf16 = one 16-byte load (the coalescing sweet spot per guide G13)
typedef uint16_t u16x8 __attribute__((ext_vector_type(8)));
typedef uint16_t u16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

// ---------------------------------------------------------------------------
// Generic element type adapters: kernels are templated on ET in {BF16, F32}.
// Each provides VecT (16-byte vector of 8/4 elems), load->float, store<-float.
// ---------------------------------------------------------------------------
struct BF16Elem {
  using T = uint16_t;
  static constexpr int VEC = 8;  // elems per 16-byte vector
  using VecT = u16x8;
  static __device__ __forceinline__ float to_f(T x) { return bf2f(x); }
  static __device__ __forceinline__ T from_f(float x) { return f2bf(x); }
};

struct F32Elem {
  using T = float;
  static constexpr int VEC = 4;
  using VecT = f32x4;
  static __device__ __forceinline__ float to_f(T x) { return x; }
  static __device__ __forceinline__ T from_f(float x) { return x; }
};

// ---------------------------------------------------------------------------
// Wave + block reductions
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-wide reduce for blockDim.x <= 1024 (<= 16 waves). `lds` needs 16 floats.
template <typename Op>
__device__ __forceinline__ float block_reduce(float v, float* lds, Op op, float ident) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : ident;
  if (wid == 0) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
    if (lane == 0) lds[0] = v;
  }
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

struct SumOp {
  __device__ float operator()(float a, float b) const { return a + b; }
};
struct MaxOp {
  __device__ float operator()(float a, float b) const { return fmaxf(a, b); }
};

// ---------------------------------------------------------------------------
// Philox4x32-10 counter-based RNG (for recompute-in-backward dropout masks).
// Stateless: mask(seed, idx) is identical in fwd and bwd, so no mask tensor
// round-trips through HBM (the fused op stays 2-tensor-traffic).
// ---------------------------------------------------------------------------
__device__ __forceinline__ void philox_round(uint32_t& c0, uint32_t& c1, uint32_t& c2,
                                             uint32_t& c3, uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t h0 = __umulhi(M0, c0), l0 = M0 * c0;
  uint32_t h1 = __umulhi(M1, c2), l1 = M1 * c2;
  uint32_t n0 = h1 ^ c1 ^ k0;
  uint32_t n1 = l1;
  uint32_t n2 = h0 ^ c3 ^ k1;
  uint32_t n3 = l0;
  c0 = n0;
  c1 = n1;
  c2 = n2;
  c3 = n3;
}

// 4 uniform u32 from (seed, ctr)
__device__ __forceinline__ void philox4(uint64_t seed, uint64_t ctr, uint32_t out[4]) {
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint32_t c0 = (uint32_t)ctr, c1 = (uint32_t)(ctr >> 32), c2 = 0x9E3779B9u, c3 = 0xBB67AE85u;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  out[0] = c0;
  out[1] = c1;
  out[2] = c2;
  out[3] = c3;
}

__device__ __forceinline__ float u32_to_uniform(uint32_t x) {
  // (0, 1]
  return (x >> 8) * (1.0f / 16777216.0f);
}

// Cheap per-element counter RNG: used by attention dropout where fwd and bwd
// kernels index elements in DIFFERENT lane layouts, so a per-element (not
// per-4-group) generator keeps regeneration cheap on both sides.
// 32-BIT ops only: 64-bit integer multiplies are emulated on gfx950 and a
// splitmix64 here measured ~25 VALU/element (PMC: 104 issued instructions per
// MFMA in flash_fwd).  This lowCbias mixer is ~9 VALU.
__device__ __forceinline__ uint32_t rnd_hash2(uint64_t seed, uint32_t lo,
                                               uint32_t hi) {
  uint32_t h = (uint32_t)seed ^ (lo * 0x9E3779B9u) ^ (hi * 0x85EBCA6Bu) ^
               (uint32_t)(seed >> 32);
  h ^= h >> 16;
  h *= 0x7FEB352Du;
  h ^= h >> 15;
  h *= 0x846CA68Bu;
  h ^= h >> 16;
  return h;
}

__device__ __forceinline__ uint32_t rnd_hash(uint64_t seed, uint64_t idx) {
  return rnd_hash2(seed, (uint32_t)idx, (uint32_t)(idx >> 32));
}

// byte-granular keep threshold: p quantized to 1/256 (one hash draw yields
// 4 keep decisions via its bytes)
__device__ __forceinline__ uint32_t drop_threshold_u8(float p) {
  float t = p * 256.0f + 0.5f;
  return t >= 255.f ? 255u : (uint32_t)t;
}

// keep-decision without the float conversion: compare the hash against a
// precomputed uint32 threshold = p * 2^32.
__device__ __forceinline__ uint32_t drop_threshold_u32(float p) {
  double t = (double)p * 4294967296.0;
  return (t >= 4294967295.0) ? 0xFFFFFFFFu : (uint32_t)t;
}

// ---------------------------------------------------------------------------
// gelu (tanh approximation, matching torch.nn.functional.gelu(approximate="tanh"))
// and its derivative; plus exact-erf gelu to match torch's default.
// ---------------------------------------------------------------------------
// fast erf (Abramowitz & Stegun 7.1.26, |err| <= 1.5e-7): ocml's erff is a
// long polynomial and made the fused bias-gelu pass VALU-bound instead of
// HBM-bound.
__device__ __forceinline__ float fast_erff(float x) {
  const float a1 = 0.254829592f, a2 = -0.284496736f, a3 = 1.421413741f;
  const float a4 = -1.453152027f, a5 = 1.061405429f, p = 0.3275911f;
  float ax = fabsf(x);
  float t = 1.0f / fmaf(p, ax, 1.0f);
  float poly = t * fmaf(t, fmaf(t, fmaf(t, fmaf(t, a5, a4), a3), a2), a1);
  float y = 1.0f - poly * __expf(-ax * ax);
  return copysignf(y, x);
}

__device__ __forceinline__ float gelu_erf(float x) {
  return 0.5f * x * (1.0f + fast_erff(x * 0.70710678118654752440f));
}

__device__ __forceinline__ float gelu_erf_grad(float x) {
  const float kInvSqrt2 = 0.70710678118654752440f;
  const float kInvSqrt2Pi = 0.3989422804014327f;
  float cdf = 0.5f * (1.0f + fast_erff(x * kInvSqrt2));
  float pdf = kInvSqrt2Pi * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

#define CDIV(a, b) (((a) + (b)-1) / (b))
