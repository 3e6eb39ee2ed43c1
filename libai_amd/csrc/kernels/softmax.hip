// Fused scale + mask + softmax (+ dropout) forward/backward for gfx950.
//
// Replaces the reference's flow._C.fused_scale_tril_softmax_mask_scale
// (causal, libai/layers/attention.py:240-246) and
// flow._C.fused_scale_mask_softmax_dropout (padding, attention.py:221-226).
//
// Rows are register-cached (one block per row, up to 4 x 16B vectors per
// thread), so the row is read from HBM exactly once per pass.  The backward
// recomputes the softmax from the saved *input* scores (one extra exp pass
// instead of saving the probability tensor), and dropout masks are philox-
// recomputed — forward traffic is read-S + write-P only.
//
// Causal masking is an in-kernel predicate (no materialized tril tensor —
// replaces reference K14, libai/models/gpt_model.py:44-51); padding masks are
// additive -10000 like the reference.
#include "common.h"

namespace {

constexpr int MAX_VPT = 4;  // vectors (16B) per thread cached in registers

template <class E, bool CAUSAL, bool GRAD>
__global__ void softmax_kernel(const typename E::T* __restrict__ s,
                               const typename E::T* __restrict__ dout,
                               const uint8_t* __restrict__ pad_mask,  // [B, SQ, SK], 1=masked
                               typename E::T* __restrict__ out, int NH, int SQ, int SK,
                               float scale, float p, uint64_t seed) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  __shared__ float red[16];

  const int64_t row = blockIdx.x;
  const int qi = (int)(row % SQ);
  const int64_t b = row / ((int64_t)NH * SQ);
  const int limit = CAUSAL ? (qi + SK - SQ) : (SK - 1);  // last valid col
  const T* sr = s + row * SK;
  T* outr = out + row * SK;
  const uint8_t* mrow = pad_mask ? pad_mask + (b * SQ + qi) * (int64_t)SK : nullptr;

  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  const int nvec = SK / V;
  const int vpt = CDIV(nvec, nth);

  float f[MAX_VPT][V];
  float m = -3.0e38f;
#pragma unroll
  for (int q = 0; q < MAX_VPT; ++q) {
    if (q >= vpt) break;
    const int iv = tid + q * nth;
    if (iv >= nvec) break;
    const int j0 = iv * V;
    if (CAUSAL && j0 > limit) {
#pragma unroll
      for (int j = 0; j < V; ++j) f[q][j] = -3.0e38f;
      continue;
    }
    VecT v = ((const VecT*)sr)[iv];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float x = E::to_f(v[j]) * scale;
      if (mrow && mrow[j0 + j]) x -= 10000.0f;
      if (CAUSAL && (j0 + j) > limit) x = -3.0e38f;
      f[q][j] = x;
      m = fmaxf(m, x);
    }
  }
  m = block_reduce(m, red, MaxOp(), -3.0e38f);

  float sum = 0.f;
#pragma unroll
  for (int q = 0; q < MAX_VPT; ++q) {
    if (q >= vpt) break;
    const int iv = tid + q * nth;
    if (iv >= nvec) break;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float e = (f[q][j] <= -3.0e38f) ? 0.f : __expf(f[q][j] - m);
      f[q][j] = e;
      sum += e;
    }
  }
  sum = block_reduce(sum, red, SumOp(), 0.f);
  const float inv = 1.0f / sum;  // every row has >=1 valid col (causal j<=qi)
  const float keep_scale = 1.0f / (1.0f - p);

  if (!GRAD) {
#pragma unroll
    for (int q = 0; q < MAX_VPT; ++q) {
      if (q >= vpt) break;
      const int iv = tid + q * nth;
      if (iv >= nvec) break;
      VecT o;
      if (p > 0.f) {
        uint32_t r[4];
#pragma unroll
        for (int h = 0; h < V / 4; ++h) {
          philox4(seed, (uint64_t)(row * nvec + iv) * (V / 4) + h, r);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float k = (u32_to_uniform(r[j]) > p) ? keep_scale : 0.f;
            o[h * 4 + j] = E::from_f(f[q][h * 4 + j] * inv * k);
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < V; ++j) o[j] = E::from_f(f[q][j] * inv);
      }
      ((VecT*)outr)[iv] = o;
    }
    return;
  }

  // backward: f now holds exp(S*scale - m); P = f * inv.
  // dP = dO * k (dropout);  dS = scale * P * (dP - rowsum(dP * P))
  float dpv[MAX_VPT][V];
  float dot = 0.f;
#pragma unroll
  for (int q = 0; q < MAX_VPT; ++q) {
    if (q >= vpt) break;
    const int iv = tid + q * nth;
    if (iv >= nvec) break;
    VecT vdo = ((const VecT*)(dout + row * SK))[iv];
    if (p > 0.f) {
      uint32_t r[4];
#pragma unroll
      for (int h = 0; h < V / 4; ++h) {
        philox4(seed, (uint64_t)(row * nvec + iv) * (V / 4) + h, r);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float k = (u32_to_uniform(r[j]) > p) ? keep_scale : 0.f;
          dpv[q][h * 4 + j] = E::to_f(vdo[h * 4 + j]) * k;
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) dpv[q][j] = E::to_f(vdo[j]);
    }
#pragma unroll
    for (int j = 0; j < V; ++j) dot += dpv[q][j] * f[q][j] * inv;
  }
  dot = block_reduce(dot, red, SumOp(), 0.f);

#pragma unroll
  for (int q = 0; q < MAX_VPT; ++q) {
    if (q >= vpt) break;
    const int iv = tid + q * nth;
    if (iv >= nvec) break;
    VecT o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float pij = f[q][j] * inv;
      o[j] = E::from_f(scale * pij * (dpv[q][j] - dot));
    }
    ((VecT*)outr)[iv] = o;
  }
}

inline int softmax_threads(int SK, int vec) {
  int nvec = SK / vec;
  int th = CDIV(nvec, MAX_VPT);
  // round up to a wave multiple, clamp to [64, 1024]
  th = ((th + WAVE - 1) / WAVE) * WAVE;
  if (th < 64) th = 64;
  if (th > 1024) th = 1024;
  return th;
}

}  // namespace

#define SOFTMAX_LAUNCHERS(SUFF, ETYPE)                                                  \
  extern "C" void softmax_fwd_##SUFF(const void* s, const uint8_t* pad_mask, void* out, \
                                     int64_t B, int NH, int SQ, int SK, float scale,    \
                                     float p, uint64_t seed, bool causal,               \
                                     hipStream_t stream) {                              \
    int th = softmax_threads(SK, ETYPE::VEC);                                           \
    dim3 grid((uint32_t)(B * NH * SQ));                                                 \
    if (causal)                                                                         \
      softmax_kernel<ETYPE, true, false><<<grid, dim3(th), 0, stream>>>(                \
          (const ETYPE::T*)s, nullptr, pad_mask, (ETYPE::T*)out, NH, SQ, SK, scale, p,  \
          seed);                                                                        \
    else                                                                                \
      softmax_kernel<ETYPE, false, false><<<grid, dim3(th), 0, stream>>>(               \
          (const ETYPE::T*)s, nullptr, pad_mask, (ETYPE::T*)out, NH, SQ, SK, scale, p,  \
          seed);                                                                        \
  }                                                                                     \
  extern "C" void softmax_bwd_##SUFF(const void* s, const void* dout,                   \
                                     const uint8_t* pad_mask, void* ds, int64_t B,      \
                                     int NH, int SQ, int SK, float scale, float p,      \
                                     uint64_t seed, bool causal, hipStream_t stream) {  \
    int th = softmax_threads(SK, ETYPE::VEC);                                           \
    dim3 grid((uint32_t)(B * NH * SQ));                                                 \
    if (causal)                                                                         \
      softmax_kernel<ETYPE, true, true><<<grid, dim3(th), 0, stream>>>(                 \
          (const ETYPE::T*)s, (const ETYPE::T*)dout, pad_mask, (ETYPE::T*)ds, NH, SQ,   \
          SK, scale, p, seed);                                                          \
    else                                                                                \
      softmax_kernel<ETYPE, false, true><<<grid, dim3(th), 0, stream>>>(                \
          (const ETYPE::T*)s, (const ETYPE::T*)dout, pad_mask, (ETYPE::T*)ds, NH, SQ,   \
          SK, scale, p, seed);                                                          \
  }

SOFTMAX_LAUNCHERS(bf16, BF16Elem)
SOFTMAX_LAUNCHERS(f32, F32Elem)
