// RoPE and fused SwiGLU kernels for gfx950.
//
// RoPE replaces the reference's unfused python rotate_half/apply_rotary
// (reference: projects/Llama/llama.py:31-64): cos/sin tables
// are host-precomputed (guide Appendix B: no on-device trig) and applied to
// strided [b, s, nh, hs] q/k views in one vectorized pass; backward is the
// same rotation with the sine negated.
//
// SwiGLU replaces the reference's fused_fast_gelu_mul-style gated MLP
// (reference: projects/MT5/layers/mlp_layer.py:123):
// y = silu(gate) * up with gate/up the two halves of one col-parallel
// projection; backward recomputes silu from the saved input.
#include "common.h"

namespace {

// ---------------------------------------------------------------------------
// RoPE: x[..., :hs/2], x[..., hs/2:] rotated pairwise with cos/sin[s, hs/2]
// half-split convention (HF Llama rotate_half):
//   out1 = x1*cos - x2*sin ;  out2 = x2*cos + x1*sin
// ---------------------------------------------------------------------------
template <class E, bool BWD>
__global__ void rope_kernel(const typename E::T* __restrict__ x,
                            typename E::T* __restrict__ out,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, int64_t sb, int64_t ss,
                            int64_t sh, int64_t ob, int64_t os, int64_t oh, int B,
                            int S, int NH, int HS, int pos0) {
  // one thread: VEC elems of x1 + matching VEC of x2
  constexpr int V = E::VEC;
  const int half = HS / 2;
  const int64_t total = (int64_t)B * S * NH * (half / V);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = (int)(i % (half / V));
    int64_t t = i / (half / V);
    const int h = (int)(t % NH);
    t /= NH;
    const int s = (int)(t % S);
    const int b = (int)(t / S);
    const typename E::T* xp = x + b * sb + (int64_t)s * ss + h * sh;
    typename E::T* op = out + b * ob + (int64_t)s * os + h * oh;
    const float* cp = cos_t + (int64_t)(pos0 + s) * half + c * V;
    const float* sp = sin_t + (int64_t)(pos0 + s) * half + c * V;
    using VecT = typename E::VecT;
    VecT x1 = *(const VecT*)(xp + c * V);
    VecT x2 = *(const VecT*)(xp + half + c * V);
    VecT o1, o2;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float cv = cp[j], sv = BWD ? -sp[j] : sp[j];
      float a = E::to_f(x1[j]), bb = E::to_f(x2[j]);
      o1[j] = E::from_f(a * cv - bb * sv);
      o2[j] = E::from_f(bb * cv + a * sv);
    }
    *(VecT*)(op + c * V) = o1;
    *(VecT*)(op + half + c * V) = o2;
  }
}

// ---------------------------------------------------------------------------
// Fused decode-step RoPE + KV-cache insert (hipGraph-captured serving step):
// one launch replaces per-layer {rope(q), rope(k), 2x index_copy, permutes}.
//   q   [B, 1, NH, HD]  (strided view, last dim contiguous) -> rotated into
//   qo  [B, NH, 1, HD]  contiguous (flash_decode input layout)
//   k   [B, 1, KVH, HD] -> rotated into ck[b, h, pos, :]
//   v   [B, 1, KVH, HD] -> copied  into cv[b, h, pos, :]
// pos is a DEVICE int64 pointer (position advances on-device under capture).
// One wavefront per (b, head) row; rows ordered q-heads, k-heads, v-heads.
// ---------------------------------------------------------------------------
template <class E, bool ROT>
__global__ void rope_kv_insert_kernel(
    const typename E::T* __restrict__ q, const typename E::T* __restrict__ k,
    const typename E::T* __restrict__ v, typename E::T* __restrict__ qo,
    typename E::T* __restrict__ ck, typename E::T* __restrict__ cv,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    const long long* __restrict__ pos_p, int B, int NH, int KVH, int HD,
    int64_t max_len, int64_t q_sb, int64_t q_sh, int64_t k_sb, int64_t k_sh,
    int64_t v_sb, int64_t v_sh, int per_batch_pos) {
  const int half = HD / 2;
  const int rows_per_b = NH + 2 * KVH;
  const int wave = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  const int lane = threadIdx.x & 63;
  if (wave >= B * rows_per_b) return;
  const int b = wave / rows_per_b;
  int r = wave % rows_per_b;
  // per-slot position (continuous batching: each sequence decodes at its
  // own offset); clamp guards an idle slot that over-advanced
  const int64_t pos = pos_p[per_batch_pos ? b : 0];
  if (pos < 0 || pos >= max_len) return;
  const float* cp = cos_t + pos * half;
  const float* sp = sin_t + pos * half;
  if (r < NH) {  // q head: rotate (or copy) -> qo[b, r, 0, :]
    const typename E::T* xp = q + b * q_sb + r * q_sh;
    typename E::T* op = qo + ((int64_t)b * NH + r) * HD;
    if constexpr (ROT) {
      for (int j = lane; j < half; j += 64) {
        float a = E::to_f(xp[j]), bb = E::to_f(xp[half + j]);
        op[j] = E::from_f(a * cp[j] - bb * sp[j]);
        op[half + j] = E::from_f(bb * cp[j] + a * sp[j]);
      }
    } else {
      for (int j = lane; j < HD; j += 64) op[j] = xp[j];
    }
  } else if (r < NH + KVH) {  // k head: rotate (or copy) -> ck[b, h, pos, :]
    const int h = r - NH;
    const typename E::T* xp = k + b * k_sb + h * k_sh;
    typename E::T* op = ck + (((int64_t)b * KVH + h) * max_len + pos) * HD;
    if constexpr (ROT) {
      for (int j = lane; j < half; j += 64) {
        float a = E::to_f(xp[j]), bb = E::to_f(xp[half + j]);
        op[j] = E::from_f(a * cp[j] - bb * sp[j]);
        op[half + j] = E::from_f(bb * cp[j] + a * sp[j]);
      }
    } else {
      for (int j = lane; j < HD; j += 64) op[j] = xp[j];
    }
  } else {  // v head: copy -> cv[b, h, pos, :]
    const int h = r - NH - KVH;
    const typename E::T* xp = v + b * v_sb + h * v_sh;
    typename E::T* op = cv + (((int64_t)b * KVH + h) * max_len + pos) * HD;
    for (int j = lane; j < HD; j += 64) op[j] = xp[j];
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: y = silu(g) * u, with g = x[..., :F], u = x[..., F:2F]
// ---------------------------------------------------------------------------
template <class E, bool BWD>
__global__ void swiglu_kernel(const typename E::T* __restrict__ x,
                              const typename E::T* __restrict__ dy,
                              typename E::T* __restrict__ out, int64_t rows, int F) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int fv = F / V;
  const int64_t total = rows * fv;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = i / fv;
    const int c = (int)(i % fv);
    const typename E::T* xr = x + r * (2 * (int64_t)F);
    VecT g = *(const VecT*)(xr + c * V);
    VecT u = *(const VecT*)(xr + F + c * V);
    if (!BWD) {
      VecT o;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float gv = E::to_f(g[j]);
        float sig = 1.0f / (1.0f + __expf(-gv));
        o[j] = E::from_f(gv * sig * E::to_f(u[j]));
      }
      ((VecT*)(out + r * (int64_t)F))[c] = o;
    } else {
      VecT d = ((const VecT*)(dy + r * (int64_t)F))[c];
      VecT dg,duv;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float gv = E::to_f(g[j]);
        float uv = E::to_f(u[j]);
        float dv = E::to_f(d[j]);
        float sig = 1.0f / (1.0f + __expf(-gv));
        float silu = gv * sig;
        float dsilu = sig * (1.0f + gv * (1.0f - sig));
        dg[j] = E::from_f(dv * uv * dsilu);
        duv[j] = E::from_f(dv * silu);
      }
      typename E::T* dxr = out + r * (2 * (int64_t)F);
      *(VecT*)(dxr + c * V) = dg;
      *(VecT*)(dxr + F + c * V) = duv;
    }
  }
}

inline int64_t grid_for(int64_t n) {
  int64_t g = CDIV(n, 256);
  return g < 4096 ? g : 4096;
}

}  // namespace

#define ROPE_LAUNCHERS(SUFF, ETYPE)                                                    \
  extern "C" void rope_fwd_##SUFF(const void* x, void* out, const float* cos_t,        \
                                  const float* sin_t, int64_t sb, int64_t ss,          \
                                  int64_t sh, int64_t ob, int64_t os, int64_t oh,      \
                                  int B, int S, int NH, int HS, int pos0,              \
                                  hipStream_t stream) {                                \
    int64_t n = (int64_t)B * S * NH * (HS / 2 / ETYPE::VEC);                           \
    rope_kernel<ETYPE, false><<<dim3((uint32_t)grid_for(n)), 256, 0, stream>>>(        \
        (const ETYPE::T*)x, (ETYPE::T*)out, cos_t, sin_t, sb, ss, sh, ob, os, oh, B,   \
        S, NH, HS, pos0);                                                              \
  }                                                                                    \
  extern "C" void rope_bwd_##SUFF(const void* dy, void* dx, const float* cos_t,        \
                                  const float* sin_t, int64_t sb, int64_t ss,          \
                                  int64_t sh, int64_t ob, int64_t os, int64_t oh,      \
                                  int B, int S, int NH, int HS, int pos0,              \
                                  hipStream_t stream) {                                \
    int64_t n = (int64_t)B * S * NH * (HS / 2 / ETYPE::VEC);                           \
    rope_kernel<ETYPE, true><<<dim3((uint32_t)grid_for(n)), 256, 0, stream>>>(         \
        (const ETYPE::T*)dy, (ETYPE::T*)dx, cos_t, sin_t, sb, ss, sh, ob, os, oh, B,   \
        S, NH, HS, pos0);                                                              \
  }                                                                                    \
  extern "C" void swiglu_fwd_##SUFF(const void* x, void* y, int64_t rows, int F,       \
                                    hipStream_t stream) {                              \
    swiglu_kernel<ETYPE, false>                                                        \
        <<<dim3((uint32_t)grid_for(rows * (F / ETYPE::VEC))), 256, 0, stream>>>(       \
            (const ETYPE::T*)x, nullptr, (ETYPE::T*)y, rows, F);                       \
  }                                                                                    \
  extern "C" void swiglu_bwd_##SUFF(const void* x, const void* dy, void* dx,           \
                                    int64_t rows, int F, hipStream_t stream) {         \
    swiglu_kernel<ETYPE, true>                                                         \
        <<<dim3((uint32_t)grid_for(rows * (F / ETYPE::VEC))), 256, 0, stream>>>(       \
            (const ETYPE::T*)x, (const ETYPE::T*)dy, (ETYPE::T*)dx, rows, F);          \
  }

ROPE_LAUNCHERS(bf16, BF16Elem)
ROPE_LAUNCHERS(f32, F32Elem)

extern "C" void rope_kv_insert_bf16(
    const void* q, const void* k, const void* v, void* qo, void* ck, void* cv,
    const float* cos_t, const float* sin_t, const long long* pos_p, int B,
    int NH, int KVH, int HD, int64_t max_len, int64_t q_sb, int64_t q_sh,
    int64_t k_sb, int64_t k_sh, int64_t v_sb, int64_t v_sh,
    int rotate, int per_batch_pos, hipStream_t stream) {
  const int waves = B * (NH + 2 * KVH);
  const int blocks = CDIV(waves, 4);  // 4 waves (256 threads) per block
  if (rotate)
    rope_kv_insert_kernel<BF16Elem, true><<<dim3(blocks), 256, 0, stream>>>(
        (const BF16Elem::T*)q, (const BF16Elem::T*)k, (const BF16Elem::T*)v,
        (BF16Elem::T*)qo, (BF16Elem::T*)ck, (BF16Elem::T*)cv, cos_t, sin_t,
        pos_p, B, NH, KVH, HD, max_len, q_sb, q_sh, k_sb, k_sh, v_sb, v_sh,
        per_batch_pos);
  else
    rope_kv_insert_kernel<BF16Elem, false><<<dim3(blocks), 256, 0, stream>>>(
        (const BF16Elem::T*)q, (const BF16Elem::T*)k, (const BF16Elem::T*)v,
        (BF16Elem::T*)qo, (BF16Elem::T*)ck, (BF16Elem::T*)cv, cos_t, sin_t,
        pos_p, B, NH, KVH, HD, max_len, q_sb, q_sh, k_sb, k_sh, v_sb, v_sh,
        per_batch_pos);
}
