// Vocab-parallel cross-entropy for gfx950.
//
// Replaces the reference's flow._C.sparse_softmax_cross_entropy call on
// [S(0), S(2)]-split logits (reference: libai/layers/cross_entropy.py:26-48):
// the full softmax is never materialized.  The forward is a SINGLE pass over
// the local logits shard computing (online, flash-style) the local row max,
// the sum of exponentials w.r.t. that local max, and the target logit when it
// falls in this rank's vocab shard.  The TP reduction (max, then corrected
// sumexp + target logit) happens in Python over [R]-sized tensors:
//   sumexp_global = allreduce_sum(sumexp_local * exp(max_local - max_global))
// Backward is one read+write pass: d logits = g * (softmax - onehot).
#include "common.h"

namespace {

template <class E>
__global__ void ce_fwd_kernel(const typename E::T* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              float* __restrict__ lmax, float* __restrict__ lsumexp,
                              float* __restrict__ tlogit, int64_t Vl,
                              int64_t vocab_start, int64_t ignore_index) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  __shared__ float red[16];

  const int64_t row = blockIdx.x;
  const typename E::T* lr = logits + row * Vl;
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  const int64_t nvec = Vl / V;
  const int64_t tgt = targets[row];
  const int64_t local_tgt =
      (tgt != ignore_index && tgt >= vocab_start && tgt < vocab_start + Vl)
          ? tgt - vocab_start
          : -1;

  // online max + sum pass
  float m = -3.0e38f, s = 0.f, tl = 0.f;
  for (int64_t i = tid; i < nvec; i += nth) {
    VecT v = ((const VecT*)lr)[i];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float x = E::to_f(v[j]);
      if (x > m) {
        s *= __expf(m - x);
        m = x;
      }
      s += __expf(x - m);
      if (i * V + j == local_tgt) tl = x;
    }
  }
  for (int64_t i = nvec * V + tid; i < Vl; i += nth) {
    float x = E::to_f(lr[i]);
    if (x > m) {
      s *= __expf(m - x);
      m = x;
    }
    s += __expf(x - m);
    if (i == local_tgt) tl = x;
  }
  // cross-thread: rescale partial sums to the block max
  float bm = block_reduce(m, red, MaxOp(), -3.0e38f);
  s *= __expf(m - bm);
  s = block_reduce(s, red, SumOp(), 0.f);
  tl = block_reduce(tl, red, SumOp(), 0.f);
  if (tid == 0) {
    lmax[row] = bm;
    lsumexp[row] = s;
    tlogit[row] = tl;
  }
}

// d logits[r, j] = g[r] * (exp(l - gmax) / gsum - onehot)
template <class E>
__global__ void ce_bwd_kernel(const typename E::T* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              const float* __restrict__ gmax,
                              const float* __restrict__ gsumexp,
                              const float* __restrict__ gscale,
                              typename E::T* __restrict__ dlogits, int64_t Vl,
                              int64_t vocab_start, int64_t ignore_index) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int64_t row = blockIdx.x;
  const typename E::T* lr = logits + row * Vl;
  typename E::T* dr = dlogits + row * Vl;
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  const int64_t nvec = Vl / V;
  const int64_t tgt = targets[row];
  const float g = (tgt == ignore_index) ? 0.f : gscale[row];
  const int64_t local_tgt =
      (tgt >= vocab_start && tgt < vocab_start + Vl) ? tgt - vocab_start : -1;
  const float m = gmax[row];
  const float inv = 1.0f / gsumexp[row];

  for (int64_t i = tid; i < nvec; i += nth) {
    VecT v = ((const VecT*)lr)[i];
    VecT o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float p = __expf(E::to_f(v[j]) - m) * inv;
      float oh = (i * V + j == local_tgt) ? 1.f : 0.f;
      o[j] = E::from_f(g * (p - oh));
    }
    ((VecT*)dr)[i] = o;
  }
  for (int64_t i = nvec * V + tid; i < Vl; i += nth) {
    float p = __expf(E::to_f(lr[i]) - m) * inv;
    float oh = (i == local_tgt) ? 1.f : 0.f;
    dr[i] = E::from_f(g * (p - oh));
  }
}

}  // namespace

#define CE_LAUNCHERS(SUFF, ETYPE)                                                    \
  extern "C" void ce_fwd_##SUFF(const void* logits, const int64_t* targets,          \
                                float* lmax, float* lsumexp, float* tlogit,          \
                                int64_t R, int64_t Vl, int64_t vocab_start,          \
                                int64_t ignore_index, hipStream_t stream) {          \
    ce_fwd_kernel<ETYPE><<<dim3((uint32_t)R), dim3(512), 0, stream>>>(               \
        (const ETYPE::T*)logits, targets, lmax, lsumexp, tlogit, Vl, vocab_start,    \
        ignore_index);                                                               \
  }                                                                                  \
  extern "C" void ce_bwd_##SUFF(const void* logits, const int64_t* targets,          \
                                const float* gmax, const float* gsumexp,             \
                                const float* gscale, void* dlogits, int64_t R,       \
                                int64_t Vl, int64_t vocab_start,                     \
                                int64_t ignore_index, hipStream_t stream) {          \
    ce_bwd_kernel<ETYPE><<<dim3((uint32_t)R), dim3(512), 0, stream>>>(               \
        (const ETYPE::T*)logits, targets, gmax, gsumexp, gscale, (ETYPE::T*)dlogits, \
        Vl, vocab_start, ignore_index);                                              \
  }

CE_LAUNCHERS(bf16, BF16Elem)
CE_LAUNCHERS(f32, F32Elem)
