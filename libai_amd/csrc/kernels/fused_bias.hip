// Fused bias+gelu and bias+dropout+residual elementwise kernels for gfx950.
//
// Replaces the reference's flow._C.fused_bias_add_gelu (libai/layers/mlp.py:95-97)
// and flow._C.fused_bias_add_dropout (mlp.py:104-106, attention.py:265-267).
// Memory-bound: 16-byte vectorized loads, fp32 math, philox-recomputed dropout
// masks (no mask tensor traffic; backward regenerates the identical mask).
#include "common.h"

namespace {

// ---------------------------------------------------------------------------
// y = gelu(x + b)   (erf gelu, matches torch.nn.functional.gelu default)
// ---------------------------------------------------------------------------
template <class E, bool GRAD>
__global__ void bias_gelu_kernel(const typename E::T* __restrict__ x,
                                 const typename E::T* __restrict__ b,
                                 const typename E::T* __restrict__ dy,
                                 typename E::T* __restrict__ out, int64_t n, int W) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int64_t nvec = n / V;
  const int wvec = W / V;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    VecT vx = ((const VecT*)x)[i];
    VecT vb;
    if (b) vb = ((const VecT*)b)[i % wvec];
    VecT o;
    if (GRAD) {
      VecT vdy = ((const VecT*)dy)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xv = E::to_f(vx[j]) + (b ? E::to_f(vb[j]) : 0.f);
        o[j] = E::from_f(E::to_f(vdy[j]) * gelu_erf_grad(xv));
      }
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xv = E::to_f(vx[j]) + (b ? E::to_f(vb[j]) : 0.f);
        o[j] = E::from_f(gelu_erf(xv));
      }
    }
    ((VecT*)out)[i] = o;
  }
}

// ---------------------------------------------------------------------------
// y = residual + dropout(x + b);  backward dx = dy * mask / (1-p)
// mask is philox(seed, elem_idx) — identical in fwd and bwd.
// ---------------------------------------------------------------------------
template <class E, bool GRAD>
__global__ void bias_dropout_res_kernel(const typename E::T* __restrict__ x,
                                        const typename E::T* __restrict__ b,
                                        const typename E::T* __restrict__ res,
                                        typename E::T* __restrict__ out, int64_t n,
                                        int W, float p, uint64_t seed) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int64_t nvec = n / V;
  const int wvec = W / V;
  const float scale = 1.0f / (1.0f - p);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    VecT vx = ((const VecT*)x)[i];
    VecT vb;
    if (b) vb = ((const VecT*)b)[i % wvec];
    float keep[V];
    if (p > 0.f) {
      uint32_t r[4];
#pragma unroll
      for (int q = 0; q < V / 4; ++q) {
        philox4(seed, (uint64_t)i * (V / 4) + q, r);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          keep[q * 4 + j] = (u32_to_uniform(r[j]) > p) ? scale : 0.f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) keep[j] = 1.f;
    }
    VecT o;
    if (GRAD) {
#pragma unroll
      for (int j = 0; j < V; ++j) o[j] = E::from_f(E::to_f(vx[j]) * keep[j]);
    } else {
      VecT vr;
      if (res) vr = ((const VecT*)res)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float v = (E::to_f(vx[j]) + (b ? E::to_f(vb[j]) : 0.f)) * keep[j];
        o[j] = E::from_f(v + (res ? E::to_f(vr[j]) : 0.f));
      }
    }
    ((VecT*)out)[i] = o;
  }
}

// ---------------------------------------------------------------------------
// generic column reduction: out[j] = sum_r in[r, j]  (for dbias)
// stage 1: thread owns one column, strides row-groups -> partials [P][W] f32
// ---------------------------------------------------------------------------
template <class E>
__global__ void colsum_partial_kernel(const typename E::T* __restrict__ in,
                                      float* __restrict__ partial, int64_t R, int W) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= W) return;
  const int P = gridDim.y;
  float s = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) s += E::to_f(in[r * W + j]);
  partial[(int64_t)blockIdx.y * W + j] = s;
}

// fold: block = 64 cols x 8 p-lanes so the strided partial reads have TLP
// (a [W]-thread fold is latency-bound: 4 blocks cannot hide 256 dependent
// HBM/L2 round trips).
template <class E>
__global__ void colsum_fold_kernel(const float* __restrict__ partial,
                                   typename E::T* __restrict__ out, int P, int W) {
  __shared__ float lds[8][64];
  const int j = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  const int pl = threadIdx.x / 64;  // 8 p-lanes
  float s = 0.f;
  if (j < W) {
    for (int p = pl; p < P; p += 8) s += partial[(int64_t)p * W + j];
  }
  lds[pl][threadIdx.x % 64] = s;
  __syncthreads();
  if (pl == 0 && j < W) {
    float acc = 0.f;
#pragma unroll
    for (int q = 0; q < 8; ++q) acc += lds[q][threadIdx.x % 64];
    out[j] = E::from_f(acc);
  }
}

inline int64_t ew_grid(int64_t nvec) {
  int64_t g = CDIV(nvec, 256);
  return g < 2048 ? g : 2048;  // grid-stride past 2048 blocks (guide G11)
}

}  // namespace

#define BIAS_LAUNCHERS(SUFF, ETYPE)                                                      \
  extern "C" void bias_gelu_fwd_##SUFF(const void* x, const void* b, void* y, int64_t n, \
                                       int W, hipStream_t stream) {                      \
    bias_gelu_kernel<ETYPE, false>                                                       \
        <<<dim3(ew_grid(n / ETYPE::VEC)), dim3(256), 0, stream>>>(                       \
            (const ETYPE::T*)x, (const ETYPE::T*)b, nullptr, (ETYPE::T*)y, n, W);        \
  }                                                                                      \
  extern "C" void bias_gelu_bwd_##SUFF(const void* x, const void* b, const void* dy,     \
                                       void* dx, int64_t n, int W, hipStream_t stream) { \
    bias_gelu_kernel<ETYPE, true>                                                        \
        <<<dim3(ew_grid(n / ETYPE::VEC)), dim3(256), 0, stream>>>(                       \
            (const ETYPE::T*)x, (const ETYPE::T*)b, (const ETYPE::T*)dy, (ETYPE::T*)dx,  \
            n, W);                                                                       \
  }                                                                                      \
  extern "C" void bias_dropout_res_fwd_##SUFF(const void* x, const void* b,              \
                                              const void* res, void* y, int64_t n,       \
                                              int W, float p, uint64_t seed,             \
                                              hipStream_t stream) {                      \
    bias_dropout_res_kernel<ETYPE, false>                                                \
        <<<dim3(ew_grid(n / ETYPE::VEC)), dim3(256), 0, stream>>>(                       \
            (const ETYPE::T*)x, (const ETYPE::T*)b, (const ETYPE::T*)res, (ETYPE::T*)y,  \
            n, W, p, seed);                                                              \
  }                                                                                      \
  extern "C" void bias_dropout_res_bwd_##SUFF(const void* dy, void* dx, int64_t n,       \
                                              int W, float p, uint64_t seed,             \
                                              hipStream_t stream) {                      \
    bias_dropout_res_kernel<ETYPE, true>                                                 \
        <<<dim3(ew_grid(n / ETYPE::VEC)), dim3(256), 0, stream>>>(                       \
            (const ETYPE::T*)dy, nullptr, nullptr, (ETYPE::T*)dx, n, W, p, seed);        \
  }                                                                                      \
  extern "C" void colsum_##SUFF(const void* in, float* partial, void* out, int64_t R,    \
                                int W, int P, hipStream_t stream) {                      \
    colsum_partial_kernel<ETYPE><<<dim3(CDIV(W, 256), P), dim3(256), 0, stream>>>(       \
        (const ETYPE::T*)in, partial, R, W);                                             \
    colsum_fold_kernel<ETYPE><<<dim3(CDIV(W, 64)), dim3(512), 0, stream>>>(              \
        partial, (ETYPE::T*)out, P, W);                                                  \
  }

BIAS_LAUNCHERS(bf16, BF16Elem)
BIAS_LAUNCHERS(f32, F32Elem)
