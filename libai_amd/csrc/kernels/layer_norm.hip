// LayerNorm + RMSNorm forward/backward for gfx950.
//
// Replaces the reference's flow._C.layer_norm_affine / flow._C.rms_norm call
// sites (reference: libai/layers/layer_norm.py:78-131).  Memory-bound: one
// block per row, 16-byte vectorized bf16 loads (guide G13), fp32 accumulate,
// mean/rstd cached for backward.  Weight grads use a two-stage column
// reduction (partials over row-groups, then a fold) to avoid atomics.
#include "common.h"

namespace {

template <class E, bool RMS>
__global__ void norm_fwd_kernel(const typename E::T* __restrict__ x,
                                const typename E::T* __restrict__ gamma,
                                const typename E::T* __restrict__ beta,
                                typename E::T* __restrict__ y,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out, int H, float eps) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  __shared__ float red[16];

  const int64_t row = blockIdx.x;
  const T* xr = x + row * (int64_t)H;
  T* yr = y + row * (int64_t)H;
  const int tid = threadIdx.x;
  const int nthread = blockDim.x;
  const int nvec = H / V;

  float sum = 0.f, sumsq = 0.f;
  for (int i = tid; i < nvec; i += nthread) {
    VecT v = ((const VecT*)xr)[i];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float f = E::to_f(v[j]);
      sum += f;
      sumsq += f * f;
    }
  }
  for (int i = nvec * V + tid; i < H; i += nthread) {
    float f = E::to_f(xr[i]);
    sum += f;
    sumsq += f * f;
  }
  sum = block_reduce(sum, red, SumOp(), 0.f);
  sumsq = block_reduce(sumsq, red, SumOp(), 0.f);

  float mean = RMS ? 0.f : sum / H;
  float var = sumsq / H - mean * mean;
  float rstd = rsqrtf(var + eps);
  if (tid == 0) {
    if (!RMS && mean_out) mean_out[row] = mean;
    if (rstd_out) rstd_out[row] = rstd;
  }

  for (int i = tid; i < nvec; i += nthread) {
    VecT v = ((const VecT*)xr)[i];
    VecT g = ((const VecT*)gamma)[i];
    VecT o;
    if (!RMS && beta != nullptr) {
      VecT b = ((const VecT*)beta)[i];
#pragma unroll
      for (int j = 0; j < V; ++j)
        o[j] = E::from_f((E::to_f(v[j]) - mean) * rstd * E::to_f(g[j]) + E::to_f(b[j]));
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j)
        o[j] = E::from_f((E::to_f(v[j]) - mean) * rstd * E::to_f(g[j]));
    }
    ((VecT*)yr)[i] = o;
  }
  for (int i = nvec * V + tid; i < H; i += nthread) {
    float b = (!RMS && beta) ? E::to_f(beta[i]) : 0.f;
    yr[i] = E::from_f((E::to_f(xr[i]) - mean) * rstd * E::to_f(gamma[i]) + b);
  }
}

// dx for LayerNorm:  dx = rstd * (dyg - mean(dyg) - xhat * mean(dyg*xhat))
// dx for RMSNorm:    dx = rstd * dyg - x * rstd^3 / H * sum(dyg * x)
template <class E, bool RMS>
__global__ void norm_bwd_dx_kernel(const typename E::T* __restrict__ dy,
                                   const typename E::T* __restrict__ x,
                                   const typename E::T* __restrict__ gamma,
                                   const float* __restrict__ mean_in,
                                   const float* __restrict__ rstd_in,
                                   typename E::T* __restrict__ dx, int H) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  __shared__ float red[16];

  const int64_t row = blockIdx.x;
  const T* dyr = dy + row * (int64_t)H;
  const T* xr = x + row * (int64_t)H;
  T* dxr = dx + row * (int64_t)H;
  const int tid = threadIdx.x;
  const int nthread = blockDim.x;
  const int nvec = H / V;
  const float mean = RMS ? 0.f : mean_in[row];
  const float rstd = rstd_in[row];

  float s1 = 0.f, s2 = 0.f;  // sum(dyg), sum(dyg * xhat)
  for (int i = tid; i < nvec; i += nthread) {
    VecT vdy = ((const VecT*)dyr)[i];
    VecT vx = ((const VecT*)xr)[i];
    VecT vg = ((const VecT*)gamma)[i];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float dyg = E::to_f(vdy[j]) * E::to_f(vg[j]);
      float xhat = (E::to_f(vx[j]) - mean) * rstd;
      s1 += dyg;
      s2 += dyg * xhat;
    }
  }
  for (int i = nvec * V + tid; i < H; i += nthread) {
    float dyg = E::to_f(dyr[i]) * E::to_f(gamma[i]);
    float xhat = (E::to_f(xr[i]) - mean) * rstd;
    s1 += dyg;
    s2 += dyg * xhat;
  }
  s1 = block_reduce(s1, red, SumOp(), 0.f) / H;
  s2 = block_reduce(s2, red, SumOp(), 0.f) / H;

  for (int i = tid; i < nvec; i += nthread) {
    VecT vdy = ((const VecT*)dyr)[i];
    VecT vx = ((const VecT*)xr)[i];
    VecT vg = ((const VecT*)gamma)[i];
    VecT o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float dyg = E::to_f(vdy[j]) * E::to_f(vg[j]);
      float xhat = (E::to_f(vx[j]) - mean) * rstd;
      float v = RMS ? rstd * (dyg - xhat * s2) : rstd * (dyg - s1 - xhat * s2);
      o[j] = E::from_f(v);
    }
    ((VecT*)dxr)[i] = o;
  }
  for (int i = nvec * V + tid; i < H; i += nthread) {
    float dyg = E::to_f(dyr[i]) * E::to_f(gamma[i]);
    float xhat = (E::to_f(xr[i]) - mean) * rstd;
    float v = RMS ? rstd * (dyg - xhat * s2) : rstd * (dyg - s1 - xhat * s2);
    dxr[i] = E::from_f(v);
  }
}

// Stage 1 of dgamma/dbeta: each thread owns one column, strides row-groups.
// grid = (CDIV(H, 256), P); partials layout [P][H] fp32 (dgamma) + [P][H] (dbeta).
template <class E, bool RMS>
__global__ void norm_bwd_wgrad_partial_kernel(const typename E::T* __restrict__ dy,
                                              const typename E::T* __restrict__ x,
                                              const float* __restrict__ mean_in,
                                              const float* __restrict__ rstd_in,
                                              float* __restrict__ pgamma,
                                              float* __restrict__ pbeta, int64_t R,
                                              int H) {
  using T = typename E::T;
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= H) return;
  const int P = gridDim.y;
  float dg = 0.f, db = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) {
    float mean = RMS ? 0.f : mean_in[r];
    float rstd = rstd_in[r];
    float dyv = E::to_f(dy[r * H + j]);
    float xhat = (E::to_f(x[r * H + j]) - mean) * rstd;
    dg += dyv * xhat;
    db += dyv;
  }
  pgamma[(int64_t)blockIdx.y * H + j] = dg;
  if (pbeta) pbeta[(int64_t)blockIdx.y * H + j] = db;
}

// fold over P row-group partials: 64 cols x 8 p-lanes per block (TLP over the
// strided reads; a [H]-thread fold is latency-bound at H~1k).
template <class E>
__global__ void wgrad_fold_kernel(const float* __restrict__ pgamma,
                                  const float* __restrict__ pbeta,
                                  typename E::T* __restrict__ dgamma,
                                  typename E::T* __restrict__ dbeta, int P, int H) {
  __shared__ float lds[2][8][64];
  const int j = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  const int pl = threadIdx.x / 64;
  float dg = 0.f, db = 0.f;
  if (j < H) {
    for (int p = pl; p < P; p += 8) {
      dg += pgamma[(int64_t)p * H + j];
      if (pbeta) db += pbeta[(int64_t)p * H + j];
    }
  }
  lds[0][pl][threadIdx.x % 64] = dg;
  lds[1][pl][threadIdx.x % 64] = db;
  __syncthreads();
  if (pl == 0 && j < H) {
    float sdg = 0.f, sdb = 0.f;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      sdg += lds[0][q][threadIdx.x % 64];
      sdb += lds[1][q][threadIdx.x % 64];
    }
    dgamma[j] = E::from_f(sdg);
    if (dbeta) dbeta[j] = E::from_f(sdb);
  }
}

inline int row_block_threads(int H, int vec) {
  int nv = H / vec;
  if (nv >= 1024) return 1024;
  if (nv >= 512) return 512;
  if (nv >= 256) return 256;
  if (nv >= 128) return 128;
  return 64;
}

}  // namespace

#define NORM_LAUNCHERS(SUFF, ETYPE)                                                     \
  extern "C" void ln_fwd_##SUFF(const void* x, const void* gamma, const void* beta,     \
                                void* y, float* mean, float* rstd, int64_t R, int H,    \
                                float eps, bool rms, hipStream_t stream) {              \
    int th = row_block_threads(H, ETYPE::VEC);                                          \
    if (rms)                                                                            \
      norm_fwd_kernel<ETYPE, true><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(         \
          (const ETYPE::T*)x, (const ETYPE::T*)gamma, (const ETYPE::T*)beta,            \
          (ETYPE::T*)y, mean, rstd, H, eps);                                            \
    else                                                                                \
      norm_fwd_kernel<ETYPE, false><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(        \
          (const ETYPE::T*)x, (const ETYPE::T*)gamma, (const ETYPE::T*)beta,            \
          (ETYPE::T*)y, mean, rstd, H, eps);                                            \
  }                                                                                     \
  extern "C" void ln_bwd_dx_##SUFF(const void* dy, const void* x, const void* gamma,    \
                                   const float* mean, const float* rstd, void* dx,      \
                                   int64_t R, int H, bool rms, hipStream_t stream) {    \
    int th = row_block_threads(H, ETYPE::VEC);                                          \
    if (rms)                                                                            \
      norm_bwd_dx_kernel<ETYPE, true><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(      \
          (const ETYPE::T*)dy, (const ETYPE::T*)x, (const ETYPE::T*)gamma, mean, rstd,  \
          (ETYPE::T*)dx, H);                                                            \
    else                                                                                \
      norm_bwd_dx_kernel<ETYPE, false><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(     \
          (const ETYPE::T*)dy, (const ETYPE::T*)x, (const ETYPE::T*)gamma, mean, rstd,  \
          (ETYPE::T*)dx, H);                                                            \
  }                                                                                     \
  extern "C" void ln_bwd_wgrad_##SUFF(const void* dy, const void* x, const float* mean, \
                                      const float* rstd, float* pgamma, float* pbeta,   \
                                      void* dgamma, void* dbeta, int64_t R, int H,      \
                                      int P, bool rms, hipStream_t stream) {            \
    dim3 grid1(CDIV(H, 256), P);                                                        \
    if (rms)                                                                            \
      norm_bwd_wgrad_partial_kernel<ETYPE, true><<<grid1, dim3(256), 0, stream>>>(      \
          (const ETYPE::T*)dy, (const ETYPE::T*)x, mean, rstd, pgamma, pbeta, R, H);    \
    else                                                                                \
      norm_bwd_wgrad_partial_kernel<ETYPE, false><<<grid1, dim3(256), 0, stream>>>(     \
          (const ETYPE::T*)dy, (const ETYPE::T*)x, mean, rstd, pgamma, pbeta, R, H);    \
    wgrad_fold_kernel<ETYPE><<<dim3(CDIV(H, 64)), dim3(512), 0, stream>>>(              \
        pgamma, pbeta, (ETYPE::T*)dgamma, (ETYPE::T*)dbeta, P, H);                      \
  }

NORM_LAUNCHERS(bf16, BF16Elem)
NORM_LAUNCHERS(f32, F32Elem)
