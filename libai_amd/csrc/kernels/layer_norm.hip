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

// Wave-per-row variants (H <= 2048, H % VEC == 0): one 64-lane wave owns a
// row, reduces with shfl_xor (no LDS, no barriers), and keeps the row's x
// vectors in registers between the stats pass and the normalize pass.  The
// block-per-row kernels above remain the fallback for wide/ragged rows.
template <class E, bool RMS, int NV>
__global__ void norm_fwd_wave_kernel(const typename E::T* __restrict__ x,
                                     const typename E::T* __restrict__ gamma,
                                     const typename E::T* __restrict__ beta,
                                     typename E::T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out, int H, float eps,
                                     int64_t R) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * (blockDim.x >> 6) + wave;
  if (row >= R) return;
  const T* xr = x + row * (int64_t)H;
  T* yr = y + row * (int64_t)H;
  const int nvec = H / V;

  VecT vx[NV];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      vx[n] = ((const VecT*)xr)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float f = E::to_f(vx[n][j]);
        sum += f;
        sumsq += f * f;
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_xor(sum, off, 64);
    sumsq += __shfl_xor(sumsq, off, 64);
  }
  const float mean = RMS ? 0.f : sum / H;
  const float rstd = rsqrtf(sumsq / H - mean * mean + eps);
  if (lane == 0) {
    if (!RMS && mean_out) mean_out[row] = mean;
    if (rstd_out) rstd_out[row] = rstd;
  }
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      VecT g = ((const VecT*)gamma)[i];
      VecT o;
      if (!RMS && beta != nullptr) {
        VecT b = ((const VecT*)beta)[i];
#pragma unroll
        for (int j = 0; j < V; ++j)
          o[j] = E::from_f((E::to_f(vx[n][j]) - mean) * rstd * E::to_f(g[j]) +
                           E::to_f(b[j]));
      } else {
#pragma unroll
        for (int j = 0; j < V; ++j)
          o[j] = E::from_f((E::to_f(vx[n][j]) - mean) * rstd * E::to_f(g[j]));
      }
      ((VecT*)yr)[i] = o;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused residual-add + norm (decode/eval path): h = x + bias + res is
// written out (the next residual) AND normalized in the same registers —
// replaces the bias_dropout_res + norm_fwd kernel pair per layer in the
// hipGraph-captured serving step (both latency-bound at decode batch
// sizes).  Forward-only: the captured decode never needs its backward.
// ---------------------------------------------------------------------------
template <class E, bool RMS, int NV>
__global__ void res_norm_fwd_wave_kernel(
    const typename E::T* __restrict__ x, const typename E::T* __restrict__ bias,
    const typename E::T* __restrict__ res, const typename E::T* __restrict__ gamma,
    const typename E::T* __restrict__ beta, typename E::T* __restrict__ h,
    typename E::T* __restrict__ y, int H, float eps, int64_t R) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * (blockDim.x >> 6) + wave;
  if (row >= R) return;
  const T* xr = x + row * (int64_t)H;
  const T* rr = res + row * (int64_t)H;
  T* hr = h + row * (int64_t)H;
  T* yr = y + row * (int64_t)H;
  const int nvec = H / V;

  VecT vh[NV];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      VecT vx = ((const VecT*)xr)[i];
      VecT vr = ((const VecT*)rr)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        // association matches the bias_dropout_res kernel exactly
        // ((x + bias) + res) so the eager and captured decode paths
        // produce bitwise-identical residual streams
        float f = E::to_f(vx[j]);
        if (bias != nullptr) f += E::to_f(((const VecT*)bias)[i][j]);
        f += E::to_f(vr[j]);
        vh[n][j] = E::from_f(f);
        const float g = E::to_f(vh[n][j]);  // stats on the STORED value
        sum += g;
        sumsq += g * g;
      }
      ((VecT*)hr)[i] = vh[n];
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_xor(sum, off, 64);
    sumsq += __shfl_xor(sumsq, off, 64);
  }
  const float mean = RMS ? 0.f : sum / H;
  const float rstd = rsqrtf(sumsq / H - mean * mean + eps);
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      VecT g = ((const VecT*)gamma)[i];
      VecT o;
      if (!RMS && beta != nullptr) {
        VecT b = ((const VecT*)beta)[i];
#pragma unroll
        for (int j = 0; j < V; ++j)
          o[j] = E::from_f((E::to_f(vh[n][j]) - mean) * rstd * E::to_f(g[j]) +
                           E::to_f(b[j]));
      } else {
#pragma unroll
        for (int j = 0; j < V; ++j)
          o[j] = E::from_f((E::to_f(vh[n][j]) - mean) * rstd * E::to_f(g[j]));
      }
      ((VecT*)yr)[i] = o;
    }
  }
}

template <class E, bool RMS, int NV>
__global__ void norm_bwd_dx_wave_kernel(const typename E::T* __restrict__ dy,
                                        const typename E::T* __restrict__ x,
                                        const typename E::T* __restrict__ gamma,
                                        const float* __restrict__ mean_in,
                                        const float* __restrict__ rstd_in,
                                        typename E::T* __restrict__ dx, int H,
                                        int64_t R) {
  using T = typename E::T;
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * (blockDim.x >> 6) + wave;
  if (row >= R) return;
  const T* dyr = dy + row * (int64_t)H;
  const T* xr = x + row * (int64_t)H;
  T* dxr = dx + row * (int64_t)H;
  const int nvec = H / V;
  const float mean = RMS ? 0.f : mean_in[row];
  const float rstd = rstd_in[row];

  VecT vdy[NV], vx[NV], vg[NV];
  float s1 = 0.f, s2 = 0.f;
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      vdy[n] = ((const VecT*)dyr)[i];
      vx[n] = ((const VecT*)xr)[i];
      vg[n] = ((const VecT*)gamma)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float dyg = E::to_f(vdy[n][j]) * E::to_f(vg[n][j]);
        const float xhat = (E::to_f(vx[n][j]) - mean) * rstd;
        s1 += dyg;
        s2 += dyg * xhat;
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s1 += __shfl_xor(s1, off, 64);
    s2 += __shfl_xor(s2, off, 64);
  }
  s1 /= H;
  s2 /= H;
#pragma unroll
  for (int n = 0; n < NV; ++n) {
    const int i = lane + n * 64;
    if (i < nvec) {
      VecT o;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float dyg = E::to_f(vdy[n][j]) * E::to_f(vg[n][j]);
        const float xhat = (E::to_f(vx[n][j]) - mean) * rstd;
        o[j] = E::from_f(RMS ? rstd * (dyg - xhat * s2)
                             : rstd * (dyg - s1 - xhat * s2));
      }
      ((VecT*)dxr)[i] = o;
    }
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
// scalar fallback for H not divisible by the vector width
template <class E, bool RMS>
__global__ void norm_bwd_wgrad_partial_scalar_kernel(
    const typename E::T* __restrict__ dy, const typename E::T* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    float* __restrict__ pgamma, float* __restrict__ pbeta, int64_t R, int H) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= H) return;
  const int P = gridDim.y;
  float dg = 0.f, db = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) {
    const float mean = RMS ? 0.f : mean_in[r];
    const float rstd = rstd_in[r];
    const float dyv = E::to_f(dy[r * H + j]);
    dg += dyv * (E::to_f(x[r * H + j]) - mean) * rstd;
    db += dyv;
  }
  pgamma[(int64_t)blockIdx.y * H + j] = dg;
  if (pbeta) pbeta[(int64_t)blockIdx.y * H + j] = db;
}

// vectorized: thread owns E::VEC columns (16B loads of dy and x per row --
// the scalar 2B-per-thread variant ran at ~2 TB/s)
template <class E, bool RMS>
__global__ void norm_bwd_wgrad_partial_kernel(const typename E::T* __restrict__ dy,
                                              const typename E::T* __restrict__ x,
                                              const float* __restrict__ mean_in,
                                              const float* __restrict__ rstd_in,
                                              float* __restrict__ pgamma,
                                              float* __restrict__ pbeta, int64_t R,
                                              int H) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wv = H / V;
  const int jv = blockIdx.x * blockDim.x + threadIdx.x;
  if (jv >= wv) return;
  const int P = gridDim.y;
  float dg[V], db[V];
#pragma unroll
  for (int j = 0; j < V; ++j) dg[j] = db[j] = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) {
    const float mean = RMS ? 0.f : mean_in[r];
    const float rstd = rstd_in[r];
    VecT vd = ((const VecT*)(dy + r * H))[jv];
    VecT vx = ((const VecT*)(x + r * H))[jv];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float dyv = E::to_f(vd[j]);
      dg[j] += dyv * (E::to_f(vx[j]) - mean) * rstd;
      db[j] += dyv;
    }
  }
#pragma unroll
  for (int j = 0; j < V; ++j) {
    pgamma[(int64_t)blockIdx.y * H + jv * V + j] = dg[j];
    if (pbeta) pbeta[(int64_t)blockIdx.y * H + jv * V + j] = db[j];
  }
}

template <class E, bool RMS>
void launch_norm_fwd(const void* x, const void* gamma, const void* beta, void* y,
                     float* mean, float* rstd, int64_t R, int H, float eps,
                     hipStream_t stream, int th) {
  const int nvl = CDIV(H / E::VEC, 64);
  if (H % E::VEC == 0 && nvl <= 4) {
    dim3 g((uint32_t)CDIV(R, 4));
#define FWD_WAVE(NV)                                                          \
  norm_fwd_wave_kernel<E, RMS, NV><<<g, dim3(256), 0, stream>>>(              \
      (const typename E::T*)x, (const typename E::T*)gamma,                   \
      (const typename E::T*)beta, (typename E::T*)y, mean, rstd, H, eps, R)
    if (nvl <= 1)
      FWD_WAVE(1);
    else if (nvl == 2)
      FWD_WAVE(2);
    else
      FWD_WAVE(4);
#undef FWD_WAVE
  } else {
    norm_fwd_kernel<E, RMS><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(
        (const typename E::T*)x, (const typename E::T*)gamma,
        (const typename E::T*)beta, (typename E::T*)y, mean, rstd, H, eps);
  }
}

template <class E, bool RMS>
void launch_res_norm_fwd(const void* x, const void* bias, const void* res,
                         const void* gamma, const void* beta, void* h, void* y,
                         int64_t R, int H, float eps, hipStream_t stream) {
  const int nvl = CDIV(H / E::VEC, 64);
  dim3 g((uint32_t)CDIV(R, 4));
#define RES_FWD_WAVE(NV)                                                      \
  res_norm_fwd_wave_kernel<E, RMS, NV><<<g, dim3(256), 0, stream>>>(          \
      (const typename E::T*)x, (const typename E::T*)bias,                    \
      (const typename E::T*)res, (const typename E::T*)gamma,                 \
      (const typename E::T*)beta, (typename E::T*)h, (typename E::T*)y, H,    \
      eps, R)
  if (nvl <= 1)
    RES_FWD_WAVE(1);
  else if (nvl == 2)
    RES_FWD_WAVE(2);
  else
    RES_FWD_WAVE(4);
#undef RES_FWD_WAVE
}

template <class E, bool RMS>
void launch_norm_bwd_dx(const void* dy, const void* x, const void* gamma,
                        const float* mean, const float* rstd, void* dx, int64_t R,
                        int H, hipStream_t stream, int th) {
  const int nvl = CDIV(H / E::VEC, 64);
  if (H % E::VEC == 0 && nvl <= 4) {
    dim3 g((uint32_t)CDIV(R, 4));
#define DX_WAVE(NV)                                                           \
  norm_bwd_dx_wave_kernel<E, RMS, NV><<<g, dim3(256), 0, stream>>>(           \
      (const typename E::T*)dy, (const typename E::T*)x,                      \
      (const typename E::T*)gamma, mean, rstd, (typename E::T*)dx, H, R)
    if (nvl <= 1)
      DX_WAVE(1);
    else if (nvl == 2)
      DX_WAVE(2);
    else
      DX_WAVE(4);
#undef DX_WAVE
  } else {
    norm_bwd_dx_kernel<E, RMS><<<dim3((uint32_t)R), dim3(th), 0, stream>>>(
        (const typename E::T*)dy, (const typename E::T*)x,
        (const typename E::T*)gamma, mean, rstd, (typename E::T*)dx, H);
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
      launch_norm_fwd<ETYPE, true>(x, gamma, beta, y, mean, rstd, R, H, eps, stream,    \
                                   th);                                                 \
    else                                                                                \
      launch_norm_fwd<ETYPE, false>(x, gamma, beta, y, mean, rstd, R, H, eps, stream,   \
                                    th);                                                \
  }                                                                                     \
  extern "C" void ln_bwd_dx_##SUFF(const void* dy, const void* x, const void* gamma,    \
                                   const float* mean, const float* rstd, void* dx,      \
                                   int64_t R, int H, bool rms, hipStream_t stream) {    \
    int th = row_block_threads(H, ETYPE::VEC);                                          \
    if (rms)                                                                            \
      launch_norm_bwd_dx<ETYPE, true>(dy, x, gamma, mean, rstd, dx, R, H, stream, th);  \
    else                                                                                \
      launch_norm_bwd_dx<ETYPE, false>(dy, x, gamma, mean, rstd, dx, R, H, stream,      \
                                       th);                                             \
  }                                                                                     \
  extern "C" void ln_bwd_wgrad_##SUFF(const void* dy, const void* x, const float* mean, \
                                      const float* rstd, float* pgamma, float* pbeta,   \
                                      void* dgamma, void* dbeta, int64_t R, int H,      \
                                      int P, bool rms, hipStream_t stream) {            \
    if (H % ETYPE::VEC == 0) {                                                          \
      dim3 grid1(CDIV(H / ETYPE::VEC, 256), P);                                         \
      if (rms)                                                                          \
        norm_bwd_wgrad_partial_kernel<ETYPE, true><<<grid1, dim3(256), 0, stream>>>(    \
            (const ETYPE::T*)dy, (const ETYPE::T*)x, mean, rstd, pgamma, pbeta, R, H);  \
      else                                                                              \
        norm_bwd_wgrad_partial_kernel<ETYPE, false><<<grid1, dim3(256), 0, stream>>>(   \
            (const ETYPE::T*)dy, (const ETYPE::T*)x, mean, rstd, pgamma, pbeta, R, H);  \
    } else {                                                                            \
      dim3 grid1(CDIV(H, 256), P);                                                      \
      if (rms)                                                                          \
        norm_bwd_wgrad_partial_scalar_kernel<ETYPE, true>                               \
            <<<grid1, dim3(256), 0, stream>>>((const ETYPE::T*)dy,                      \
                                              (const ETYPE::T*)x, mean, rstd, pgamma,   \
                                              pbeta, R, H);                             \
      else                                                                              \
        norm_bwd_wgrad_partial_scalar_kernel<ETYPE, false>                              \
            <<<grid1, dim3(256), 0, stream>>>((const ETYPE::T*)dy,                      \
                                              (const ETYPE::T*)x, mean, rstd, pgamma,   \
                                              pbeta, R, H);                             \
    }                                                                                   \
    (void)dgamma;                                                                       \
    (void)dbeta;                                                                        \
  }

NORM_LAUNCHERS(bf16, BF16Elem)
NORM_LAUNCHERS(f32, F32Elem)

extern "C" void res_ln_fwd_bf16(const void* x, const void* bias,
                                const void* res, const void* gamma,
                                const void* beta, void* h, void* y, int64_t R,
                                int H, float eps, bool rms,
                                hipStream_t stream) {
  if (rms)
    launch_res_norm_fwd<BF16Elem, true>(x, bias, res, gamma, beta, h, y, R, H,
                                        eps, stream);
  else
    launch_res_norm_fwd<BF16Elem, false>(x, bias, res, gamma, beta, h, y, R, H,
                                         eps, stream);
}
