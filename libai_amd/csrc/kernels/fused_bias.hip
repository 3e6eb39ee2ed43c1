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
// 2D launch: thread owns ONE vector-column (bias loaded once, no per-element
// modulo -- a flat grid-stride variant spent ~30 of its 406 loop instructions
// on the emulated 64-bit `i % wvec`), rows strided by gridDim.y.
template <class E, bool GRAD>
__global__ void bias_gelu_kernel(const typename E::T* __restrict__ x,
                                 const typename E::T* __restrict__ b,
                                 const typename E::T* __restrict__ dy,
                                 typename E::T* __restrict__ out,
                                 float* __restrict__ db_partial, int64_t n,
                                 int W) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wvec = W / V;
  const int64_t R = n / W;
  const int cv = blockIdx.x * blockDim.x + threadIdx.x;
  if (cv >= wvec) return;
  VecT vb;
  if (b) vb = ((const VecT*)b)[cv];
  float db[GRAD ? V : 1] = {0.f};
  for (int64_t r = blockIdx.y; r < R; r += gridDim.y) {
    const int64_t i = r * wvec + cv;
    VecT vx = ((const VecT*)x)[i];
    VecT o;
    if (GRAD) {
      VecT vdy = ((const VecT*)dy)[i];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xv = E::to_f(vx[j]) + (b ? E::to_f(vb[j]) : 0.f);
        float g = E::to_f(vdy[j]) * gelu_erf_grad(xv);
        o[j] = E::from_f(g);
        db[j] += g;  // bias grad rides along: no second pass over dx
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
  if (GRAD && db_partial != nullptr) {
#pragma unroll
    for (int j = 0; j < V; ++j)
      db_partial[(int64_t)blockIdx.y * W + cv * V + j] = db[j];
  }
}

// ---------------------------------------------------------------------------
// y = residual + dropout(x + b);  backward dx = dy * mask / (1-p)
// mask is philox(seed, elem_idx) — identical in fwd and bwd.
// ---------------------------------------------------------------------------
// Dropout mask: one rnd_hash draw yields 4 byte-granular keep decisions
// (p quantized to 1/256 like the attention dropout; a philox4 here cost ~60
// VALU per 4 elements vs ~10 for the mixer).  fwd and bwd regenerate the
// identical mask from (seed, element index).
template <class E, bool GRAD>
__global__ void bias_dropout_res_kernel(const typename E::T* __restrict__ x,
                                        const typename E::T* __restrict__ b,
                                        const typename E::T* __restrict__ res,
                                        typename E::T* __restrict__ out,
                                        float* __restrict__ db_partial, int64_t n,
                                        int W, float p, uint64_t seed) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wvec = W / V;
  const int64_t R = n / W;
  const float scale = 1.0f / (1.0f - p);
  const uint32_t thr8 = drop_threshold_u8(p);
  const int cv = blockIdx.x * blockDim.x + threadIdx.x;
  if (cv >= wvec) return;
  VecT vb;
  if (b) vb = ((const VecT*)b)[cv];
  float db[GRAD ? V : 1] = {0.f};
  for (int64_t r = blockIdx.y; r < R; r += gridDim.y) {
    const int64_t i = r * wvec + cv;
    VecT vx = ((const VecT*)x)[i];
    float keep[V];
    if (p > 0.f) {
#pragma unroll
      for (int q = 0; q < V / 4; ++q) {
        const uint64_t idx = (uint64_t)i * (V / 4) + q;
        const uint32_t h = rnd_hash(seed, idx);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          keep[q * 4 + j] = (((h >> (j * 8)) & 0xFFu) >= thr8) ? scale : 0.f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) keep[j] = 1.f;
    }
    VecT o;
    if (GRAD) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float g = E::to_f(vx[j]) * keep[j];
        o[j] = E::from_f(g);
        db[j] += g;
      }
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
  if (GRAD && db_partial != nullptr) {
#pragma unroll
    for (int j = 0; j < V; ++j)
      db_partial[(int64_t)blockIdx.y * W + cv * V + j] = db[j];
  }
}

// ---------------------------------------------------------------------------
// generic column reduction: out[j] = sum_r in[r, j]  (for dbias)
// stage 1: thread owns one column, strides row-groups -> partials [P][W] f32
// ---------------------------------------------------------------------------
// vectorized: thread owns E::VEC columns (16B row loads; the scalar variant
// read 2B per thread per row and ran at ~2 TB/s)
template <class E>
__global__ void colsum_partial_kernel(const typename E::T* __restrict__ in,
                                      float* __restrict__ partial, int64_t R, int W) {
  using VecT = typename E::VecT;
  constexpr int V = E::VEC;
  const int wv = W / V;
  const int jv = blockIdx.x * blockDim.x + threadIdx.x;
  if (jv >= wv) return;
  const int P = gridDim.y;
  float s[V];
#pragma unroll
  for (int j = 0; j < V; ++j) s[j] = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) {
    VecT v = ((const VecT*)(in + r * W))[jv];
#pragma unroll
    for (int j = 0; j < V; ++j) s[j] += E::to_f(v[j]);
  }
#pragma unroll
  for (int j = 0; j < V; ++j) partial[(int64_t)blockIdx.y * W + jv * V + j] = s[j];
}

// scalar fallback for W not divisible by the vector width
template <class E>
__global__ void colsum_partial_scalar_kernel(const typename E::T* __restrict__ in,
                                             float* __restrict__ partial, int64_t R,
                                             int W) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= W) return;
  const int P = gridDim.y;
  float s = 0.f;
  for (int64_t r = blockIdx.y; r < R; r += P) s += E::to_f(in[r * W + j]);
  partial[(int64_t)blockIdx.y * W + j] = s;
}

inline int64_t ew_grid(int64_t nvec) {
  int64_t g = CDIV(nvec, 256);
  return g < 2048 ? g : 2048;  // grid-stride past 2048 blocks (guide G11)
}

// 2D launch shape for the fixed-column kernels: x covers the vector-columns,
// y strides rows, total blocks ~4096 (>> 256 CUs, bounded queue cost)
inline dim3 col_grid(int64_t n, int W, int V, int* block_out) {
  const int wvec = W / V;
  const int block = wvec < 256 ? wvec : 256;
  const int gx = (int)CDIV(wvec, block);
  int64_t rows = n / W;
  int64_t gy = 4096 / gx;
  if (gy > rows) gy = rows;
  if (gy < 1) gy = 1;
  *block_out = block;
  return dim3(gx, (uint32_t)gy);
}

}  // namespace

extern "C" int bias_col_grid_rows(int64_t n, int W, int V) {
  int blk;
  dim3 g = col_grid(n, W, V, &blk);
  return (int)g.y;
}

#define BIAS_LAUNCHERS(SUFF, ETYPE)                                                      \
  extern "C" void bias_gelu_fwd_##SUFF(const void* x, const void* b, void* y, int64_t n, \
                                       int W, hipStream_t stream) {                      \
    int blk;                                                                             \
    dim3 g = col_grid(n, W, ETYPE::VEC, &blk);                                           \
    bias_gelu_kernel<ETYPE, false><<<g, dim3(blk), 0, stream>>>(                         \
        (const ETYPE::T*)x, (const ETYPE::T*)b, nullptr, (ETYPE::T*)y, nullptr, n,       \
        W);                                                                              \
  }                                                                                      \
  extern "C" void bias_gelu_bwd_##SUFF(const void* x, const void* b, const void* dy,     \
                                       void* dx, float* db_partial, int64_t n, int W,    \
                                       hipStream_t stream) {                             \
    int blk;                                                                             \
    dim3 g = col_grid(n, W, ETYPE::VEC, &blk);                                           \
    bias_gelu_kernel<ETYPE, true><<<g, dim3(blk), 0, stream>>>(                          \
        (const ETYPE::T*)x, (const ETYPE::T*)b, (const ETYPE::T*)dy, (ETYPE::T*)dx,      \
        db_partial, n, W);                                                               \
  }                                                                                      \
  extern "C" void bias_dropout_res_fwd_##SUFF(const void* x, const void* b,              \
                                              const void* res, void* y, int64_t n,       \
                                              int W, float p, uint64_t seed,             \
                                              hipStream_t stream) {                      \
    int blk;                                                                             \
    dim3 g = col_grid(n, W, ETYPE::VEC, &blk);                                           \
    bias_dropout_res_kernel<ETYPE, false><<<g, dim3(blk), 0, stream>>>(                  \
        (const ETYPE::T*)x, (const ETYPE::T*)b, (const ETYPE::T*)res, (ETYPE::T*)y,      \
        nullptr, n, W, p, seed);                                                         \
  }                                                                                      \
  extern "C" void bias_dropout_res_bwd_##SUFF(const void* dy, void* dx,                  \
                                              float* db_partial, int64_t n, int W,       \
                                              float p, uint64_t seed,                    \
                                              hipStream_t stream) {                      \
    int blk;                                                                             \
    dim3 g = col_grid(n, W, ETYPE::VEC, &blk);                                           \
    bias_dropout_res_kernel<ETYPE, true><<<g, dim3(blk), 0, stream>>>(                   \
        (const ETYPE::T*)dy, nullptr, nullptr, (ETYPE::T*)dx, db_partial, n, W, p,       \
        seed);                                                                           \
  }                                                                                      \
  extern "C" void colsum_##SUFF(const void* in, float* partial, void* out, int64_t R,    \
                                int W, int P, hipStream_t stream) {                      \
    if (W % ETYPE::VEC == 0)                                                             \
      colsum_partial_kernel<ETYPE>                                                       \
          <<<dim3(CDIV(W / ETYPE::VEC, 256), P), dim3(256), 0, stream>>>(                \
              (const ETYPE::T*)in, partial, R, W);                                       \
    else                                                                                 \
      colsum_partial_scalar_kernel<ETYPE>                                                \
          <<<dim3(CDIV(W, 256), P), dim3(256), 0, stream>>>(                             \
              (const ETYPE::T*)in, partial, R, W);                                       \
    (void)out;                                                                           \
  }

BIAS_LAUNCHERS(bf16, BF16Elem)
BIAS_LAUNCHERS(f32, F32Elem)
