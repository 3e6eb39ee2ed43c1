// Multi-tensor fused AdamW + grad-norm kernels for gfx950.
//
// Replaces the reference's fused model-update path (OneFlow
// allow_fuse_model_update_ops + param-group clip_grad, reference:
// libai/models/utils/graph_base.py:74-76, libai/optim/build.py:86-88).
//
// The host packs per-chunk descriptors (pointers as int64, chunk length) into
// one device buffer; a single launch updates every parameter chunk.  bf16
// training keeps fp32 master weights: the kernel updates the master and
// rewrites the bf16 model copy in the same pass.  Gradient clipping is a
// multi-tensor squared-L2 kernel + a host-side scalar, applied in the update
// via grad_scale (one pass over grads total).
#include "common.h"

namespace {

constexpr int CHUNK = 1 << 16;  // elems per chunk (bf16: 128 KiB)

struct AdamChunkDesc {
  // int64 fields so Python can build the buffer as a [N, 6] int64 tensor
  int64_t param;   // bf16 model weights (0 if fp32-native)
  int64_t master;  // fp32 master weights (= param storage when fp32-native)
  int64_t grad;    // gradient (same dtype as param flag of the launcher)
  int64_t m;       // fp32 exp_avg
  int64_t v;       // fp32 exp_avg_sq
  int64_t n;       // number of elements in this chunk
};

template <class E>
__global__ void adamw_kernel(const AdamChunkDesc* __restrict__ chunks, float lr,
                             float beta1, float beta2, float eps, float wd,
                             float bc1, float bc2,  // 1-b1^t, 1-b2^t
                             float grad_scale) {
  const AdamChunkDesc c = chunks[blockIdx.x];
  const int64_t n = c.n;
  typename E::T* p16 = (typename E::T*)c.param;
  float* master = (float*)c.master;
  const typename E::T* g = (const typename E::T*)c.grad;
  float* m = (float*)c.m;
  float* v = (float*)c.v;
  const float inv_bc1 = 1.0f / bc1, inv_bc2 = 1.0f / bc2;

  for (int64_t i = threadIdx.x; i < n; i += blockDim.x) {
    float gv = E::to_f(g[i]) * grad_scale;
    float pv = master[i];
    float mv = m[i] = beta1 * m[i] + (1.0f - beta1) * gv;
    float vv = v[i] = beta2 * v[i] + (1.0f - beta2) * gv * gv;
    float mhat = mv * inv_bc1;
    float vhat = vv * inv_bc2;
    pv -= lr * (mhat / (sqrtf(vhat) + eps) + wd * pv);
    master[i] = pv;
    if (p16) p16[i] = E::from_f(pv);
  }
}

struct NormChunkDesc {
  int64_t ptr;
  int64_t n;
};

template <class E>
__global__ void l2norm_sq_kernel(const NormChunkDesc* __restrict__ chunks,
                                 float* __restrict__ out) {
  __shared__ float red[16];
  const NormChunkDesc c = chunks[blockIdx.x];
  const typename E::T* x = (const typename E::T*)c.ptr;
  float s = 0.f;
  for (int64_t i = threadIdx.x; i < c.n; i += blockDim.x) {
    float v = E::to_f(x[i]);
    s += v * v;
  }
  s = block_reduce(s, red, SumOp(), 0.f);
  if (threadIdx.x == 0) atomicAdd(out, s);
}

}  // namespace

extern "C" int adamw_chunk_elems() { return CHUNK; }

#define ADAMW_LAUNCHERS(SUFF, ETYPE)                                                   \
  extern "C" void adamw_step_##SUFF(const void* chunks, int nchunks, float lr,         \
                                    float beta1, float beta2, float eps, float wd,     \
                                    float bc1, float bc2, float grad_scale,            \
                                    hipStream_t stream) {                              \
    adamw_kernel<ETYPE><<<dim3(nchunks), dim3(256), 0, stream>>>(                      \
        (const AdamChunkDesc*)chunks, lr, beta1, beta2, eps, wd, bc1, bc2,             \
        grad_scale);                                                                   \
  }                                                                                    \
  extern "C" void l2norm_sq_##SUFF(const void* chunks, int nchunks, float* out,        \
                                   hipStream_t stream) {                               \
    l2norm_sq_kernel<ETYPE><<<dim3(nchunks), dim3(256), 0, stream>>>(                  \
        (const NormChunkDesc*)chunks, out);                                            \
  }

ADAMW_LAUNCHERS(bf16, BF16Elem)
ADAMW_LAUNCHERS(f32, F32Elem)
