// Fused flash-style scaled-dot-product attention for gfx950 (bf16, MFMA).
//
// Replaces the reference's unfused QK^T -> fused_scale_tril_softmax -> PV
// chain (reference libai/layers/attention.py:211-253) with a
// single online-softmax kernel: the S x S score matrix never touches HBM.
//
// Structure (v1, per the CDNA4 guide's 8-warp attention recipe, simplified):
//   * workgroup = 4 waves x 64 = 256 threads; each wave owns QB=32 query rows
//     -> 128 q rows per workgroup; K/V tiles of KVB=64 keys staged in LDS and
//     shared by the 4 waves.
//   * QK^T via v_mfma_f32_32x32x16_bf16 with SWAPPED operands
//     (S[kv][q] = K x Q^T), so each lane holds score slices of ONE query
//     (col = lane&31) and the online-softmax row reduce is 16 in-lane f32 ops
//     + one shfl_xor(32).
//   * P -> bf16 B-fragments in-register via pack + permlane32_swap (guide
//     T12); PV uses V B-fragments delivered transposed by ds_read_b64_tr_b16
//     from a [kv/4][d/16][4][16]-blocked LDS image (guide T10; mapping
//     verified empirically in csrc/tools/frag_probe.cpp).
//   * K LDS tile is XOR-swizzled (byte ^= (row&15)<<4) for conflict-free
//     ds_read_b128 A-fragments (guide T2/G4).
//   * causal masking is an in-kernel predicate; attention dropout is philox
//     keyed on (batch*head, q, kv) so the backward regenerates the mask.
//
// Saved for backward: O and per-row logsumexp (lse = m + log(l)).
#include "common.h"

namespace {

typedef __bf16 bf16_t;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));
typedef float f32x16_t __attribute__((ext_vector_type(16)));

#define QB 32    // q rows per wave
#define NW 4     // waves per workgroup
#define QBLK (QB * NW)
#define KVB 64   // kv rows per LDS tile

// LDS K tile: [KVB][D] row-major bf16, 16-byte chunks XOR-swizzled by row.
__device__ __forceinline__ int k_lds_off(int row, int col /*bf16 units*/) {
  int byte = row * 128 /*64*2B... D template handles*/ + col * 2;
  return byte;  // swizzle applied by caller (needs D)
}

// pack two f32 into one u32 of two bf16 (compiler emits v_cvt_pk_bf16_f32)
__device__ __forceinline__ uint32_t pack_bf16x2(float lo, float hi) {
  union {
    uint32_t u;
    uint16_t h[2];
  } r;
  r.h[0] = f2bf(lo);
  r.h[1] = f2bf(hi);
  return r.u;
}

template <int D>  // head dim: 64 or 128
__global__ __launch_bounds__(256, 2) void flash_fwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, bf16_t* __restrict__ o, float* __restrict__ lse,
    int64_t q_sb, int64_t q_ss, int64_t q_sh,  // elem strides: batch, seq, head
    int64_t k_sb, int64_t k_ss, int64_t k_sh, int64_t v_sb, int64_t v_ss,
    int64_t v_sh, int64_t o_sb, int64_t o_ss, int64_t o_sh, int H, int Sq, int Sk,
    float scale, float p_drop, uint64_t seed, int causal) {
  constexpr int DT = D / 32;      // 32-wide d tiles (O accum tiles)
  constexpr int KC = D / 16;      // 16-deep k-chunks per QK^T mfma chain
  static_assert(D == 64 || D == 128, "head dim 64/128 only");

  __shared__ __align__(16) bf16_t k_lds[KVB * D];
  __shared__ __align__(16) bf16_t v_lds[KVB * D];  // [kv/4][d/16][4][16] blocks

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int hi = lane >> 5;  // k-half selector in A/B fragments

  const int bh = blockIdx.y;  // b * H + h
  const int b = bh / H, h = bh % H;
  const int q_block = blockIdx.x * QBLK;
  const int q_base = q_block + wave * QB;  // this wave's first q row

  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* kp = k + b * k_sb + h * k_sh;
  const bf16_t* vp = v + b * v_sb + h * v_sh;

  // ---- load this wave's Q B-fragments (lane: q row = l31, k = hi*8+0..7) --
  bf16x8_t qfrag[KC];
  {
    const int qrow = q_base + l31;
#pragma unroll
    for (int c = 0; c < KC; ++c)
      qfrag[c] = *(const bf16x8_t*)(qp + (int64_t)qrow * q_ss + c * 16 + hi * 8);
  }

  // ---- accumulators ----
  f32x16_t oacc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) oacc[t] = f32x16_t{};
  float m_run = -3.0e38f, l_run = 0.f;

  const int kv_end = causal ? min(Sk, q_block + QBLK) : Sk;
  const int n_tiles = CDIV(kv_end, KVB);
  const float keep_scale = (p_drop > 0.f) ? 1.0f / (1.0f - p_drop) : 1.0f;

  for (int tile = 0; tile < n_tiles; ++tile) {
    const int kv0 = tile * KVB;
    // ---- stage K and V tiles (all 256 threads) ----
    // thread t: row = t>>2 (+32 for second half when 256*16B < tile bytes)
    __syncthreads();
    {
      // KVB * D bf16 = KVB*D*2 bytes; each thread writes (KVB*D/8)/256 chunks
      constexpr int CHUNKS = (KVB * D / 8) / 256;
#pragma unroll
      for (int cc = 0; cc < CHUNKS; ++cc) {
        const int flat = tid + cc * 256;         // 16B chunk index
        const int row = flat / (D / 8);          // kv row
        const int col8 = flat % (D / 8);         // 8-elem col chunk
        const int kvr = kv0 + row;
        bf16x8_t kv8 = (kvr < Sk)
                           ? *(const bf16x8_t*)(kp + (int64_t)kvr * k_ss + col8 * 8)
                           : bf16x8_t{};
        // K: row-major with (row&15)<<4 byte XOR swizzle
        int kbyte = (row * D + col8 * 8) * 2;
        kbyte ^= (row & 15) << 4;
        *(bf16x8_t*)((char*)k_lds + kbyte) = kv8;
        bf16x8_t vv8 = (kvr < Sk)
                           ? *(const bf16x8_t*)(vp + (int64_t)kvr * v_ss + col8 * 8)
                           : bf16x8_t{};
        // V: [kv>>2][d>>4][kv&3][d&15] blocked image (for tr16 reads)
        const int d0 = col8 * 8;
        int vidx = (((row >> 2) * (D / 16) + (d0 >> 4)) * 64) + (row & 3) * 16 +
                   (d0 & 15);
        *(bf16x8_t*)(v_lds + vidx) = vv8;
      }
    }
    __syncthreads();

    if (causal && kv0 > q_base + QB - 1) continue;  // tile beyond this wave

    // ---- S = K x Q^T : two 32x32 tiles (kv halves) ----
    f32x16_t s0{}, s1{};
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      // A-frag: K[kv = half*32 + l31][hs = c*16 + hi*8 + 0..7]
      int byte0 = ((l31)*D + c * 16 + hi * 8) * 2;
      byte0 ^= (l31 & 15) << 4;
      bf16x8_t ka = *(const bf16x8_t*)((char*)k_lds + byte0);
      s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[c], s0, 0, 0, 0);
      int byte1 = ((l31 + 32) * D + c * 16 + hi * 8) * 2;
      byte1 ^= ((l31 + 32) & 15) << 4;
      bf16x8_t kb = *(const bf16x8_t*)((char*)k_lds + byte1);
      s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb, qfrag[c], s1, 0, 0, 0);
    }

    // ---- online softmax (lane owns query q_base + l31) ----
    const int qg = q_base + l31;  // this lane's global q row
    float sv[2][16];
    float pmax = -3.0e38f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kva = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      int kvb = kva + 32;
      float a = s0[r] * scale, bvl = s1[r] * scale;
      if (kva >= Sk || (causal && kva > qg)) a = -3.0e38f;
      if (kvb >= Sk || (causal && kvb > qg)) bvl = -3.0e38f;
      sv[0][r] = a;
      sv[1][r] = bvl;
      pmax = fmaxf(pmax, fmaxf(a, bvl));
    }
    pmax = fmaxf(pmax, __shfl_xor(pmax, 32));

    const float m_new = fmaxf(m_run, pmax);
    const float alpha = (m_run <= -3.0e38f) ? 0.f : __expf(m_run - m_new);
    m_run = m_new;
    float lsum = 0.f;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float e = (sv[t][r] <= -3.0e38f) ? 0.f : __expf(sv[t][r] - m_new);
        sv[t][r] = e;
        lsum += e;
      }
    lsum += __shfl_xor(lsum, 32);
    l_run = l_run * alpha + lsum;
    // rescale O accumulators (each lane's accum belongs to its own query)
#pragma unroll
    for (int t = 0; t < DT; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[t][r] *= alpha;

    // ---- dropout on P (philox on (bh, q, kv)) ----
    if (p_drop > 0.f) {
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {
          // 4 consecutive regs share a philox call; counter from first kv
          uint32_t rnd[4];
          int kvf = kv0 + t * 32 + (r4 * 8) + 4 * hi;  // kv of reg r4*4
          // unique counter per (bh, q, kv group of 4): kv pattern within a
          // group r: (r&3) consecutive -> use kv base of the 4-reg run
          uint64_t ctr =
              (((uint64_t)bh * Sq + qg) * (uint64_t)Sk + kvf) >> 2;
          philox4(seed, ctr, rnd);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float kp_ = (u32_to_uniform(rnd[j]) > p_drop) ? keep_scale : 0.f;
            sv[t][r4 * 4 + j] *= kp_;
          }
        }
      }
    }

    // ---- P -> bf16 B-fragments via pack + permlane32_swap ----
    // chunk c covers kv = kv0 + t*32 + c16*16; B word w: lanes<32 kv pairs
    // (2w, 2w+1), lanes>=32 kv pairs (8+2w, 8+2w+1) of the chunk.
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      uint32_t pw[8];  // two chunks x 4 words
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        // regs for kv offsets {0,1,2,3} of this chunk: r = c16*8 + {0..3}
        // regs for kv offsets {8..11}: r = c16*8 + {4..7}
        uint32_t x0 = pack_bf16x2(sv[t][c16 * 8 + 0], sv[t][c16 * 8 + 1]);
        uint32_t z0 = pack_bf16x2(sv[t][c16 * 8 + 2], sv[t][c16 * 8 + 3]);
        uint32_t y0 = pack_bf16x2(sv[t][c16 * 8 + 4], sv[t][c16 * 8 + 5]);
        uint32_t w0 = pack_bf16x2(sv[t][c16 * 8 + 6], sv[t][c16 * 8 + 7]);
        auto rx = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        auto rz = __builtin_amdgcn_permlane32_swap(z0, w0, false, false);
        pw[c16 * 4 + 0] = rx[0];
        pw[c16 * 4 + 1] = rz[0];
        pw[c16 * 4 + 2] = rx[1];
        pw[c16 * 4 + 3] = rz[1];
      }
      // ---- PV: O[q][d] += P x V, B-frag of V via tr16 reads ----
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        bf16x8_t pfrag;
        memcpy(&pfrag, &pw[c16 * 4], 16);
        const int kvc = t * 32 + c16 * 16;  // chunk kv base within tile
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          // lane needs V[kv = kvc + hi*8 + j'][d = dt*32 + l31]
          // image block of (kv row group, d group): two tr reads (4 kv each)
          bf16x4_t va, vb2;
          {
            const int kvr = kvc + hi * 8;          // rows kvr..kvr+3
            const int d0 = dt * 32 + (l31 & ~15);  // 16-d group
            int base = (((kvr >> 2) * (D / 16) + (d0 >> 4)) * 64);
            va = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4_t*)(v_lds + base +
                                                              (lane & 15) * 4));
            const int kvr2 = kvr + 4;
            int base2 = (((kvr2 >> 2) * (D / 16) + (d0 >> 4)) * 64);
            vb2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4_t*)(v_lds + base2 +
                                                              (lane & 15) * 4));
          }
          bf16x8_t vfrag;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            vfrag[j] = va[j];
            vfrag[4 + j] = vb2[j];
          }
          oacc[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pfrag, oacc[dt], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: O = oacc / l, write [q][d]; lse = m + log(l) ----
  const int qg = q_base + l31;
  const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
  bf16_t* op = o + b * o_sb + h * o_sh + (int64_t)qg * o_ss;
  if (qg < Sq) {
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        // regs r4*4..r4*4+3 are d = dt*32 + {0..3} + 8*r4 + 4*hi  (contiguous)
        u16x4 pack;
#pragma unroll
        for (int j = 0; j < 4; ++j)
          pack[j] = f2bf(oacc[dt][r4 * 4 + j] * inv_l);
        *(u16x4*)(op + dt * 32 + 8 * r4 + 4 * hi) = pack;
      }
    }
    if (hi == 0 && lse != nullptr)
      lse[((int64_t)bh * Sq) + qg] = m_run + __logf(l_run > 0.f ? l_run : 1.f);
  }
}

}  // namespace

extern "C" void flash_fwd_bf16(const void* q, const void* k, const void* v, void* o,
                               float* lse, int64_t q_sb, int64_t q_ss, int64_t q_sh,
                               int64_t k_sb, int64_t k_ss, int64_t k_sh, int64_t v_sb,
                               int64_t v_ss, int64_t v_sh, int64_t o_sb, int64_t o_ss,
                               int64_t o_sh, int B, int H, int Sq, int Sk, int D,
                               float scale, float p_drop, uint64_t seed, int causal,
                               hipStream_t stream) {
  dim3 grid(CDIV(Sq, QBLK), B * H);
  dim3 block(256);
  if (D == 64)
    flash_fwd_kernel<64><<<grid, block, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, lse, q_sb,
        q_ss, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_ss, o_sh, H, Sq, Sk,
        scale, p_drop, seed, causal);
  else if (D == 128)
    flash_fwd_kernel<128><<<grid, block, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, lse, q_sb,
        q_ss, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_ss, o_sh, H, Sq, Sk,
        scale, p_drop, seed, causal);
}

// ---------------------------------------------------------------------------
// apply the SAME philox attention-dropout mask the flash forward used, to an
// arbitrary [B*H, Sq, Sk] tensor (used by the recompute backward):
//   x *= keep(seed, bh, q, kv) / (1-p)
// ---------------------------------------------------------------------------
namespace {
template <class E>
__global__ void attn_dropout_apply_kernel(typename E::T* __restrict__ x, int64_t BH,
                                          int64_t Sq, int64_t Sk, float p,
                                          float keep_scale, uint64_t seed) {
  const int64_t total4 = BH * Sq * (Sk >> 2);
  for (int64_t i4 = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i4 < total4;
       i4 += (int64_t)gridDim.x * blockDim.x) {
    uint32_t rnd[4];
    philox4(seed, (uint64_t)i4, rnd);
    typename E::T* xp = x + i4 * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float kp = (u32_to_uniform(rnd[j]) > p) ? keep_scale : 0.f;
      xp[j] = E::from_f(E::to_f(xp[j]) * kp);
    }
  }
}
}  // namespace

#define ATTN_DROP_LAUNCHER(SUFF, ETYPE)                                            \
  extern "C" void attn_dropout_apply_##SUFF(void* x, int64_t BH, int64_t Sq,       \
                                            int64_t Sk, float p, uint64_t seed,    \
                                            hipStream_t stream) {                  \
    int64_t n4 = BH * Sq * (Sk >> 2);                                              \
    int64_t g = CDIV(n4, 256);                                                     \
    if (g > 4096) g = 4096;                                                        \
    attn_dropout_apply_kernel<ETYPE><<<dim3((uint32_t)g), dim3(256), 0, stream>>>( \
        (ETYPE::T*)x, BH, Sq, Sk, p, 1.0f / (1.0f - p), seed);                     \
  }

ATTN_DROP_LAUNCHER(bf16, BF16Elem)
ATTN_DROP_LAUNCHER(f32, F32Elem)
