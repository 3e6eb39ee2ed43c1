// Fused flash-style scaled-dot-product attention for gfx950 (bf16, MFMA):
// forward + full backward.
//
// Replaces the reference's unfused QK^T -> fused_scale_tril_softmax -> PV
// chain (reference libai/layers/attention.py:211-253) with
// online-softmax kernels: the S x S score matrix never touches HBM in either
// direction.
//
// Shared machinery (fragment layouts verified empirically by
// csrc/tools/frag_probe.cpp on MI355X):
//   * v_mfma_f32_32x32x16_bf16 everywhere.  A-frag: lane i=l&31, k=(l>>5)*8+j.
//     B-frag: lane j=l&31, same k split.  D-frag: lane col j=l&31,
//     row=(r&3)+8*(r>>2)+4*(l>>5).
//   * row images: [ROWS][D] bf16 in LDS, 16B chunks XOR-swizzled by
//     (row&15)<<4 -> conflict-free ds_read_b128 row fragments (guide T2/G4).
//   * tr images: [r/4][d/16][4][16]-blocked bf16 in LDS; ds_read_b64_tr_b16
//     delivers transposed fragments (lane gets column d, 4 rows per read).
//   * D-layout f32 regs -> contiguous-k bf16 fragments via pack +
//     permlane32_swap (guide T12).
//   * attention dropout = per-element splitmix hash on (bh, q, kv): fwd and
//     both bwd kernels regenerate identical masks in their own lane layouts.
//
// Forward saves O and per-row logsumexp.  Backward (flash-attention-2 style):
//   Drow = rowsum(dO*O);  P = exp(scale*S - lse)
//   dV = Pd^T dO;  dPd = dO V^T;  dS = scale * P (dPd*mask - Drow)
//   dK = dS^T Q;   dQ = dS K
// bwd_kv computes dK,dV (grid over kv tiles); bwd_q computes dQ (grid over
// q tiles); both recompute S on the fly.
#include "common.h"

namespace {

typedef __bf16 bf16_t;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));
typedef float f32x16_t __attribute__((ext_vector_type(16)));

#define QB 32    // rows per wave
#define NW 4     // waves per workgroup
#define QBLK (QB * NW)
#define KVB 64   // kv rows per LDS tile (forward)

// A/B experiment knobs (guide T5: s_setprio around MFMA clusters, +4-7% on
// attention when wave roles diverge; FLASH_STAGGER decorrelates the waves'
// subtile order so they don't hit the same stall points in lockstep).
#ifdef FLASH_SETPRIO
#define PRIO_HI() __builtin_amdgcn_s_setprio(1)
#define PRIO_LO() __builtin_amdgcn_s_setprio(0)
#else
#define PRIO_HI()
#define PRIO_LO()
#endif

// native bf16 converts (v_cvt_pk_bf16_f32-class) instead of the bit-manip
// RNE helper: the repack runs per score element and the manual rounding was
// ~6 VALU per value.
__device__ __forceinline__ uint32_t pack_bf16x2(float lo, float hi) {
  union {
    uint32_t u;
    bf16_t h[2];
  } r;
  r.h[0] = (bf16_t)lo;
  r.h[1] = (bf16_t)hi;
  return r.u;
}

// D-layout regs (8 consecutive: rows {0..3}+4hi and {8..11}+4hi of a 16-row
// chunk) -> one contiguous-k bf16x8 fragment.
__device__ __forceinline__ bf16x8_t repack_chunk(const float* sv8) {
  uint32_t x0 = pack_bf16x2(sv8[0], sv8[1]);
  uint32_t z0 = pack_bf16x2(sv8[2], sv8[3]);
  uint32_t y0 = pack_bf16x2(sv8[4], sv8[5]);
  uint32_t w0 = pack_bf16x2(sv8[6], sv8[7]);
  auto rx = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
  auto rz = __builtin_amdgcn_permlane32_swap(z0, w0, false, false);
  uint32_t words[4] = {(uint32_t)rx[0], (uint32_t)rz[0], (uint32_t)rx[1],
                       (uint32_t)rz[1]};
  bf16x8_t out;
  memcpy(&out, words, 16);
  return out;
}

// ---- LDS images -----------------------------------------------------------

// row image write: 16B chunk of row `row` at col chunk col8*8, swizzled
template <int D>
__device__ __forceinline__ void row_img_write(bf16_t* lds, int row, int col8,
                                              bf16x8_t val) {
  int byte = (row * D + col8 * 8) * 2;
  byte ^= (row & 15) << 4;
  *(bf16x8_t*)((char*)lds + byte) = val;
}

// row fragment: element [row][c*16 + hi*8 .. +8]
template <int D>
__device__ __forceinline__ bf16x8_t row_img_frag(const bf16_t* lds, int row, int c,
                                                 int hi) {
  int byte = (row * D + c * 16 + hi * 8) * 2;
  byte ^= (row & 15) << 4;
  return *(const bf16x8_t*)((const char*)lds + byte);
}

// tr image write: 16B chunk (row, 8 cols at d0)
template <int D>
__device__ __forceinline__ void tr_img_write(bf16_t* lds, int row, int d0,
                                             bf16x8_t val) {
  int idx = (((row >> 2) * (D / 16) + (d0 >> 4)) * 64) + (row & 3) * 16 + (d0 & 15);
  *(bf16x8_t*)(lds + idx) = val;
}

// transposed fragment: lane gets [rbase + hi*8 + j][dt*32 + l31] for j=0..7
template <int D>
__device__ __forceinline__ bf16x8_t tr_img_frag(const bf16_t* lds, int rbase, int dt,
                                                int lane) {
  const int hi = lane >> 5;
  const int l31 = lane & 31;
  const int r0 = rbase + hi * 8;
  const int d0 = dt * 32 + (l31 & ~15);
  const bf16_t* p0 =
      lds + (((r0 >> 2) * (D / 16) + (d0 >> 4)) * 64) + (lane & 15) * 4;
  const bf16_t* p1 =
      lds + ((((r0 + 4) >> 2) * (D / 16) + (d0 >> 4)) * 64) + (lane & 15) * 4;
  bf16x4_t a = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4_t*)p0);
  bf16x4_t b = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4_t*)p1);
  bf16x8_t out;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[j] = a[j];
    out[4 + j] = b[j];
  }
  return out;
}

// cooperative staging of a [ROWS][D] global tile into row and/or tr images
template <int ROWS, int D, bool ROW_IMG, bool TR_IMG>
__device__ __forceinline__ void stage_tile(bf16_t* row_lds, bf16_t* tr_lds,
                                           const bf16_t* gp, int row0, int bound,
                                           int64_t stride, int tid) {
  constexpr int CHUNKS_TOTAL = ROWS * D / 8;
  constexpr int ITER = CDIV(CHUNKS_TOTAL, 256);
#pragma unroll
  for (int cc = 0; cc < ITER; ++cc) {
    const int flat = tid + cc * 256;
    if (flat >= CHUNKS_TOTAL) break;
    const int row = flat / (D / 8);
    const int col8 = flat % (D / 8);
    const int gr = row0 + row;
    bf16x8_t val =
        (gr < bound) ? *(const bf16x8_t*)(gp + (int64_t)gr * stride + col8 * 8)
                     : bf16x8_t{};
    if (ROW_IMG) row_img_write<D>(row_lds, row, col8, val);
    if (TR_IMG) tr_img_write<D>(tr_lds, row, col8 * 8, val);
  }
}

// Attention dropout: ONE 32-bit hash keyed on (bh, q, kv>>2) yields four
// byte-granular keep decisions (kv&3 selects the byte; p quantized to 1/256,
// well below dropout's statistical resolution).  The forward and dQ backward
// iterate kv in aligned 4-runs at fixed q, so they amortize the hash 4x; the
// dK/dV backward (q-major layout) pays one hash per element.
// 32-bit keying: (q*Sk4 + kv4) is collision-free below 2^32 (seq up to ~128k)
// and avoids gfx950's emulated 64-bit multiplies in the per-element bwd path;
// the head index feeds the mixer's second word.
__device__ __forceinline__ uint32_t drop_hash4(uint64_t seed, uint64_t bh,
                                               int64_t Sq, int64_t Sk4, int q,
                                               int kv4) {
  (void)Sq;
  return rnd_hash2(seed, (uint32_t)q * (uint32_t)Sk4 + (uint32_t)kv4,
                   (uint32_t)bh);
}

__device__ __forceinline__ float drop_keep_byte(uint32_t h4, int kv_lo,
                                                uint32_t thr8, float ks) {
  return (((h4 >> (kv_lo * 8)) & 0xFFu) >= thr8) ? ks : 0.f;
}

__device__ __forceinline__ float drop_keep(uint64_t seed, uint64_t bh, int64_t Sq,
                                           int64_t Sk, int q, int kv, uint32_t thr8,
                                           float ks) {
  uint32_t h = drop_hash4(seed, bh, Sq, Sk >> 2, q, kv >> 2);
  return drop_keep_byte(h, kv & 3, thr8, ks);
}

// ===========================================================================
// forward
// ===========================================================================
template <int D>
__global__ __launch_bounds__(256, D == 64 ? 3 : 2) void flash_fwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, bf16_t* __restrict__ o, float* __restrict__ lse,
    const int* __restrict__ kv_len, int64_t q_sb, int64_t q_ss, int64_t q_sh,
    int64_t k_sb, int64_t k_ss, int64_t k_sh, int64_t v_sb, int64_t v_ss,
    int64_t v_sh, int64_t o_sb, int64_t o_ss, int64_t o_sh, int H, int Sq, int Sk,
    float scale, float p_drop, uint64_t seed, int causal, int kv_group) {
  constexpr int DT = D / 32;
  constexpr int KC = D / 16;

  // 2 x KVB rows staged per barrier round: halves barrier count per unit of
  // compute (PMC: 43% of forward wave cycles were barrier/wait-parked)
  __shared__ __align__(16) bf16_t k_lds[2 * KVB * D];
  __shared__ __align__(16) bf16_t v_lds[2 * KVB * D];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q_block = blockIdx.x * QBLK;
  const int q_base = q_block + wave * QB;

  const int hk = h / kv_group;  // GQA: query head -> its kv head
  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* kp = k + b * k_sb + hk * k_sh;
  const bf16_t* vp = v + b * v_sb + hk * v_sh;

  // Q fragments pre-scaled by scale*log2e: scores come out of the MFMA
  // already in the exp2 domain, saving a mul per score element and the
  // exp->exp2 conversion mul (softmax runs on v_exp_f32 directly).
  const float qscale = scale * 1.4426950408889634f;
  bf16x8_t qfrag[KC];
  {
    const int qrow = min(q_base + l31, Sq - 1);
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      bf16x8_t raw = *(const bf16x8_t*)(qp + (int64_t)qrow * q_ss + c * 16 + hi * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) raw[j] = (bf16_t)((float)raw[j] * qscale);
      qfrag[c] = raw;
    }
  }

  f32x16_t oacc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) oacc[t] = f32x16_t{};
  float m_run = -3.0e38f, l_run = 0.f;

  // right-padding support: keys at/after kv_len[b] are masked out (BERT-style
  // per-sequence valid length; replaces the reference's additive -10000 pad
  // mask, attention.py:221-226)
  const int sk_eff = (kv_len != nullptr) ? kv_len[b] : Sk;
  const int kv_end = causal ? min(sk_eff, q_block + QBLK) : sk_eff;
  const int n_rounds = CDIV(kv_end, 2 * KVB);
  const float ks = (p_drop > 0.f) ? 1.0f / (1.0f - p_drop) : 1.0f;
  const int qg = q_base + l31;

  for (int round = 0; round < n_rounds; ++round) {
    const int kvR = round * 2 * KVB;
    __syncthreads();
    stage_tile<2 * KVB, D, true, false>(k_lds, nullptr, kp, kvR, Sk, k_ss, tid);
    stage_tile<2 * KVB, D, false, true>(nullptr, v_lds, vp, kvR, Sk, v_ss, tid);
    __syncthreads();

    for (int sub = 0; sub < 2; ++sub) {
    const int kv0 = kvR + sub * KVB;
    const int ro = sub * KVB;  // row offset inside the staged images
    if (kv0 >= kv_end || (causal && kv0 > q_base + QB - 1)) continue;

    // S = K x Q^T : two 32x32 tiles
    f32x16_t s0{}, s1{};
    PRIO_HI();
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      bf16x8_t ka = row_img_frag<D>(k_lds, ro + l31, c, hi);
      s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[c], s0, 0, 0, 0);
      bf16x8_t kb = row_img_frag<D>(k_lds, ro + l31 + 32, c, hi);
      s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb, qfrag[c], s1, 0, 0, 0);
    }
    PRIO_LO();

    // scores are already in the exp2 domain (Q pre-scale); m/l run in it too
    float sv[2][16];
    float pmax = -3.0e38f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kva = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      int kvb = kva + 32;
      float a = s0[r], bb = s1[r];
      if (kva >= sk_eff || (causal && kva > qg)) a = -3.0e38f;
      if (kvb >= sk_eff || (causal && kvb > qg)) bb = -3.0e38f;
      sv[0][r] = a;
      sv[1][r] = bb;
      pmax = fmaxf(pmax, fmaxf(a, bb));
    }
    pmax = fmaxf(pmax, __shfl_xor(pmax, 32));

    const float m_new = fmaxf(m_run, pmax);
    const float alpha = (m_run <= -3.0e38f) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_new);
    m_run = m_new;
    float lsum = 0.f;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float e = (sv[t][r] <= -3.0e38f) ? 0.f : __builtin_amdgcn_exp2f(sv[t][r] - m_new);
        sv[t][r] = e;
        lsum += e;
      }
    lsum += __shfl_xor(lsum, 32);
    l_run = l_run * alpha + lsum;
#pragma unroll
    for (int t = 0; t < DT; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[t][r] *= alpha;

    if (p_drop > 0.f) {
      const uint32_t thr8 = drop_threshold_u8(p_drop);
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {
          // regs r4*4..r4*4+3 cover kv run kvb..kvb+3 (4-aligned)
          const int kvb = kv0 + t * 32 + 8 * r4 + 4 * hi;
          const uint32_t h = drop_hash4(seed, bh, Sq, Sk >> 2, qg, kvb >> 2);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            sv[t][r4 * 4 + j] *= drop_keep_byte(h, j, thr8, ks);
        }
    }

    // PV: O^T[d][q] += V^T x P
    PRIO_HI();
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        bf16x8_t pfrag = repack_chunk(&sv[t][c16 * 8]);
        const int kvc = t * 32 + c16 * 16;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          bf16x8_t vfrag = tr_img_frag<D>(v_lds, ro + kvc, dt, lane);
          oacc[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pfrag, oacc[dt], 0, 0, 0);
        }
      }
    }
    PRIO_LO();
    }  // sub
  }

  const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
  bf16_t* op = o + b * o_sb + h * o_sh + (int64_t)qg * o_ss;
  if (qg < Sq) {
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        bf16x4_t pk;
#pragma unroll
        for (int j = 0; j < 4; ++j) pk[j] = (bf16_t)(oacc[dt][r4 * 4 + j] * inv_l);
        *(bf16x4_t*)(op + dt * 32 + 8 * r4 + 4 * hi) = pk;
      }
    if (hi == 0 && lse != nullptr)  // convert the exp2-domain max back to ln
      lse[((int64_t)bh * Sq) + qg] =
          m_run * 0.6931471805599453f + __logf(l_run > 0.f ? l_run : 1.f);
  }
}

// ===========================================================================
// Drow = rowsum(dO * O): one wave per 8 rows (8 lanes x 8 elems for D=64)
// out layout [B*H, Sq] (matches lse); inputs [B, Sq, H, D] strided
// ===========================================================================
template <int D>
__global__ void rowdot_kernel(const bf16_t* __restrict__ dout,
                              const bf16_t* __restrict__ o, float* __restrict__ drow,
                              int64_t sb, int64_t ss, int64_t sh, int H, int Sq,
                              int64_t nrows) {
  constexpr int LPR = D / 8;   // lanes per row
  constexpr int RPW = 64 / LPR;  // rows per wave
  const int64_t wave_id = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  const int lane = threadIdx.x & 63;
  const int64_t row = wave_id * RPW + lane / LPR;  // flat (b*Sq+q)*H+h
  if (row >= nrows) return;
  const int64_t BSH = (int64_t)Sq * H;
  const int64_t b = row / BSH;
  const int64_t rem = row % BSH;
  const int64_t qi = rem / H;
  const int64_t h = rem % H;
  const bf16_t* dp = dout + b * sb + qi * ss + h * sh + (lane % LPR) * 8;
  const bf16_t* op = o + b * sb + qi * ss + h * sh + (lane % LPR) * 8;
  u16x8 dv = *(const u16x8*)dp;
  u16x8 ov = *(const u16x8*)op;
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) s += bf2f(dv[j]) * bf2f(ov[j]);
#pragma unroll
  for (int off = LPR / 2; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
  if (lane % LPR == 0) drow[(b * H + h) * Sq + qi] = s;
}

// ===========================================================================
// backward over kv tiles: dK, dV   (block = 128 kv rows, wave owns 32)
// ===========================================================================
// FLASH_BWD_DBUF: double-buffered q/do staging (2-phase pipeline) at
// 2 blocks/CU — next tile's global loads fly during this tile's compute,
// one barrier per tile instead of two.
#ifdef FLASH_BWD_DBUF
#define BWD_KV_NBUF 2
#define BWD_KV_OCC 2
#else
#define BWD_KV_NBUF 1
#define BWD_KV_OCC 3
#endif
template <int D>
__global__ __launch_bounds__(256, D == 64 ? BWD_KV_OCC : 2) void flash_bwd_kv_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    const int* __restrict__ kv_len, bf16_t* __restrict__ dk,
    bf16_t* __restrict__ dv, int64_t q_sb, int64_t q_ss,
    int64_t q_sh, int64_t k_sb, int64_t k_ss, int64_t k_sh, int64_t v_sb,
    int64_t v_ss, int64_t v_sh, int64_t do_sb, int64_t do_ss, int64_t do_sh,
    int64_t dk_sb, int64_t dk_ss, int64_t dk_sh, int64_t dv_sb, int64_t dv_ss,
    int64_t dv_sh, int H, int Sq, int Sk, float scale, float p_drop, uint64_t seed,
    int causal, int kv_group) {
  constexpr int DT = D / 32;
  constexpr int KC = D / 16;
  // D=64: 64-row staging tiles + persistent K/V fragments.  D=128: 32-row
  // tiles and K/V fragments re-read from global per use (L2-resident) --
  // the persistent variant spills 78 VGPRs.
  constexpr int QTILE = (D == 64) ? 2 * QB : QB;
  constexpr int NSUB = QTILE / QB;

  __shared__ __align__(16) bf16_t q_row[BWD_KV_NBUF][QTILE * D];
  __shared__ __align__(16) bf16_t q_tr[BWD_KV_NBUF][QTILE * D];
  __shared__ __align__(16) bf16_t do_row[BWD_KV_NBUF][QTILE * D];
  __shared__ __align__(16) bf16_t do_tr[BWD_KV_NBUF][QTILE * D];
  __shared__ float lse_lds[BWD_KV_NBUF][QTILE];
  __shared__ float drow_lds[BWD_KV_NBUF][QTILE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  // GQA: grid.y spans B * H_kv; each kv head's dK/dV accumulates the
  // contributions of its kv_group query heads (looped below)
  const int Hkv = H / kv_group;
  const int bhk = blockIdx.y;
  const int b = bhk / Hkv, hk = bhk % Hkv;
  const int kv_block = blockIdx.x * QBLK;   // 128 kv rows per block
  const int kv_base = kv_block + wave * QB; // this wave's 32 kv rows
  const int kvg = kv_base + l31;            // lane's kv row

  const bf16_t* kp = k + b * k_sb + hk * k_sh;
  const bf16_t* vp = v + b * v_sb + hk * v_sh;

  // per-lane K and V row fragments (B-operands: lane j = kv).  Persistent in
  // registers for D=64; re-read from global (L2) per use for D=128.
  // K/V row fragments are re-read from global per use (L2-resident: one K row
  // per lane, reused across every q tile).  Keeping them persistent cost 32
  // VGPRs and held the kernel at 2 waves/SIMD.
  const int kvr_ld = min(kvg, Sk - 1);
  auto get_kf = [&](int c) {
    return *(const bf16x8_t*)(kp + (int64_t)kvr_ld * k_ss + c * 16 + hi * 8);
  };
  auto get_vf = [&](int c) {
    return *(const bf16x8_t*)(vp + (int64_t)kvr_ld * v_ss + c * 16 + hi * 8);
  };

  f32x16_t dk_acc[DT], dv_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) {
    dk_acc[t] = f32x16_t{};
    dv_acc[t] = f32x16_t{};
  }

  const float ks = (p_drop > 0.f) ? 1.0f / (1.0f - p_drop) : 1.0f;
  const int sk_eff = (kv_len != nullptr) ? kv_len[b] : Sk;
  const int qt_start = causal ? (kv_block / QTILE) : 0;
  // fully-padded kv block: accumulators stay zero, skip straight to the
  // (zero) writes.  kv_block is uniform across the block, so this does not
  // break the __syncthreads inside the loop.
  const int n_qtiles = (kv_block < sk_eff) ? CDIV(Sq, QTILE) : 0;

  for (int g = 0; g < kv_group; ++g) {
  const int h = hk * kv_group + g;
  const int bh = b * H + h;  // q-head index: lse/drow rows + dropout key
  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* dop = dout + b * do_sb + h * do_sh;

  auto stage_all = [&](int buf, int qt) {
    const int q0 = qt * QTILE;
    stage_tile<QTILE, D, true, true>(q_row[buf], q_tr[buf], qp, q0, Sq, q_ss, tid);
    stage_tile<QTILE, D, true, true>(do_row[buf], do_tr[buf], dop, q0, Sq, do_ss,
                                     tid);
    if (tid < QTILE) {
      int qr = min(q0 + tid, Sq - 1);
      lse_lds[buf][tid] = lse[(int64_t)bh * Sq + qr];
      drow_lds[buf][tid] = drow[(int64_t)bh * Sq + qr];
    }
  };

  auto compute_tile = [&](int buf, int q0) {
    if (causal && q0 + QTILE - 1 < kv_base) return;  // entirely above diag

#pragma unroll
    for (int sub_i = 0; sub_i < NSUB; ++sub_i) {
#ifdef FLASH_STAGGER
      // per-wave subtile order: decorrelates the waves' stall points
      // (subtiles are independent; accumulation order is irrelevant)
      const int sub = (sub_i + wave) % NSUB;
#else
      const int sub = sub_i;
#endif
      const int q0s = q0 + sub * QB;
      if (causal && q0s + QB - 1 < kv_base) continue;
      const int ro = sub * QB;  // row offset inside the staged images

      // S[q][kv] = Q x K^T  (lane col = kv)
      f32x16_t s{};
      f32x16_t dpd{};
      PRIO_HI();
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_t qa = row_img_frag<D>(q_row[buf], ro + l31, c, hi);
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, get_kf(c), s, 0, 0, 0);
        bf16x8_t da = row_img_frag<D>(do_row[buf], ro + l31, c, hi);
        dpd = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, get_vf(c), dpd, 0, 0, 0);
      }
      PRIO_LO();

      // per-chunk: compute Pd/dS for 8 regs, repack, feed the MFMAs — the
      // short lifetimes keep the kernel at 3 waves/SIMD (full pd[16]/ds[16]
      // arrays held it at 2).
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        float pd8[8], ds8[8];
        // dropout hashes, quad-distributed: each element needs the draw for
        // (its qrow, lane's kv4); the 4 lanes of a quad share kv4 and the 4
        // qrows of a run are consecutive, so lane (l&3) computes the run's
        // (qbase + l&3) hash and DPP quad-perm broadcasts each one back.
        // (A straight per-element hash makes the compiler hoist 16 hash
        // bases per subtile -> 25+ VGPR spills into the hot loop.)
        const uint32_t thr8 = drop_threshold_u8(p_drop);
        const int kv4q = (kv_base + (l31 & ~3)) >> 2;
        const int byte_sh = (l31 & 3) * 8;
#pragma unroll
        for (int r4 = 0; r4 < 2; ++r4) {
          const int qbase = q0s + c16 * 16 + 8 * r4 + 4 * hi;
          uint32_t hq = 0;
          if (p_drop > 0.f)
            hq = drop_hash4(seed, bh, Sq, Sk >> 2, qbase + (l31 & 3), kv4q);
          uint32_t hqj[4];
          hqj[0] = (uint32_t)__builtin_amdgcn_mov_dpp((int)hq, 0x00, 0xF, 0xF, true);
          hqj[1] = (uint32_t)__builtin_amdgcn_mov_dpp((int)hq, 0x55, 0xF, 0xF, true);
          hqj[2] = (uint32_t)__builtin_amdgcn_mov_dpp((int)hq, 0xAA, 0xF, 0xF, true);
          hqj[3] = (uint32_t)__builtin_amdgcn_mov_dpp((int)hq, 0xFF, 0xF, 0xF, true);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const int k = r4 * 4 + j;
            const int r = c16 * 8 + k;
            const int qrow = qbase + j;
            const bool valid =
                qrow < Sq && kvg < sk_eff && (!causal || qrow >= kvg);
            float p = 0.f;
            if (valid) p = __expf(s[r] * scale - lse_lds[buf][qrow - q0]);
            float keep = valid ? 1.f : 0.f;
            if (p_drop > 0.f && valid)
              keep = (((hqj[j] >> byte_sh) & 0xFFu) >= thr8) ? ks : 0.f;
            pd8[k] = p * keep;
            ds8[k] = valid
                         ? scale * p * (dpd[r] * keep - drow_lds[buf][qrow - q0])
                         : 0.f;
          }
        }
        bf16x8_t pdf = repack_chunk(pd8);
        bf16x8_t dsf = repack_chunk(ds8);
        const int qc = ro + c16 * 16;
        PRIO_HI();
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          bf16x8_t dof = tr_img_frag<D>(do_tr[buf], qc, dt, lane);
          dv_acc[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, pdf, dv_acc[dt], 0, 0, 0);
          bf16x8_t qtf = tr_img_frag<D>(q_tr[buf], qc, dt, lane);
          dk_acc[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(qtf, dsf, dk_acc[dt], 0, 0, 0);
        }
        PRIO_LO();
      }
    }
  };  // compute_tile

#ifdef FLASH_BWD_DBUF
  {
    int cur = 0;
    if (qt_start < n_qtiles) {
      __syncthreads();  // previous g's reads complete before re-staging
      stage_all(0, qt_start);
      __syncthreads();
    }
    for (int qt = qt_start; qt < n_qtiles; ++qt) {
      if (qt + 1 < n_qtiles)
        stage_all(cur ^ 1, qt + 1);  // next tile's loads fly under compute
      compute_tile(cur, qt * QTILE);
      __syncthreads();
      cur ^= 1;
    }
  }
#else
  for (int qt = qt_start; qt < n_qtiles; ++qt) {
    __syncthreads();
    stage_all(0, qt);
    __syncthreads();
    compute_tile(0, qt * QTILE);
  }
#endif

  }  // group loop (GQA)

  // write dK, dV rows (lane owns kv row kvg; d in 4-element runs)
  if (kvg < Sk) {
    bf16_t* dkp = dk + b * dk_sb + hk * dk_sh + (int64_t)kvg * dk_ss;
    bf16_t* dvp = dv + b * dv_sb + hk * dv_sh + (int64_t)kvg * dv_ss;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        bf16x4_t a, c;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          a[j] = (bf16_t)dk_acc[dt][r4 * 4 + j];
          c[j] = (bf16_t)dv_acc[dt][r4 * 4 + j];
        }
        *(bf16x4_t*)(dkp + dt * 32 + 8 * r4 + 4 * hi) = a;
        *(bf16x4_t*)(dvp + dt * 32 + 8 * r4 + 4 * hi) = c;
      }
  }
}

// ===========================================================================
// backward over q tiles: dQ   (block = 128 q rows, wave owns 32; fwd-like)
// ===========================================================================
template <int D>
__global__ __launch_bounds__(256, 2) void flash_bwd_q_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    const int* __restrict__ kv_len, bf16_t* __restrict__ dq, int64_t q_sb,
    int64_t q_ss, int64_t q_sh, int64_t k_sb,
    int64_t k_ss, int64_t k_sh, int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int64_t do_sb, int64_t do_ss, int64_t do_sh, int64_t dq_sb, int64_t dq_ss,
    int64_t dq_sh, int H, int Sq, int Sk, float scale, float p_drop, uint64_t seed,
    int causal, int kv_group) {
  constexpr int DT = D / 32;
  constexpr int KC = D / 16;
  constexpr int KVTILE = (D == 64) ? 2 * QB : QB;
  constexpr int NSUB = KVTILE / QB;

  __shared__ __align__(16) bf16_t k_row[KVTILE * D];
  __shared__ __align__(16) bf16_t k_tr[KVTILE * D];
  __shared__ __align__(16) bf16_t v_row[KVTILE * D];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q_block = blockIdx.x * QBLK;
  const int q_base = q_block + wave * QB;
  const int qg = q_base + l31;

  const int hk = h / kv_group;  // GQA
  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* kp = k + b * k_sb + hk * k_sh;
  const bf16_t* vp = v + b * v_sb + hk * v_sh;
  const bf16_t* dop = dout + b * do_sb + h * do_sh;

  // per-lane Q and dO row fragments (B-operands: lane j = q); persistent for
  // D=64, re-read from global (L2) per use for D=128 (register budget).
  const int qr_ld = min(qg, Sq - 1);
  bf16x8_t qf[D == 64 ? KC : 1], dof[D == 64 ? KC : 1];
  if constexpr (D == 64) {
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      qf[c] = *(const bf16x8_t*)(qp + (int64_t)qr_ld * q_ss + c * 16 + hi * 8);
      dof[c] = *(const bf16x8_t*)(dop + (int64_t)qr_ld * do_ss + c * 16 + hi * 8);
    }
  }
  auto get_qf = [&](int c) {
    if constexpr (D == 64) return qf[c];
    else return *(const bf16x8_t*)(qp + (int64_t)qr_ld * q_ss + c * 16 + hi * 8);
  };
  auto get_dof = [&](int c) {
    if constexpr (D == 64) return dof[c];
    else return *(const bf16x8_t*)(dop + (int64_t)qr_ld * do_ss + c * 16 + hi * 8);
  };
  const float lse_lane = lse[(int64_t)bh * Sq + min(qg, Sq - 1)];
  const float drow_lane = drow[(int64_t)bh * Sq + min(qg, Sq - 1)];

  f32x16_t dq_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) dq_acc[t] = f32x16_t{};

  const float ks = (p_drop > 0.f) ? 1.0f / (1.0f - p_drop) : 1.0f;
  const int sk_eff = (kv_len != nullptr) ? kv_len[b] : Sk;
  const int kv_end = causal ? min(sk_eff, q_block + QBLK) : sk_eff;
  const int n_tiles = CDIV(kv_end, KVTILE);

  for (int tile = 0; tile < n_tiles; ++tile) {
    const int kv0 = tile * KVTILE;
    __syncthreads();
    stage_tile<KVTILE, D, true, true>(k_row, k_tr, kp, kv0, Sk, k_ss, tid);
    stage_tile<KVTILE, D, true, false>(v_row, nullptr, vp, kv0, Sk, v_ss, tid);
    __syncthreads();

    if (causal && kv0 > q_base + QB - 1) continue;

#pragma unroll
    for (int sub = 0; sub < NSUB; ++sub) {
      const int kv0s = kv0 + sub * QB;
      if (kv0s >= kv_end || (causal && kv0s > q_base + QB - 1)) continue;
      const int ro = sub * QB;

      // S^T[kv][q] = K x Q^T ;  dPd^T[kv][q] = V x dO^T   (lane col = q)
      f32x16_t st{};
      f32x16_t dpdt{};
      PRIO_HI();
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_t ka = row_img_frag<D>(k_row, ro + l31, c, hi);
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, get_qf(c), st, 0, 0, 0);
        bf16x8_t va = row_img_frag<D>(v_row, ro + l31, c, hi);
        dpdt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, get_dof(c), dpdt, 0, 0, 0);
      }
      PRIO_LO();

      float ds[16];
      const uint32_t thr8 = drop_threshold_u8(p_drop);
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        const int kvb = kv0s + 8 * r4 + 4 * hi;  // 4-aligned kv run
        const uint32_t h = (p_drop > 0.f)
                               ? drop_hash4(seed, bh, Sq, Sk >> 2, qg, kvb >> 2)
                               : 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int r = r4 * 4 + j;
          const int kv = kvb + j;
          const bool valid = kv < sk_eff && qg < Sq && (!causal || kv <= qg);
          float p = valid ? __expf(st[r] * scale - lse_lane) : 0.f;
          float keep = (p_drop > 0.f && valid) ? drop_keep_byte(h, j, thr8, ks)
                                               : (valid ? 1.f : 0.f);
          ds[r] = valid ? scale * p * (dpdt[r] * keep - drow_lane) : 0.f;
        }
      }

      // dQ^T[d][q] += K^T x dS^T
      PRIO_HI();
#pragma unroll
      for (int c16 = 0; c16 < 2; ++c16) {
        bf16x8_t dsf = repack_chunk(&ds[c16 * 8]);
        const int kvc = ro + c16 * 16;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          bf16x8_t ktf = tr_img_frag<D>(k_tr, kvc, dt, lane);
          dq_acc[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(ktf, dsf, dq_acc[dt], 0, 0, 0);
        }
      }
      PRIO_LO();
    }
  }

  if (qg < Sq) {
    bf16_t* dqp = dq + b * dq_sb + h * dq_sh + (int64_t)qg * dq_ss;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        bf16x4_t a;
#pragma unroll
        for (int j = 0; j < 4; ++j) a[j] = (bf16_t)dq_acc[dt][r4 * 4 + j];
        *(bf16x4_t*)(dqp + dt * 32 + 8 * r4 + 4 * hi) = a;
      }
  }
}

// dropout-mask application for external recompute paths / tests
template <class E>
__global__ void attn_dropout_apply_kernel(typename E::T* __restrict__ x, int64_t BH,
                                          int64_t Sq, int64_t Sk, float p,
                                          float keep_scale, uint64_t seed) {
  const int64_t total = BH * Sq * Sk;
  const uint32_t thr8 = drop_threshold_u8(p);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = i / Sk;
    const int kv = (int)(i % Sk);
    uint32_t h = rnd_hash2(seed, (uint32_t)(row % Sq) * (uint32_t)(Sk >> 2) +
                                     (uint32_t)(kv >> 2),
                           (uint32_t)(row / Sq));
    float kp = drop_keep_byte(h, kv & 3, thr8, keep_scale);
    x[i] = E::from_f(E::to_f(x[i]) * kp);
  }
}

}  // namespace

extern "C" void flash_fwd_bf16(const void* q, const void* k, const void* v, void* o,
                               float* lse, const int* kv_len, int64_t q_sb,
                               int64_t q_ss, int64_t q_sh, int64_t k_sb,
                               int64_t k_ss, int64_t k_sh, int64_t v_sb,
                               int64_t v_ss, int64_t v_sh, int64_t o_sb, int64_t o_ss,
                               int64_t o_sh, int B, int H, int Sq, int Sk, int D,
                               float scale, float p_drop, uint64_t seed, int causal,
                               int kv_group, hipStream_t stream) {
  dim3 grid(CDIV(Sq, QBLK), B * H);
  dim3 block(256);
  if (D == 64)
    flash_fwd_kernel<64><<<grid, block, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, lse,
        kv_len, q_sb, q_ss, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_ss,
        o_sh, H, Sq, Sk, scale, p_drop, seed, causal, kv_group);
  else if (D == 128)
    flash_fwd_kernel<128><<<grid, block, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, lse,
        kv_len, q_sb, q_ss, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_ss,
        o_sh, H, Sq, Sk, scale, p_drop, seed, causal, kv_group);
}

extern "C" void flash_bwd_bf16(
    const void* q, const void* k, const void* v, const void* o, const void* dout,
    const float* lse, float* drow_ws, const int* kv_len, void* dq, void* dk,
    void* dv, int64_t q_sb,
    int64_t q_ss, int64_t q_sh, int64_t k_sb, int64_t k_ss, int64_t k_sh,
    int64_t v_sb, int64_t v_ss, int64_t v_sh, int64_t o_sb, int64_t o_ss,
    int64_t o_sh, int64_t do_sb, int64_t do_ss, int64_t do_sh, int64_t dq_sb,
    int64_t dq_ss, int64_t dq_sh, int64_t dk_sb, int64_t dk_ss, int64_t dk_sh,
    int64_t dv_sb, int64_t dv_ss, int64_t dv_sh, int B, int H, int Sq, int Sk, int D,
    float scale, float p_drop, uint64_t seed, int causal, int kv_group,
    hipStream_t stream) {
  // Drow = rowsum(dO * O)  (dO and O share layout; use dO's strides for both:
  // the wrapper guarantees o was allocated with the same layout)
  {
    int rpw = 64 / (D / 8);
    int64_t rows = (int64_t)B * Sq * H;
    int64_t waves = CDIV(rows, rpw);
    uint32_t blocks = (uint32_t)CDIV(waves * 64, 256);
    if (D == 64)
      rowdot_kernel<64><<<dim3(blocks), 256, 0, stream>>>(
          (const bf16_t*)dout, (const bf16_t*)o, drow_ws, do_sb, do_ss, do_sh, H, Sq,
          rows);
    else
      rowdot_kernel<128><<<dim3(blocks), 256, 0, stream>>>(
          (const bf16_t*)dout, (const bf16_t*)o, drow_ws, do_sb, do_ss, do_sh, H, Sq,
          rows);
  }
#define BWD_ARGS_KV                                                                  \
  (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (const bf16_t*)dout, lse,    \
      drow_ws, kv_len, (bf16_t*)dk, (bf16_t*)dv, q_sb, q_ss, q_sh, k_sb, k_ss,       \
      k_sh, v_sb, v_ss, v_sh, do_sb, do_ss, do_sh, dk_sb, dk_ss, dk_sh, dv_sb,       \
      dv_ss, dv_sh, H, Sq, Sk, scale, p_drop, seed, causal, kv_group
#define BWD_ARGS_Q                                                                   \
  (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (const bf16_t*)dout, lse,    \
      drow_ws, kv_len, (bf16_t*)dq, q_sb, q_ss, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss,  \
      v_sh, do_sb, do_ss, do_sh, dq_sb, dq_ss, dq_sh, H, Sq, Sk, scale, p_drop,      \
      seed, causal, kv_group
  dim3 block(256);
  const int Hkv = H / kv_group;  // bwd_kv grid spans the KV heads
  if (D == 64) {
    flash_bwd_kv_kernel<64>
        <<<dim3(CDIV(Sk, QBLK), B * Hkv), block, 0, stream>>>(BWD_ARGS_KV);
    flash_bwd_q_kernel<64>
        <<<dim3(CDIV(Sq, QBLK), B * H), block, 0, stream>>>(BWD_ARGS_Q);
  } else if (D == 128) {
    flash_bwd_kv_kernel<128>
        <<<dim3(CDIV(Sk, QBLK), B * Hkv), block, 0, stream>>>(BWD_ARGS_KV);
    flash_bwd_q_kernel<128>
        <<<dim3(CDIV(Sq, QBLK), B * H), block, 0, stream>>>(BWD_ARGS_Q);
  }
}

#define ATTN_DROP_LAUNCHER(SUFF, ETYPE)                                            \
  extern "C" void attn_dropout_apply_##SUFF(void* x, int64_t BH, int64_t Sq,       \
                                            int64_t Sk, float p, uint64_t seed,    \
                                            hipStream_t stream) {                  \
    int64_t n = BH * Sq * Sk;                                                      \
    int64_t g = CDIV(n, 256);                                                      \
    if (g > 4096) g = 4096;                                                        \
    attn_dropout_apply_kernel<ETYPE><<<dim3((uint32_t)g), dim3(256), 0, stream>>>( \
        (ETYPE::T*)x, BH, Sq, Sk, p, 1.0f / (1.0f - p), seed);                     \
  }

ATTN_DROP_LAUNCHER(bf16, BF16Elem)
ATTN_DROP_LAUNCHER(f32, F32Elem)
