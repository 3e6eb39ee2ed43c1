// Fused single-query decode attention for gfx950 (bf16): one new token's
// q against the KV cache, online softmax, no materialized scores.
//
// Replaces the unfused bmm+softmax+bmm decode chain (reference capability:
// flow._C.fused_multi_head_attention_inference_v2,
// projects/GLM/layers/attention_layer.py:111 — SURVEY.md K16 "fused MHA
// inference for serving").
//
// Decomposition: one workgroup per (batch, q-head); each of the 4 waves owns
// a strided slice of the cache keys (lane = one key: the k-row dot is a
// coalesced 16B-vector stream), keeps a per-wave online (m, l, acc[d=lane])
// state — the V accumulation indexes dims by LANE so each j-step reads one
// coalesced 128B line of V — and the 4 wave states merge through LDS.
// GQA: query head h reads kv head h / kv_group.
#include "common.h"

namespace {

typedef __bf16 bf16_t;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

template <int D>
__global__ __launch_bounds__(256, 4) void flash_decode_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
    const int* __restrict__ kv_len, int64_t q_sb, int64_t q_sh, int64_t k_sb,
    int64_t k_ss, int64_t k_sh, int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int64_t o_sb, int64_t o_sh, int H, int Skv, float scale, int kv_group) {
  constexpr int DPL = D / 64;  // dims per lane
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  const int hk = h / kv_group;
  const int skv = (kv_len != nullptr) ? kv_len[b] : Skv;

  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* kp = k + b * k_sb + hk * k_sh;
  const bf16_t* vp = v + b * v_sb + hk * v_sh;

  // q in bf16 registers (2 VGPRs per 8 values; scale folded after the dot)
  u16x8 qreg[D / 8];
#pragma unroll
  for (int c = 0; c < D / 8; ++c) qreg[c] = *(const u16x8*)(qp + c * 8);

  __shared__ float p_lds[4][64];
  __shared__ float merge_m[4], merge_l[4];
  __shared__ float merge_acc[4][D];

  float m_run = -3.0e38f, l_run = 0.f;
  float acc[DPL];
#pragma unroll
  for (int t = 0; t < DPL; ++t) acc[t] = 0.f;

  // each wave walks keys [wave*64 + chunk*256 ...)
  for (int key0 = wave * 64; key0 < skv; key0 += 256) {
    const int key = key0 + lane;
    float s = -3.0e38f;
    if (key < skv) {
      const bf16_t* kr = kp + (int64_t)key * k_ss;
      float dot = 0.f;
#pragma unroll
      for (int c = 0; c < D / 8; ++c) {
        u16x8 kv8 = *(const u16x8*)(kr + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) dot += bf2f(qreg[c][j]) * bf2f(kv8[j]);
      }
      s = dot * scale;
    }
    const float cmax = wave_reduce_max(s);
    const float m_new = fmaxf(m_run, cmax);
    const float alpha = (m_run <= -3.0e38f) ? 0.f : __expf(m_run - m_new);
    const float p = (s <= -3.0e38f) ? 0.f : __expf(s - m_new);
    m_run = m_new;
    l_run = l_run * alpha + wave_reduce_sum(p);
#pragma unroll
    for (int t = 0; t < DPL; ++t) acc[t] *= alpha;
    p_lds[wave][lane] = p;
    // no barrier needed: p_lds[wave] is only read by this wave
    const int limit = min(64, skv - key0);
    for (int j = 0; j < limit; ++j) {
      const float pj = p_lds[wave][j];
      const bf16_t* vr = vp + (int64_t)(key0 + j) * v_ss;
#pragma unroll
      for (int t = 0; t < DPL; ++t)
        acc[t] += pj * bf2f(*(const uint16_t*)(vr + t * 64 + lane));
    }
  }

  // merge the 4 wave states (standard flash-decode combine)
  if (lane == 0) {
    merge_m[wave] = m_run;
    merge_l[wave] = l_run;
  }
#pragma unroll
  for (int t = 0; t < DPL; ++t) merge_acc[wave][t * 64 + lane] = acc[t];
  __syncthreads();
  if (wave == 0) {
    float M = -3.0e38f;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, merge_m[w]);
    float L = 0.f;
    float out[DPL];
#pragma unroll
    for (int t = 0; t < DPL; ++t) out[t] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float a = (merge_m[w] <= -3.0e38f) ? 0.f : __expf(merge_m[w] - M);
      L += merge_l[w] * a;
#pragma unroll
      for (int t = 0; t < DPL; ++t) out[t] += a * merge_acc[w][t * 64 + lane];
    }
    const float inv = (L > 0.f) ? 1.0f / L : 0.f;
    bf16_t* op = o + b * o_sb + h * o_sh;
#pragma unroll
    for (int t = 0; t < DPL; ++t)
      *(uint16_t*)(op + t * 64 + lane) = f2bf(out[t] * inv);
  }
}

// ---------------------------------------------------------------------------
// Split-KV variant (flash-decoding): one workgroup per (batch, head, 512-key
// split).  B*H workgroups underfill the 256-CU chip for serving batch sizes
// (b32 h16 = 512 WGs -> measured 36 us/call, 25% of the captured decode
// step); splitting the KV range multiplies the grid by ceil(Skv/512).
// Split boundaries are FIXED multiples of 512 (not Skv/S), so each split's
// online-softmax chain is identical for any Skv and the cross-split merge
// of empty splits is an exact no-op -> bitwise-identical outputs across
// cache capacities (the captured-vs-eager greedy parity tests rely on it).
// ---------------------------------------------------------------------------
constexpr int SPLIT_KEYS = 512;

template <int D>
__global__ __launch_bounds__(256, 4) void flash_decode_split_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, float* __restrict__ part_m,
    float* __restrict__ part_l, float* __restrict__ part_acc,
    const int* __restrict__ kv_len, int64_t q_sb, int64_t q_sh, int64_t k_sb,
    int64_t k_ss, int64_t k_sh, int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int H, int S, int Skv, float scale, int kv_group) {
  constexpr int DPL = D / 64;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  const int bhs = blockIdx.x;
  const int split = bhs % S;
  const int bh = bhs / S;
  const int b = bh / H, h = bh % H;
  const int hk = h / kv_group;
  const int skv = (kv_len != nullptr) ? kv_len[b] : Skv;
  const int lo = split * SPLIT_KEYS;
  const int hi = min(lo + SPLIT_KEYS, skv);
  const int64_t pidx = (int64_t)bh * S + split;

  if (lo >= skv) {  // null split: exact no-op under the merge
    if (tid == 0) {
      part_m[pidx] = -3.0e38f;
      part_l[pidx] = 0.f;
    }
    if (tid < 64) {
#pragma unroll
      for (int t = 0; t < DPL; ++t) part_acc[pidx * D + t * 64 + tid] = 0.f;
    }
    return;
  }

  const bf16_t* qp = q + b * q_sb + h * q_sh;
  const bf16_t* kp = k + b * k_sb + hk * k_sh;
  const bf16_t* vp = v + b * v_sb + hk * v_sh;

  u16x8 qreg[D / 8];
#pragma unroll
  for (int c = 0; c < D / 8; ++c) qreg[c] = *(const u16x8*)(qp + c * 8);

  __shared__ float p_lds[4][64];
  __shared__ float merge_m[4], merge_l[4];
  __shared__ float merge_acc[4][D];

  float m_run = -3.0e38f, l_run = 0.f;
  float acc[DPL];
#pragma unroll
  for (int t = 0; t < DPL; ++t) acc[t] = 0.f;

  for (int key0 = lo + wave * 64; key0 < hi; key0 += 256) {
    const int key = key0 + lane;
    float s = -3.0e38f;
    if (key < hi) {
      const bf16_t* kr = kp + (int64_t)key * k_ss;
      float dot = 0.f;
#pragma unroll
      for (int c = 0; c < D / 8; ++c) {
        u16x8 kv8 = *(const u16x8*)(kr + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) dot += bf2f(qreg[c][j]) * bf2f(kv8[j]);
      }
      s = dot * scale;
    }
    const float cmax = wave_reduce_max(s);
    const float m_new = fmaxf(m_run, cmax);
    const float alpha = (m_run <= -3.0e38f) ? 0.f : __expf(m_run - m_new);
    const float p = (s <= -3.0e38f) ? 0.f : __expf(s - m_new);
    m_run = m_new;
    l_run = l_run * alpha + wave_reduce_sum(p);
#pragma unroll
    for (int t = 0; t < DPL; ++t) acc[t] *= alpha;
    p_lds[wave][lane] = p;
    const int limit = min(64, hi - key0);
    for (int j = 0; j < limit; ++j) {
      const float pj = p_lds[wave][j];
      const bf16_t* vr = vp + (int64_t)(key0 + j) * v_ss;
#pragma unroll
      for (int t = 0; t < DPL; ++t)
        acc[t] += pj * bf2f(*(const uint16_t*)(vr + t * 64 + lane));
    }
  }

  if (lane == 0) {
    merge_m[wave] = m_run;
    merge_l[wave] = l_run;
  }
#pragma unroll
  for (int t = 0; t < DPL; ++t) merge_acc[wave][t * 64 + lane] = acc[t];
  __syncthreads();
  if (wave == 0) {
    float M = -3.0e38f;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, merge_m[w]);
    float L = 0.f;
    float out[DPL];
#pragma unroll
    for (int t = 0; t < DPL; ++t) out[t] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float a = (merge_m[w] <= -3.0e38f) ? 0.f : __expf(merge_m[w] - M);
      L += merge_l[w] * a;
#pragma unroll
      for (int t = 0; t < DPL; ++t) out[t] += a * merge_acc[w][t * 64 + lane];
    }
    if (lane == 0) {
      part_m[pidx] = M;
      part_l[pidx] = L;
    }
#pragma unroll
    for (int t = 0; t < DPL; ++t)
      part_acc[pidx * D + t * 64 + lane] = out[t];
  }
}

// one wave per (batch, head): combine the S split states, normalize, write o
template <int D>
__global__ __launch_bounds__(64, 8) void flash_decode_merge_kernel(
    const float* __restrict__ part_m, const float* __restrict__ part_l,
    const float* __restrict__ part_acc, bf16_t* __restrict__ o, int H, int S,
    int64_t o_sb, int64_t o_sh) {
  constexpr int DPL = D / 64;
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  const int lane = threadIdx.x;
  float M = -3.0e38f;
  for (int s = 0; s < S; ++s) M = fmaxf(M, part_m[(int64_t)bh * S + s]);
  float L = 0.f;
  float out[DPL];
#pragma unroll
  for (int t = 0; t < DPL; ++t) out[t] = 0.f;
  for (int s = 0; s < S; ++s) {
    const int64_t pidx = (int64_t)bh * S + s;
    const float m = part_m[pidx];
    const float a = (m <= -3.0e38f) ? 0.f : __expf(m - M);
    L += part_l[pidx] * a;
#pragma unroll
    for (int t = 0; t < DPL; ++t)
      out[t] += a * part_acc[pidx * D + t * 64 + lane];
  }
  const float inv = (L > 0.f) ? 1.0f / L : 0.f;
  bf16_t* op = o + b * o_sb + h * o_sh;
#pragma unroll
  for (int t = 0; t < DPL; ++t)
    *(uint16_t*)(op + t * 64 + lane) = f2bf(out[t] * inv);
}

}  // namespace

extern "C" int flash_decode_num_splits(int B, int H, int Skv) {
  // engage split-KV when B*H underfills the chip AND the cache is long
  // enough to split; boundaries are fixed 512-key multiples (see kernel)
  const int smax = (Skv + SPLIT_KEYS - 1) / SPLIT_KEYS;
  if (smax <= 1 || B * H >= 2048) return 1;
  return smax < 16 ? smax : 16;
}

extern "C" void flash_decode_bf16(const void* q, const void* k, const void* v,
                                  void* o, const int* kv_len, int64_t q_sb,
                                  int64_t q_sh, int64_t k_sb, int64_t k_ss,
                                  int64_t k_sh, int64_t v_sb, int64_t v_ss,
                                  int64_t v_sh, int64_t o_sb, int64_t o_sh, int B,
                                  int H, int Skv, int D, float scale,
                                  int kv_group, hipStream_t stream) {
  dim3 grid(B * H);
  if (D == 64)
    flash_decode_kernel<64><<<grid, dim3(256), 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, kv_len,
        q_sb, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_sh, H, Skv,
        scale, kv_group);
  else if (D == 128)
    flash_decode_kernel<128><<<grid, dim3(256), 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, kv_len,
        q_sb, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, o_sb, o_sh, H, Skv,
        scale, kv_group);
}

extern "C" void flash_decode_split_bf16(
    const void* q, const void* k, const void* v, void* o, float* part_m,
    float* part_l, float* part_acc, const int* kv_len, int64_t q_sb,
    int64_t q_sh, int64_t k_sb, int64_t k_ss, int64_t k_sh, int64_t v_sb,
    int64_t v_ss, int64_t v_sh, int64_t o_sb, int64_t o_sh, int B, int H,
    int S, int Skv, int D, float scale, int kv_group, hipStream_t stream) {
  dim3 grid(B * H * S), mgrid(B * H);
  if (D == 64) {
    flash_decode_split_kernel<64><<<grid, dim3(256), 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, part_m, part_l,
        part_acc, kv_len, q_sb, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, H, S,
        Skv, scale, kv_group);
    flash_decode_merge_kernel<64><<<mgrid, dim3(64), 0, stream>>>(
        part_m, part_l, part_acc, (bf16_t*)o, H, S, o_sb, o_sh);
  } else if (D == 128) {
    flash_decode_split_kernel<128><<<grid, dim3(256), 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, part_m, part_l,
        part_acc, kv_len, q_sb, q_sh, k_sb, k_ss, k_sh, v_sb, v_ss, v_sh, H, S,
        Skv, scale, kv_group);
    flash_decode_merge_kernel<128><<<mgrid, dim3(64), 0, stream>>>(
        part_m, part_l, part_acc, (bf16_t*)o, H, S, o_sb, o_sh);
  }
}
