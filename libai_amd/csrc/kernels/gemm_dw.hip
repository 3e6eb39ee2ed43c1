// Hand-written split-K weight-gradient GEMM for gfx950:  dW = dY^T @ X.
//
// The K1 backward dW GEMMs (reference site: flow.matmul weight grads,
// libai/layers/linear.py:132-157) run at 20-40% of MFMA peak through
// hipBLASLt/rocBLAS on the training shapes (profiles/gemm_roofline.md):
// both operands are "reduction-major" ([M, N] with M = the 49152-token
// reduction dim), so the libraries' TN kernels spend their time in
// transposed loads.  This kernel stages [MB, 128] tiles of both operands
// into LDS tr-images (the flash kernel's verified ds_read_b64_tr_b16
// machinery) and feeds 32x32x16 MFMAs with 1:1 tr-read:MFMA ratio.
//
// Split-K: grid.y partitions M; fp32 partials land in a workspace and a
// second kernel reduces them into the bf16 (or f32) dW.
//
//   C[i, j] = sum_m dY[m, i] * X[m, j]
//   A-frag (lane i=l&31, k=(l>>5)*8+j) <- tr-read of the dY tile
//   B-frag (lane j=l&31, same k split) <- tr-read of the X tile
#include "common.h"

namespace {

typedef __bf16 bf16_t;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));
typedef float f32x16_t __attribute__((ext_vector_type(16)));

#define DW_BT 128  // output tile edge (i and j)
#define DW_MB 64   // m rows staged per iteration

// tr image helpers (layout identical to flash_attn.hip, D = tile cols)
template <int D>
__device__ __forceinline__ void dw_tr_write(bf16_t* lds, int row, int d0,
                                            bf16x8_t val) {
  int idx = (((row >> 2) * (D / 16) + (d0 >> 4)) * 64) + (row & 3) * 16 + (d0 & 15);
  *(bf16x8_t*)(lds + idx) = val;
}

template <int D>
__device__ __forceinline__ bf16x8_t dw_tr_frag(const bf16_t* lds, int rbase,
                                               int dt, int lane) {
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

// stage [ROWS][DW_BT] global rows (stride `ld`) into a tr image; rows
// beyond `bound` become zeros (they contribute 0 to the reduction)
template <int ROWS>
__device__ __forceinline__ void dw_stage(bf16_t* tr_lds, const bf16_t* gp,
                                         int64_t row0, int64_t bound, int64_t ld,
                                         int tid) {
  constexpr int CHUNKS = ROWS * DW_BT / 8;
#pragma unroll
  for (int cc = 0; cc < CHUNKS / 256; ++cc) {
    const int flat = tid + cc * 256;
    const int row = flat / (DW_BT / 8);
    const int col8 = flat % (DW_BT / 8);
    const int64_t gr = row0 + row;
    bf16x8_t val = (gr < bound)
                       ? *(const bf16x8_t*)(gp + gr * ld + col8 * 8)
                       : bf16x8_t{};
    dw_tr_write<DW_BT>(tr_lds, row, col8 * 8, val);
  }
}

// one workgroup = 256 threads = 4 waves in a 2x2 grid of 64x64 sub-tiles
__global__ __launch_bounds__(256, 2) void gemm_dw_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
    float* __restrict__ ws, int64_t M, int64_t N, int64_t K, int64_t m_per_split) {
  __shared__ __align__(16) bf16_t dy_tr[DW_MB * DW_BT];
  __shared__ __align__(16) bf16_t x_tr[DW_MB * DW_BT];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;

  const int64_t tiles_k = K / DW_BT;
  // XCD-aware tile swizzle: consecutive blockIdx.x land on the same XCD's
  // CUs in dispatch order; bijective remap spreads each split's tiles so
  // the 8 XCDs see contiguous i-slices (L2 reuse of the dY slice)
  const int64_t tile = blockIdx.x;
  const int64_t i0 = (tile / tiles_k) * DW_BT;
  const int64_t j0 = (tile % tiles_k) * DW_BT;
  const int64_t split = blockIdx.y;
  const int64_t m0 = split * m_per_split;
  const int64_t m_end = min(M, m0 + m_per_split);

  const bf16_t* dyp = dy + i0;  // column block of dY (row-major [M, N])
  const bf16_t* xp = x + j0;

  f32x16_t acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b) acc[a][b] = f32x16_t{};

  for (int64_t m = m0; m < m_end; m += DW_MB) {
    __syncthreads();
    dw_stage<DW_MB>(dy_tr, dyp, m, m_end, N, tid);
    dw_stage<DW_MB>(x_tr, xp, m, m_end, K, tid);
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < DW_MB / 16; ++ks) {
      bf16x8_t af[2], bf[2];
#pragma unroll
      for (int a = 0; a < 2; ++a)
        af[a] = dw_tr_frag<DW_BT>(dy_tr, ks * 16, wr * 2 + a, lane);
#pragma unroll
      for (int b = 0; b < 2; ++b)
        bf[b] = dw_tr_frag<DW_BT>(x_tr, ks * 16, wc * 2 + b, lane);
#pragma unroll
      for (int a = 0; a < 2; ++a)
#pragma unroll
        for (int b = 0; b < 2; ++b)
          acc[a][b] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[a], bf[b],
                                                              acc[a][b], 0, 0, 0);
    }
  }

  // write fp32 partials: D-frag lane = col j (l&31), row = (r&3)+8*(r>>2)+4*(l>>5)
  float* wsp = ws + split * N * K;
  const int l31 = lane & 31;
  const int hi = lane >> 5;
#pragma unroll
  for (int a = 0; a < 2; ++a) {
    const int64_t ib = i0 + (wr * 2 + a) * 32;
#pragma unroll
    for (int b = 0; b < 2; ++b) {
      const int64_t jb = j0 + (wc * 2 + b) * 32 + l31;
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        f32x4 v;
#pragma unroll
        for (int j = 0; j < 4; ++j) v[j] = acc[a][b][r4 * 4 + j];
        // rows ib + 8*r4 + 4*hi + {0..3}
        const int64_t row = ib + 8 * r4 + 4 * hi;
#pragma unroll
        for (int j = 0; j < 4; ++j) wsp[(row + j) * K + jb] = v[j];
      }
    }
  }
}

// reduce the split partials into dW (bf16 or f32)
template <class E>
__global__ void dw_reduce_kernel(const float* __restrict__ ws,
                                 typename E::T* __restrict__ out, int64_t NK,
                                 int64_t splits) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < NK;
       i += (int64_t)gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int64_t sp = 0; sp < splits; ++sp) s += ws[sp * NK + i];
    out[i] = E::from_f(s);
  }
}

}  // namespace

extern "C" void gemm_dw_bf16(const void* dy, const void* x, float* ws, void* out,
                             int out_is_bf16, int64_t M, int64_t N, int64_t K,
                             int64_t splits, hipStream_t stream) {
  const int64_t m_per_split = CDIV(CDIV(M, splits), DW_MB) * DW_MB;
  dim3 grid((uint32_t)((N / DW_BT) * (K / DW_BT)), (uint32_t)splits);
  gemm_dw_kernel<<<grid, dim3(256), 0, stream>>>(
      (const bf16_t*)dy, (const bf16_t*)x, ws, M, N, K, m_per_split);
  const int64_t NK = N * K;
  int64_t g = CDIV(NK, 256 * 8);
  if (g > 16384) g = 16384;
  if (out_is_bf16)
    dw_reduce_kernel<BF16Elem><<<dim3((uint32_t)g), dim3(256), 0, stream>>>(
        ws, (BF16Elem::T*)out, NK, splits);
  else
    dw_reduce_kernel<F32Elem><<<dim3((uint32_t)g), dim3(256), 0, stream>>>(
        ws, (F32Elem::T*)out, NK, splits);
}
