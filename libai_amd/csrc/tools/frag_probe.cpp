// Standalone probe: verifies gfx950 MFMA fragment layouts and the
// ds_read_b64_tr_b16 LDS-transpose mapping empirically (guide G9: A=I checks
// with asymmetric operands).  Build on a GPU box:
//   hipcc --offload-arch=gfx950 -O3 frag_probe.cpp -o frag_probe && ./frag_probe
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

#define CHECK(x)                                                          \
  do {                                                                    \
    hipError_t e = (x);                                                   \
    if (e != hipSuccess) {                                                \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);     \
      exit(1);                                                            \
    }                                                                     \
  } while (0)

// Hypothesis under test (mfma_f32_32x32x16_bf16):
//   A[i][k]: lane l holds i = l&31, k = (l>>5)*8 + j   (j = 0..7)
//   B[k][j]: lane l holds j = l&31, k = (l>>5)*8 + j'
//   D[i][j]: lane l holds j = l&31, i = (reg&3) + 8*(reg>>2) + 4*(l>>5)
__global__ void mfma_probe(const __bf16* A, const __bf16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(l & 31) * 16 + (l >> 5) * 8 + j];   // A[i][k] row-major 32x16
    b[j] = B[((l >> 5) * 8 + j) * 32 + (l & 31)]; // B[k][j] row-major 16x32
  }
  f32x16 acc{};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    int i = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    int j = l & 31;
    D[i * 32 + j] = acc[r];
  }
}

// tr_read probe: fill LDS with pattern lds16[n] = n, each lane does one
// ds_read_b64_tr_b16 at per-lane byte address = lane*8, dump the 4 u16 the
// lane received -> reveals the (lane, elem) -> lds index mapping.
__global__ void tr_probe(unsigned short* out, int base_mode) {
  __shared__ __align__(16) unsigned short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = i;
  __syncthreads();
  unsigned idx;  // element (u16) index the lane points at
  if (base_mode == 0) idx = threadIdx.x * 4;            // linear 8B per lane
  else if (base_mode == 1) idx = 0;                     // uniform
  else idx = (threadIdx.x & 15) * 4;                    // repeat per 16-group
  typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)&lds[idx]);
  union { bf16x4 v; unsigned short u[4]; } cvt;
  cvt.v = v;
  for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = cvt.u[j];
}

int main() {
  // ---- MFMA layout check with asymmetric A and B ----
  std::vector<__bf16> A(32 * 16), B(16 * 32);
  std::vector<float> Dref(32 * 32, 0.f), D(32 * 32);
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 16; ++k) A[i * 16 + k] = (__bf16)((i * 7 + k * 3) % 13 - 6);
  for (int k = 0; k < 16; ++k)
    for (int j = 0; j < 32; ++j) B[k * 32 + j] = (__bf16)((k * 5 + j * 11) % 9 - 4);
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float s = 0;
      for (int k = 0; k < 16; ++k) s += (float)A[i * 16 + k] * (float)B[k * 32 + j];
      Dref[i * 32 + j] = s;
    }
  __bf16 *dA, *dB;
  float* dD;
  CHECK(hipMalloc(&dA, A.size() * 2));
  CHECK(hipMalloc(&dB, B.size() * 2));
  CHECK(hipMalloc(&dD, D.size() * 4));
  CHECK(hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice));
  mfma_probe<<<1, 64>>>(dA, dB, dD);
  CHECK(hipDeviceSynchronize());
  CHECK(hipMemcpy(D.data(), dD, D.size() * 4, hipMemcpyDeviceToHost));
  int bad = 0;
  for (int i = 0; i < 32 * 32; ++i)
    if (D[i] != Dref[i]) {
      if (bad < 5) printf("MFMA mismatch at %d: got %f want %f\n", i, D[i], Dref[i]);
      bad++;
    }
  printf("MFMA 32x32x16 layout: %s (%d mismatches)\n", bad ? "WRONG" : "OK", bad);

  // ---- tr_read mapping dump ----
  unsigned short* dout;
  CHECK(hipMalloc(&dout, 64 * 4 * 2));
  std::vector<unsigned short> out(64 * 4);
  for (int mode = 0; mode < 3; ++mode) {
    tr_probe<<<1, 64>>>(dout, mode);
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(out.data(), dout, out.size() * 2, hipMemcpyDeviceToHost));
    printf("tr_read mode %d (lane: e0 e1 e2 e3):\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  l%02d: %4d %4d %4d %4d", l, out[l * 4], out[l * 4 + 1],
             out[l * 4 + 2], out[l * 4 + 3]);
      if (l % 4 == 3) printf("\n");
    }
  }
  return 0;
}
