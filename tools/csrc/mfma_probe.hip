// MFMA fragment-layout probe for gfx950 mfma_f32_16x16x32_bf16.
//
// Validates the assumed lane->element mappings before any conv/GEMM kernel
// uses them (guide §3: "Always A=I-check with ASYMMETRIC B"):
//   A (16x32): row = lane&15,  k = (lane>>4)*8 + j   (j = 0..7)
//   B (32x16): col = lane&15,  k = (lane>>4)*8 + j
//   C/D (16x16): col = lane&15, row = (lane>>4)*4 + reg
// One wave computes D = A*B and the host compares against a CPU reference.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void probe_kernel(const __bf16* __restrict__ a, const __bf16* __restrict__ b,
                             float* __restrict__ d) {
  int lane = threadIdx.x;  // 64 lanes, one wave
  bf16x8 af, bf;
  int row_a = lane & 15;
  int kb = (lane >> 4) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[row_a * 32 + kb + j];      // A row-major [16][32]
    bf[j] = b[(kb + j) * 16 + (lane & 15)];  // B row-major [32][16], col = lane&15
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = (lane >> 4) * 4 + r;
    int col = lane & 15;
    d[row * 16 + col] = acc[r];
  }
}

int main() {
  const int M = 16, N = 16, K = 32;
  __bf16 *ha = new __bf16[M * K], *hb = new __bf16[K * N];
  float* hd = new float[M * N];
  float* ref = new float[M * N];
  // asymmetric, exactly-representable values
  for (int i = 0; i < M * K; ++i) ha[i] = (__bf16)(float)((i % 7) - 3);
  for (int i = 0; i < K * N; ++i) hb[i] = (__bf16)(float)(((i * 3) % 5) - 2);
  for (int m = 0; m < M; ++m)
    for (int n = 0; n < N; ++n) {
      float s = 0.f;
      for (int k = 0; k < K; ++k) s += (float)ha[m * K + k] * (float)hb[k * N + n];
      ref[m * N + n] = s;
    }
  __bf16 *da, *db;
  float* dd;
  hipMalloc(&da, M * K * sizeof(__bf16));
  hipMalloc(&db, K * N * sizeof(__bf16));
  hipMalloc(&dd, M * N * sizeof(float));
  hipMemcpy(da, ha, M * K * sizeof(__bf16), hipMemcpyHostToDevice);
  hipMemcpy(db, hb, K * N * sizeof(__bf16), hipMemcpyHostToDevice);
  probe_kernel<<<1, 64>>>(da, db, dd);
  hipError_t err = hipDeviceSynchronize();
  if (err != hipSuccess) {
    printf("PROBE LAUNCH FAILED: %s\n", hipGetErrorString(err));
    return 2;
  }
  hipMemcpy(hd, dd, M * N * sizeof(float), hipMemcpyDeviceToHost);
  float maxdiff = 0.f;
  for (int i = 0; i < M * N; ++i) maxdiff = fmaxf(maxdiff, fabsf(hd[i] - ref[i]));
  printf("mfma_f32_16x16x32_bf16 layout probe: maxdiff=%g -> %s\n", maxdiff,
         maxdiff < 1e-3 ? "LAYOUT OK" : "LAYOUT WRONG");
  if (maxdiff >= 1e-3) {
    for (int r = 0; r < 4; ++r) {
      printf("row %d got:", r);
      for (int c = 0; c < 8; ++c) printf(" %6.1f", hd[r * 16 + c]);
      printf("  ref:");
      for (int c = 0; c < 8; ++c) printf(" %6.1f", ref[r * 16 + c]);
      printf("\n");
    }
  }
  return maxdiff < 1e-3 ? 0 : 1;
}
