// Empirical probe for gfx950 __builtin_amdgcn_mfma_f32_32x32x16_bf16
// operand layouts, with asymmetric random A (32x16) and B (16x32) per the
// guide's transpose-detecting methodology (ERRATA #3/#16).
//
// Assumed (to verify):
//   A: lane l supplies A[l&31][k], k = 8*(l>>5) + i  (i = 0..7)
//   B: lane l supplies B[k][l&31], same k-pattern
//   D: lane l, reg r holds D[row][col], col = l&31,
//      row = (r&3) + 8*(r>>2) + 4*(l>>5)              (r = 0..15)
// Also checks k-permutation invariance (any lane k-pattern OK if A and B
// agree), which held for 16x16x32 (csrc/tools/mfma_probe.hip).
// Build: hipcc --offload-arch=gfx950 -O2 mfma32_probe.hip -o mfma32_probe
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ __forceinline__ int kpat(int pat, int hi, int i) {
  switch (pat) {
    case 0: return hi * 8 + i;               // contiguous 8
    case 1: return hi * 4 + (i & 3) + 8 * (i >> 2);  // split halves
    default: return hi + i * 2;              // stride-2 interleave
  }
}

__global__ void probe(const __bf16* A /*32x16*/, const __bf16* B /*16x32*/,
                      float* D /*n_pat x 32x32*/) {
  const int l = threadIdx.x;
  const int hi = l >> 5;
  const int rc = l & 31;
  for (int p = 0; p < 3; p++) {
    bf16x8 a, b;
#pragma unroll
    for (int i = 0; i < 8; i++) {
      a[i] = A[rc * 16 + kpat(p, hi, i)];
      b[i] = B[kpat(p, hi, i) * 32 + rc];
    }
    f32x16 c;
#pragma unroll
    for (int r = 0; r < 16; r++) c[r] = 0.f;
    c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
    float* out = D + p * 1024;
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
      out[row * 32 + rc] = c[r];
    }
  }
}

int main() {
  __bf16 *A, *B;
  float* D;
  hipMallocManaged(&A, 32 * 16 * sizeof(__bf16));
  hipMallocManaged(&B, 16 * 32 * sizeof(__bf16));
  hipMallocManaged(&D, 3 * 1024 * sizeof(float));
  srand(7);
  auto rnd = [] { return (float)(rand() % 17 - 8); };  // exact in bf16
  float Af[32][16], Bf[16][32];
  for (int i = 0; i < 32; i++)
    for (int k = 0; k < 16; k++) {
      Af[i][k] = rnd();
      A[i * 16 + k] = (__bf16)Af[i][k];
    }
  for (int k = 0; k < 16; k++)
    for (int j = 0; j < 32; j++) {
      Bf[k][j] = rnd();
      B[k * 32 + j] = (__bf16)Bf[k][j];
    }
  float ref[32][32];
  for (int i = 0; i < 32; i++)
    for (int j = 0; j < 32; j++) {
      float s = 0;
      for (int k = 0; k < 16; k++) s += Af[i][k] * Bf[k][j];
      ref[i][j] = s;
    }
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, A, B, D);
  hipDeviceSynchronize();
  for (int p = 0; p < 3; p++) {
    int bad = 0;
    for (int i = 0; i < 32 && bad < 5; i++)
      for (int j = 0; j < 32; j++) {
        float got = D[p * 1024 + i * 32 + j];
        if (fabsf(got - ref[i][j]) > 0.5f) {
          if (bad < 3)
            printf("pat%d mismatch D[%d][%d] got %.1f want %.1f\n", p, i, j,
                   got, ref[i][j]);
          bad++;
        }
      }
    printf("k-pattern %d: %s\n", p, bad ? "FAIL" : "PASS");
  }
  return 0;
}
