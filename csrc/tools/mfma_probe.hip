// Empirical probe: determine the A/B operand lane->element mappings of
// gfx950 __builtin_amdgcn_mfma_f32_16x16x32_bf16, with asymmetric random
// A (16x32) and B (32x16) per the guide's transpose-detecting methodology.
// The C/D mapping (col=lane&15, row=(lane>>4)*4+reg) is HW-verified
// [learn_hip m89/m91]; this probe confirms it transitively.
//
// Candidate k-patterns for lane l, element i (g = l>>4):
//   P0: k = g*8 + i          (contiguous 8 per lane)
//   P1: k = g*4 + (i&3) + 16*(i>>2)   (two 16-k halves of contiguous 4)
//   P2: k = g + i*4          (stride-4 interleave)
// A: element = A[l&15][k];  B: element = B[k][l&15].
// Build: hipcc --offload-arch=gfx950 -O2 mfma_probe.hip -o mfma_probe
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ int kpat(int pat, int g, int i) {
  switch (pat) {
    case 0: return g * 8 + i;
    case 1: return g * 4 + (i & 3) + 16 * (i >> 2);
    default: return g + i * 4;
  }
}

__global__ void probe(const __bf16* A /*16x32*/, const __bf16* B /*32x16*/,
                      float* D /*9 x 16x16*/) {
  const int l = threadIdx.x;
  const int g = l >> 4;
  const int rc = l & 15;
  for (int pa = 0; pa < 3; pa++) {
    for (int pb = 0; pb < 3; pb++) {
      bf16x8 a, b;
#pragma unroll
      for (int i = 0; i < 8; i++) {
        a[i] = A[rc * 32 + kpat(pa, g, i)];
        b[i] = B[kpat(pb, g, i) * 16 + rc];
      }
      f32x4 c = {0.f, 0.f, 0.f, 0.f};
      c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
      float* out = D + (pa * 3 + pb) * 256;
#pragma unroll
      for (int r = 0; r < 4; r++) out[(g * 4 + r) * 16 + rc] = c[r];
    }
  }
}

int main() {
  __bf16 hA[16 * 32], hB[32 * 16];
  float refD[16 * 16];
  srand(7);
  float fA[16 * 32], fB[32 * 16];
  for (int i = 0; i < 16 * 32; i++) {
    fA[i] = (rand() % 1000 - 500) / 250.0f;
    hA[i] = (__bf16)fA[i];
    fA[i] = (float)hA[i];
  }
  for (int i = 0; i < 32 * 16; i++) {
    fB[i] = (rand() % 1000 - 500) / 250.0f;
    hB[i] = (__bf16)fB[i];
    fB[i] = (float)hB[i];
  }
  for (int m = 0; m < 16; m++)
    for (int n = 0; n < 16; n++) {
      float s = 0;
      for (int k = 0; k < 32; k++) s += fA[m * 32 + k] * fB[k * 16 + n];
      refD[m * 16 + n] = s;
    }
  __bf16 *dA, *dB;
  float* dD;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dD, 9 * 256 * sizeof(float));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dA, dB, dD);
  float out[9 * 256];
  hipMemcpy(out, dD, sizeof(out), hipMemcpyDeviceToHost);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    printf("HIP error: %s\n", hipGetErrorString(e));
    return 1;
  }
  for (int pa = 0; pa < 3; pa++)
    for (int pb = 0; pb < 3; pb++) {
      float maxerr = 0, maxerr_t = 0;
      for (int m = 0; m < 16; m++)
        for (int n = 0; n < 16; n++) {
          float got = out[(pa * 3 + pb) * 256 + m * 16 + n];
          float err = fabsf(got - refD[m * 16 + n]);
          float err_t = fabsf(got - refD[n * 16 + m]);
          if (err > maxerr) maxerr = err;
          if (err_t > maxerr_t) maxerr_t = err_t;
        }
      printf("A=P%d B=P%d maxerr=%.4f maxerr_transposed=%.4f %s\n", pa, pb,
             maxerr, maxerr_t,
             maxerr < 0.15   ? "<== MATCH"
             : maxerr_t < 0.15 ? "<== MATCH-TRANSPOSED"
                               : "");
    }
  return 0;
}
