// Batched-grouped LoRA matvec (BGMV) for gfx950 — the punica-equivalent
// that lets LoRA requests run inside captured decode hipGraphs.
//
// out[t, off:off+W] += scale[idx[t]] * B[idx[t]] @ (A[idx[t]] @ x[t])
//
// Capability parity: the reference serves adapters through
// /v1/load_lora_adapter (reference loraadapter_controller.go:553-592) and
// relies on vLLM's punica kernels to keep adapter traffic off the critical
// path; here the same role is a single fused per-token kernel: one
// workgroup per token, phase 1 computes y = A x with wave-level reductions
// (rank rows round-robined over the 4 waves), phase 2 expands columns
// B y with the adapter's B panel staying L2-resident across the tokens
// that share it (per-slot panels are <1 MiB << 4 MiB XCD L2).
//
// idx[t] < 0 means "no adapter for this row" and exits immediately, so the
// kernel is safe to leave captured in the decode graph at all times.
#include "ps_common.h"

#define BGMV_RMAX 64

__global__ __launch_bounds__(256) void bgmv_kernel(
    unsigned short* __restrict__ out,      // [T, out_stride] bf16
    const unsigned short* __restrict__ x,  // [T, x_stride] bf16
    const unsigned short* __restrict__ A,  // [S, R, IN] bf16 (layer slice)
    const unsigned short* __restrict__ B,  // [S, W, R] bf16 (layer slice)
    const float* __restrict__ scale,       // [S]
    const int* __restrict__ idx,           // [T] slot per token (-1 = none)
    int IN, int W, int R, long out_stride, long x_stride, int col_off) {
  const int t = blockIdx.x;
  const int s = idx[t];
  if (s < 0) return;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  __shared__ float y[BGMV_RMAX];

  // ---- phase 1: y[r] = dot(A[s, r, :], x[t, :]) ----
  const unsigned short* xrow = x + (long)t * x_stride;
  const unsigned short* Abase = A + ((long)s * R) * IN;
  for (int r = wave; r < R; r += 4) {
    const unsigned short* arow = Abase + (long)r * IN;
    float acc = 0.f;
    for (int i = lane * 8; i < IN; i += 64 * 8) {
      ps_bf16x8 av = *(const ps_bf16x8*)(arow + i);
      ps_bf16x8 xv = *(const ps_bf16x8*)(xrow + i);
#pragma unroll
      for (int j = 0; j < 8; j++)
        acc += ps_bf16_to_f32(av[j]) * ps_bf16_to_f32(xv[j]);
    }
    acc = ps_group_sum<64>(acc);
    if (lane == 0) y[r] = acc * scale[s];
  }
  __syncthreads();

  // ---- phase 2: out[t, col_off + c] += dot(B[s, c, :], y) ----
  const unsigned short* Bbase = B + ((long)s * W) * R;
  unsigned short* orow = out + (long)t * out_stride + col_off;
  for (int c = tid; c < W; c += 256) {
    const unsigned short* brow = Bbase + (long)c * R;
    float acc = 0.f;
    for (int r = 0; r < R; r++)
      acc += ps_bf16_to_f32(brow[r]) * y[r];
    orow[c] = ps_f32_to_bf16(ps_bf16_to_f32(orow[c]) + acc);
  }
}

extern "C" {

int ps_lora_bgmv(void* out, const void* x, const void* A, const void* B,
                 const void* scale, const void* idx, int T, int IN, int W,
                 int R, long out_stride, long x_stride, int col_off,
                 hipStream_t stream) {
  if (R > BGMV_RMAX || T == 0) return -1;
  bgmv_kernel<<<dim3(T), 256, 0, stream>>>(
      (unsigned short*)out, (const unsigned short*)x,
      (const unsigned short*)A, (const unsigned short*)B,
      (const float*)scale, (const int*)idx, IN, W, R, out_stride, x_stride,
      col_off);
  return 0;
}

}  // extern "C"
