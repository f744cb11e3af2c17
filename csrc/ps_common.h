// Common device helpers for the MI355X (gfx950 / CDNA4) serving kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; all cross-lane idioms use width-64 shuffles or
//    explicit sub-group widths (16-lane token groups for head_dim=128).
//  - bf16 memory traffic is always vectorized as ushort8 (16 B per lane),
//    the coalescing sweet spot on CDNA4 (Guideline 13).
#pragma once

#include <hip/hip_runtime.h>

#define PS_DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) unsigned short ps_bf16x8;  // 16 B
typedef __attribute__((ext_vector_type(4))) unsigned short ps_bf16x4;  // 8 B
typedef __attribute__((ext_vector_type(4))) float ps_f32x4;

PS_DEV float ps_bf16_to_f32(unsigned short u) {
  union {
    float f;
    unsigned int i;
  } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

PS_DEV unsigned short ps_f32_to_bf16(float f) {
  union {
    float f;
    unsigned int i;
  } v;
  v.f = f;
  unsigned int x = v.i;
  // round-to-nearest-even; NaN flushed to canonical quiet NaN
  if ((x & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;
  unsigned int r = (x >> 16) & 1u;
  x += 0x7fffu + r;
  return (unsigned short)(x >> 16);
}

// XOR-tree reduction across a power-of-two subgroup of the wave.
template <int WIDTH>
PS_DEV float ps_group_sum(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WIDTH);
  return v;
}

template <int WIDTH>
PS_DEV float ps_group_max(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WIDTH));
  return v;
}

#define PS_NEG_INF (-3.0e38f)

static inline int ps_cdiv(int a, int b) { return (a + b - 1) / b; }
