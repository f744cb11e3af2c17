// Common device helpers for the MI355X (gfx950 / CDNA4) serving kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; all cross-lane idioms use width-64 shuffles or
//    explicit sub-group widths (16-lane token groups for head_dim=128).
//  - bf16 memory traffic is always vectorized as ushort8 (16 B per lane),
//    the coalescing sweet spot on CDNA4 (Guideline 13).
#pragma once

#include <hip/hip_fp8.h>
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

// ---- KV-cache element traits: bf16 (ushort) or OCP fp8 e4m3 (uchar) ----
typedef __attribute__((ext_vector_type(8))) unsigned char ps_fp8x8;  // 8 B

PS_DEV float ps_fp8_to_f32(unsigned char u) {
  __hip_fp8_e4m3 v;
  v.__x = u;
  return (float)v;
}
PS_DEV unsigned char ps_f32_to_fp8(float f) {
  return __hip_fp8_e4m3(f).__x;
}

typedef __attribute__((ext_vector_type(2))) float ps_f32x2;

// Packed fp8->f32: v_cvt_pk_f32_fp8 converts a byte pair per instruction
// (half the VALU of scalar v_cvt_f32_fp8 per element).
PS_DEV void ps_fp8x8_to_f32(ps_fp8x8 v, float* out) {
  union {
    ps_fp8x8 v8;
    int i2[2];
  } u;
  u.v8 = v;
#pragma unroll
  for (int w = 0; w < 2; w++) {
    ps_f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(u.i2[w], false);
    ps_f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(u.i2[w], true);
    out[w * 4 + 0] = lo.x;
    out[w * 4 + 1] = lo.y;
    out[w * 4 + 2] = hi.x;
    out[w * 4 + 3] = hi.y;
  }
}

typedef __attribute__((ext_vector_type(2))) __bf16 ps_rawbf16x2;

// Packed fp8x8 -> bf16x8 (for LDS staging in the MFMA prefill kernel):
// 4x v_cvt_pk_f32_fp8 + 4x v_cvt_pk_bf16_f32 instead of ~8 scalar chains.
PS_DEV ps_bf16x8 ps_fp8x8_to_bf16x8(ps_fp8x8 v) {
  float f[8];
  ps_fp8x8_to_f32(v, f);
  union {
    ps_bf16x8 v8;
    unsigned int i4[4];
  } out;
#pragma unroll
  for (int w = 0; w < 4; w++) {
    ps_f32x2 p = {f[2 * w], f[2 * w + 1]};
    ps_rawbf16x2 b = __builtin_convertvector(p, ps_rawbf16x2);
    out.i4[w] = *(unsigned int*)&b;
  }
  return out.v8;
}

template <typename KVT>
struct ps_kv_traits;

typedef __attribute__((ext_vector_type(4))) unsigned char ps_fp8x4;

template <>
struct ps_kv_traits<unsigned short> {  // bf16 cache
  using vec8 = ps_bf16x8;
  using vec4 = ps_bf16x4;
  static PS_DEV float to_f32(unsigned short u) { return ps_bf16_to_f32(u); }
  static PS_DEV unsigned short from_f32(float f) { return ps_f32_to_bf16(f); }
  static PS_DEV unsigned short to_bf16(unsigned short u) { return u; }
  static PS_DEV void to_f32x8(ps_bf16x8 v, float* out) {
#pragma unroll
    for (int j = 0; j < 8; j++) out[j] = ps_bf16_to_f32(v[j]);
  }
  static PS_DEV ps_bf16x8 to_bf16x8(ps_bf16x8 v) { return v; }
};

template <>
struct ps_kv_traits<unsigned char> {  // fp8 e4m3 cache
  using vec8 = ps_fp8x8;
  using vec4 = ps_fp8x4;
  static PS_DEV float to_f32(unsigned char u) { return ps_fp8_to_f32(u); }
  static PS_DEV unsigned char from_f32(float f) { return ps_f32_to_fp8(f); }
  static PS_DEV unsigned short to_bf16(unsigned char u) {
    return ps_f32_to_bf16(ps_fp8_to_f32(u));
  }
  static PS_DEV void to_f32x8(ps_fp8x8 v, float* out) {
    ps_fp8x8_to_f32(v, out);
  }
  static PS_DEV ps_bf16x8 to_bf16x8(ps_fp8x8 v) {
    return ps_fp8x8_to_bf16x8(v);
  }
};

static inline int ps_cdiv(int a, int b) { return (a + b - 1) / b; }
