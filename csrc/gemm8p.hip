// 8-phase deep-pipelined bf16 GEMM for gfx950: out[M,N] = x[M,K] @ W[N,K]^T.
//
// Structure per the CDNA4 guide's 256^2 8-phase template (§5 "The 256² 8-phase
// template", techniques T1-T5): 256x256 output tile, BK=64, 8 waves (2M x 4N,
// 512 threads), LDS = 2 double-buffered K-tiles x (A,B) x 256x64 bf16 =
// 128 KiB. Each K-loop iteration covers TWO K-tiles in 8 phases; every phase
// {12x ds_read_b128 -> issue ONE half-tile (128 rows x 64 k) of a future
// K-tile via global_load_lds(16B) -> barrier -> lgkmcnt(0) -> setprio(1) ->
// 16 MFMA (one C-quadrant x K=64) -> setprio(0) -> barrier}. Counted
// s_waitcnt vmcnt(2) at the starts of phases 0 and 4 keep staged loads in
// flight across barriers — never drained to 0 in the main loop (guide
// T3+T4: the whole gain of the 8-phase schedule).
//
// Staging schedule (phase -> half-tile), derived so every stage targets a
// slot whose last ds_read finished a full barrier earlier (global-load->LDS
// writes are unordered vs the LDS pipe) and every stage is FIFO-guaranteed
// landed by a counted vmcnt(2) before its first read:
//   p0: B(t+1)h0  p1: B(t+1)h1  p2: A(t+1)h1  p3: A(t+2)h0
//   p4: A(t+2)h1  p5: B(t+2)h0  p6: B(t+2)h1  p7: A(t+3)h0
// where iteration i computes K-tiles t=2i (buf0, phases 0-3) and t+1
// (buf1, phases 4-7); quadrant order (m-half, n-half) = 00,01,10,11.
//
// LDS bank conflicts: rows are 128 B, so 16 lanes reading a column slice
// alias one bank 16-way; an XOR swizzle byte ^= ((row&7)<<4) (guide
// Guideline 4, +89% on the attention kernel) spreads each 8-row stripe over
// 8 distinct 16-B slots. global_load_lds writes LINEARLY (wave-uniform base
// + lane*16), so the swizzle is applied by PRE-SWIZZLING the per-lane global
// source address (guide ERRATA #21 "linear dest + inverse-swz source + swz
// on read"); the involution makes source and read permutations identical.
//
// Split-K for skinny-M shapes (decode batches): gridDim.y K-ranges write
// fp32 partials to ws[S,M,N]; gemm8p_combine sums them to bf16. K must be a
// multiple of 128 per split range (all Llama projections qualify).
//
// mfma_f32_16x16x32_bf16; A/B k-pattern lane=(l>>4)*8+i verified in
// csrc/tools/mfma_probe.hip; C/D layout col=lane&15, row=(l>>4)*4+reg.
#include "ps_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 ps_g8bf16x8;
typedef __attribute__((ext_vector_type(4))) float ps_g8f32x4;

PS_DEV ps_g8bf16x8 ps_as_g8bf16(ps_bf16x8 u) {
  union {
    ps_bf16x8 u16;
    ps_g8bf16x8 bf;
  } v;
  v.u16 = u;
  return v.bf;
}

// byte-offset swizzle within a 16 KiB half-tile (involution, 16B-granular).
// Two candidates, A/B-selectable at compile time: the attention-style
// 8-row stripe spread (default) and the guide's st_16x32 (XOR byte bit 5
// with bit 9 within each 1 KiB subtile, m201). Measured on MI355X at the
// Llama serving shapes: attn-style wins every cell (e.g. gate_up M=2048
// 958 vs 766 TF) — the frag reads here stride 128 B, not 256, so the
// 8-row stripe spread matches the bank geometry better.
#ifndef PS_G8_SWZ_VARIANT
#define PS_G8_SWZ_VARIANT 0
#endif
PS_DEV int ps_g8_swz(int b) {
#if PS_G8_SWZ_VARIANT == 1
  return b ^ (((b >> 9) & 1) << 5);
#else
  return b ^ (((b >> 7) & 7) << 4);
#endif
}

// LDS byte offset of a generic pointer to a __shared__ object
PS_DEV unsigned ps_g8_lds_off(const void* p) {
  return (unsigned)(unsigned long long)(
      const __attribute__((address_space(3))) char*)p;
}

// ds_read_b128 issued as opaque inline asm: the AMDGPU memory legalizer
// otherwise inserts s_waitcnt vmcnt(0) before every phase's LDS reads to
// order them against the in-flight global_load_lds writes — draining the
// pipeline each phase and reducing the schedule to 1-phase performance
// (guide T3/T4 and common-mistake notes). Correct ordering is owned by the
// phase barriers + the two counted vmcnt(4) waits.
PS_DEV ps_g8bf16x8 ps_g8_ds_read_b128(unsigned addr) {
  ps_g8bf16x8 r;
  asm volatile("ds_read_b128 %0, %1" : "=v"(r) : "v"(addr));
  return r;
}

#define PS_G8_BM 256
#define PS_G8_BN 256
#define PS_G8_BK 64

__global__ __launch_bounds__(512, 2) void gemm8p_kernel(
    unsigned short* __restrict__ out_bf16,  // [M, N] (splits == 1)
    float* __restrict__ ws,                 // [S, M, N] (splits > 1)
    const unsigned short* __restrict__ x,   // [M, K] (row stride x_stride)
    const unsigned short* __restrict__ w,   // [N, K]
    int M, int N, int K, long x_stride) {
  // ---- block -> (m0, n0) with bijective XCD swizzle (guide T1/#11) ----
  const int mt = (M + PS_G8_BM - 1) / PS_G8_BM;
  const int nt = N / PS_G8_BN;
  const int nwg = mt * nt;
  int wg = blockIdx.x;
  {
    const int q = nwg >> 3, r = nwg & 7, xcd = wg & 7, lin = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lin;
  }
  if (wg >= nwg) return;
  const int m0 = (wg % mt) * PS_G8_BM;
  const int n0 = (wg / mt) * PS_G8_BN;

  const int splits = gridDim.y;
  const int kpairs = K / 128;
  const int per_split = (kpairs + splits - 1) / splits;
  const int kt_begin = blockIdx.y * per_split * 2;        // in 64-wide tiles
  const int kt_end = min(K / 64, kt_begin + per_split * 2);
  if (kt_begin >= kt_end) return;
  const int n_iter = (kt_end - kt_begin) / 2;  // iterations of 2 K-tiles

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = lane >> 4;   // k-group
  const int rc = lane & 15;  // row (A/C) / col (B) within a 16-tile
  const int wm = wave >> 2;  // wave M position (2)
  const int wn = wave & 3;   // wave N position (4)

  // LDS: [dbuf][half][128 rows][64 k] per operand
  __shared__ __align__(16) unsigned short a_lds[2][2][128][64];
  __shared__ __align__(16) unsigned short b_lds[2][2][128][64];

  // ---- staging: one half-tile = 16 KiB = 8 waves x 2 calls x 1 KiB ----
  // wave v, call j covers segments (v*2+j)*64 + lane; dest is linear,
  // source address pre-swizzled.
  const int seg_base0 = wave * 2;  // call j adds j
  auto stage_half = [&](unsigned short (*dst)[64],  // [128][64] half
                        const unsigned short* src_base, long src_stride,
                        int row_limit, int row0_g, int k0_g) {
#pragma unroll
    for (int j = 0; j < 2; j++) {
      const int seg = (seg_base0 + j) * 64 + lane;
      const int b = ps_g8_swz(seg * 16);
      const int row_l = b >> 7;
      const int k_l = (b & 127) >> 1;
      const unsigned short* src =
          src_base + (long)min(row0_g + row_l, row_limit) * src_stride +
          k0_g + k_l;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)(
              &dst[0][0] + (seg_base0 + j) * 512),
          16, 0, 0);
    }
  };
  // stage dispatcher: half-tiles indexed 0..7 within an iteration pair
  // (A t+2 h0, A t+2 h1, B t+2 h0, B t+2 h1, A t+3 h0, A t+3 h1,
  //  B t+3 h0, B t+3 h1); tile kt is absolute (64-wide index).
  auto stage = [&](int which, int kt) {
    if (kt >= kt_end) return;  // epilogue: nothing left to stage
    const int buf = kt & 1;
    const int h = which & 1;
    const bool is_a = (which & 2) == 0;
    if (is_a)
      stage_half(a_lds[buf][h], x, x_stride, M - 1, m0 + h * 128, kt * 64);
    else
      stage_half(b_lds[buf][h], w, K, N - 1, n0 + h * 128, kt * 64);
  };

  // ---- accumulators: wave's 128x64 C block = 8 m-frags x 4 n-frags ----
  ps_g8f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; i++)
#pragma unroll
    for (int j = 0; j < 4; j++) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // ---- prologue: emulate the steady-state "previous iteration" stage
  // FIFO [A(t0)h0, A(t0)h1, B(t0)h0, B(t0)h1, A(t1)h0] so the loop's
  // counted waits see the same outstanding-load order from iteration 0.
  stage(0, kt_begin);      // A t0 h0
  stage(1, kt_begin);      // A t0 h1
  stage(2, kt_begin);      // B t0 h0
  stage(3, kt_begin);      // B t0 h1
  stage(0, kt_begin + 1);  // A t1 h0

  // one phase: quadrant (mh, nh) of K-tile in buf, then stage `s_which`
  // of tile `s_kt`. ds_reads happen FIRST (so the slot being overwritten
  // by this phase's stage has no readers left), stage issue second.
  const unsigned a_lds0 = ps_g8_lds_off(&a_lds[0][0][0][0]);
  const unsigned b_lds0 = ps_g8_lds_off(&b_lds[0][0][0][0]);

  auto phase = [&](int buf, int mh, int nh, int s_which, int s_kt) {
    // A rows: half mh, rows mt*16 + rc (mt 0..3); B: wave's n-half
    const int bh = wn >> 1;  // B half this wave reads
    const int brow0 = (wn & 1) * 64 + nh * 32;
    const unsigned a_base = a_lds0 + (buf * 2 + wm) * 16384;
    const unsigned b_base = b_lds0 + (buf * 2 + bh) * 16384;
    ps_g8bf16x8 af[2][4], bf[2][2];
#pragma unroll
    for (int ks = 0; ks < 2; ks++) {
#pragma unroll
      for (int mtl = 0; mtl < 4; mtl++) {
        const int row = mh * 64 + mtl * 16 + rc;
        const int byte = ps_g8_swz(row * 128 + (ks * 32 + g * 8) * 2);
        af[ks][mtl] = ps_g8_ds_read_b128(a_base + byte);
      }
#pragma unroll
      for (int ntl = 0; ntl < 2; ntl++) {
        const int row = brow0 + ntl * 16 + rc;
        const int byte = ps_g8_swz(row * 128 + (ks * 32 + g * 8) * 2);
        bf[ks][ntl] = ps_g8_ds_read_b128(b_base + byte);
      }
    }
    stage(s_which, s_kt);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);  // guide rule #18: pin MFMAs after
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ks++)
#pragma unroll
      for (int mtl = 0; mtl < 4; mtl++)
#pragma unroll
        for (int ntl = 0; ntl < 2; ntl++)
          acc[mh * 4 + mtl][nh * 2 + ntl] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[ks][mtl], bf[ks][ntl],
                  acc[mh * 4 + mtl][nh * 2 + ntl], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

  for (int it = 0; it < n_iter; it++) {
    const int t = kt_begin + 2 * it;      // first K-tile of the pair
    const int b0 = t & 1, b1 = b0 ^ 1;    // buf of t / t+1
    // Stage schedule: a stage at phase p may only target a slot whose
    // LAST ds_read was in phase <= p-1 — the closing barrier of p-1
    // comes after every wave's lgkmcnt(0) drain, so no wave still has a
    // read of that slot in the LDS pipe when the staged write can land
    // (global-load->LDS writes are NOT ordered against ds_reads).
    //   p0: B(t+1)h0  p1: B(t+1)h1  p2: A(t+1)h1  p3: A(t+2)h0
    //   p4: A(t+2)h1  p5: B(t+2)h0  p6: B(t+2)h1  p7: A(t+3)h0
    // Counted waits (guide T4): vmcnt(2) at the two half-iteration
    // starts; FIFO completion guarantees the 4 half-tiles the next 4
    // phases read have landed while 1 half-tile stays in flight.
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase(b0, 0, 0, /*B(t+1)h0*/ 2, t + 1);
    phase(b0, 0, 1, /*B(t+1)h1*/ 3, t + 1);
    phase(b0, 1, 0, /*A(t+1)h1*/ 1, t + 1);
    phase(b0, 1, 1, /*A(t+2)h0*/ 0, t + 2);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase(b1, 0, 0, /*A(t+2)h1*/ 1, t + 2);
    phase(b1, 0, 1, /*B(t+2)h0*/ 2, t + 2);
    phase(b1, 1, 0, /*B(t+2)h1*/ 3, t + 2);
    phase(b1, 1, 1, /*A(t+3)h0*/ 0, t + 3);
  }

  // Drain every outstanding staged load before exit: an LDS write landing
  // after s_endpgm would hit a workgroup slot the hardware may already
  // have reassigned.
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // ---- epilogue: store the wave's 128x64 block ----
  const int row_base = m0 + wm * 128;
  const int col_base = n0 + wn * 64;
  if (ws != nullptr) {
    float* dst = ws + (long)blockIdx.y * M * N;
#pragma unroll
    for (int mtl = 0; mtl < 8; mtl++)
#pragma unroll
      for (int ntl = 0; ntl < 4; ntl++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int row = row_base + mtl * 16 + g * 4 + r;
          if (row < M)
            dst[(long)row * N + col_base + ntl * 16 + rc] =
                acc[mtl][ntl][r];
        }
  } else {
#pragma unroll
    for (int mtl = 0; mtl < 8; mtl++)
#pragma unroll
      for (int ntl = 0; ntl < 4; ntl++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int row = row_base + mtl * 16 + g * 4 + r;
          if (row < M)
            out_bf16[(long)row * N + col_base + ntl * 16 + rc] =
                ps_f32_to_bf16(acc[mtl][ntl][r]);
        }
  }
}

__global__ void gemm8p_combine_kernel(unsigned short* __restrict__ out,
                                      const float* __restrict__ ws, long MN,
                                      int splits) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  float acc = 0.f;
  for (int s = 0; s < splits; s++) acc += ws[(long)s * MN + i];
  out[i] = ps_f32_to_bf16(acc);
}

extern "C" {

// Returns split count the heuristic picks (for ws sizing), or -1 if the
// shape is unsupported (N % 256, K % 128).
int ps_gemm8p_splits(int M, int N, int K) {
  if (N % 256 != 0 || K % 128 != 0) return -1;
  const int mt = (M + 255) / 256;
  const int nwg = mt * (N / 256);
  int splits = 384 / nwg;
  if (splits < 1) splits = 1;
  const int kpairs = K / 128;
  if (splits > kpairs) splits = kpairs;
  return splits;
}

int ps_gemm8p(void* out_bf16, void* ws, const void* x, const void* w, int M,
              int N, int K, long x_stride, hipStream_t stream) {
  int splits = ps_gemm8p_splits(M, N, K);
  if (splits < 0) return -1;
  if (ws == nullptr) splits = 1;
  const int mt = (M + 255) / 256;
  dim3 grid(((mt * (N / 256) + 7) / 8) * 8, splits);
  gemm8p_kernel<<<grid, 512, 0, stream>>>(
      (unsigned short*)out_bf16, splits > 1 ? (float*)ws : nullptr,
      (const unsigned short*)x, (const unsigned short*)w, M, N, K, x_stride);
  if (splits > 1) {
    const long MN = (long)M * N;
    gemm8p_combine_kernel<<<dim3((MN + 255) / 256), 256, 0, stream>>>(
        (unsigned short*)out_bf16, (const float*)ws, MN, splits);
  }
  return 0;
}

}  // extern "C"
