// MFMA chunked-prefill attention for gfx950 (head_dim 128, bf16, paged KV).
//
// Structure (MI355X-first; mfma_f32_16x16x32_bf16, per-wave tiles):
//   grid = (num_q_tiles, num_q_heads), block = 256 (4 waves)
//   Each workgroup owns one (q-tile of <=64 rows, q-head); wave w computes
//   rows [16w, 16w+16). KV is consumed in 32-token chunks (2 paged blocks):
//     QK^T: A = Q[16 x 128] (registers), B = K^T via direct global 16-B lane
//           loads (KV pages are L2-resident across the 4 waves x GQ heads
//           that re-read them -- no LDS staging for K, guide mistake #7).
//     softmax: online, per-row state in C-frag register layout
//           (row = (lane>>4)*4 + reg, col = lane&15  [HW-verified]).
//     PV:   P routed through a small per-wave LDS tile ([16][40] stride
//           avoids bank conflicts) to convert C-layout -> A-layout;
//           V staged TRANSPOSED in shared LDS ([128][40]) so B-frags are
//           single ds_read_b128s.
//   Operand k-pattern: lane l supplies elements k = (l>>4)*8 + i for both A
//   and B (k-permutation invariance verified on hardware: csrc/tools/
//   mfma_probe.hip, gpurun_out/mfma_probe.log).
#include "ps_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 ps_mbf16x8;
typedef __attribute__((ext_vector_type(4))) float ps_mf32x4;

PS_DEV ps_mbf16x8 ps_as_mbf16(ps_bf16x8 u) {
  union {
    ps_bf16x8 u16;
    ps_mbf16x8 bf;
  } v;
  v.u16 = u;
  return v.bf;
}

// V^T is stored [128 dims][32 tokens] with the token GROUP (8-token units,
// 16 B) XOR-swizzled by the dim's low bits: physical group = (tok>>3)^(d&3).
// This keeps every access a 16-B-aligned ds_read/write unit while spreading
// banks (2-way worst case on both the staging writes and the B-frag reads).
#define PS_PL_STRIDE 40  // P row stride in tokens

// tile_info: int4 per tile = (seq_row, q_token_start, q_pos_start, n_rows)
template <int HEAD_DIM>
__global__ __launch_bounds__(256) void paged_attn_prefill_mfma_kernel(
    unsigned short* __restrict__ out,            // [T, QH, HD]
    const unsigned short* __restrict__ q,        // [T, QH, HD]
    const unsigned short* __restrict__ k_cache,  // [NB, KH, 16, HD]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ tile_info,     // [NT, 4]
    int max_blocks, float scale, int KH, int GQ, long q_stride) {
  constexpr int D = HEAD_DIM;  // 128
  constexpr int BS = 16;       // page size in tokens
  constexpr int NK = D / 32;   // mfma k-steps over head dim (4)

  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / GQ;
  const int QH = gridDim.y;
  const int seq_row = tile_info[tile * 4 + 0];
  const int q_tok0 = tile_info[tile * 4 + 1];
  const int q_pos0 = tile_info[tile * 4 + 2];
  const int n_rows = tile_info[tile * 4 + 3];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = lane >> 4;    // 0..3
  const int rc = lane & 15;   // row (A/C) or col (B) index

  const int* bt = block_tables + (long)seq_row * max_blocks;
  const int ctx_limit = q_pos0 + n_rows;  // causal bound for the tile
  const int n_pages = (ctx_limit + BS - 1) / BS;
  const int n_chunks = (ctx_limit + 31) / 32;

  // shared: V^T staging + per-wave P tiles
  __shared__ __align__(16) unsigned short v_t[D][32];
  __shared__ __align__(16) unsigned short p_lds[4][16][PS_PL_STRIDE];

  // ---- load Q fragments (row rc of this wave's 16) ----
  const int my_local_row = wave * 16 + rc;
  const int q_row_clamped = min(my_local_row, n_rows - 1);
  const unsigned short* qrow =
      q + (long)(q_tok0 + q_row_clamped) * q_stride + (long)qh * D;
  ps_mbf16x8 q_frag[NK];
#pragma unroll
  for (int kk = 0; kk < NK; kk++)
    q_frag[kk] = ps_as_mbf16(*(const ps_bf16x8*)(qrow + kk * 32 + g * 8));

  // online-softmax state: rows wave*16 + g*4 + r
  float m_run[4], l_run[4];
  ps_mf32x4 o_acc[8];  // 8 dim-slices x 4 rows
#pragma unroll
  for (int r = 0; r < 4; r++) {
    m_run[r] = PS_NEG_INF;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int s = 0; s < 8; s++) o_acc[s] = {0.f, 0.f, 0.f, 0.f};

  const int wave_pos_max = q_pos0 + min(wave * 16 + 15, n_rows - 1);

  // V^T staging is software-pipelined (guide T14 async-STAGE): this
  // thread's V row for chunk c+1 is issued as global loads during chunk c's
  // compute phase and written to LDS after the end-of-compute barrier.
  const int tv = tid & 31;          // staged token (0..31 within chunk)
  const int d0 = (tid >> 5) * 16;   // staged dim range [d0, d0+16)
  const int tg = tv >> 3;
  const int tl = tv & 7;
  auto v_row_ptr = [&](int chunk) {
    const int pg_idx = min((chunk * 32 + tv) / BS, n_pages - 1);
    const long pg = bt[pg_idx];
    return v_cache +
           ((pg * KH + kvh) * BS + ((chunk * 32 + tv) & (BS - 1))) * D;
  };
  ps_bf16x8 vstage[2];
  {
    const unsigned short* vrow = v_row_ptr(0);
    vstage[0] = *(const ps_bf16x8*)(vrow + d0);
    vstage[1] = *(const ps_bf16x8*)(vrow + d0 + 8);
  }

  for (int chunk = 0; chunk < n_chunks; chunk++) {
    const int tok0 = chunk * 32;
    // write the pre-fetched V^T tile for this chunk
#pragma unroll
    for (int h = 0; h < 2; h++) {
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const int d = d0 + h * 8 + j;
        v_t[d][(tg ^ (d & 3)) * 8 + tl] = vstage[h][j];
      }
    }
    __syncthreads();
    // issue next chunk's V loads now; HBM latency hides under the MFMA
    // phase below and the ds_write happens after the trailing barrier
    if (chunk + 1 < n_chunks) {
      const unsigned short* vrow = v_row_ptr(chunk + 1);
      vstage[0] = *(const ps_bf16x8*)(vrow + d0);
      vstage[1] = *(const ps_bf16x8*)(vrow + d0 + 8);
    }

    const bool wave_active = (wave * 16 < n_rows) && (tok0 <= wave_pos_max);
    if (wave_active) {
      // ---- QK^T for the two 16-token column tiles ----
      float p_vals[2][4];  // [col-tile][reg] probability values
      float row_corr[4];
#pragma unroll
      for (int r = 0; r < 4; r++) row_corr[r] = 1.f;
      float m_new[4];
#pragma unroll
      for (int r = 0; r < 4; r++) m_new[r] = m_run[r];

      ps_mf32x4 s_frag[2];
#pragma unroll
      for (int ct = 0; ct < 2; ct++) {
        s_frag[ct] = {0.f, 0.f, 0.f, 0.f};
        const int tok = tok0 + ct * 16 + rc;  // this lane's kv token (col)
        const int pg_idx = min(tok / BS, n_pages - 1);
        const long pg = bt[pg_idx];
        const unsigned short* krow =
            k_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < NK; kk++) {
          ps_mbf16x8 k_frag =
              ps_as_mbf16(*(const ps_bf16x8*)(krow + kk * 32 + g * 8));
          s_frag[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[kk], k_frag, s_frag[ct], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      // ---- mask + online softmax ----
      // lane holds S[row g*4+r][col ct*16+rc]
#pragma unroll
      for (int ct = 0; ct < 2; ct++) {
        const int kv_pos = tok0 + ct * 16 + rc;
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int lrow = wave * 16 + g * 4 + r;
          const int q_pos = q_pos0 + lrow;
          const bool valid = (lrow < n_rows) && (kv_pos <= q_pos);
          const float sv = valid ? s_frag[ct][r] * scale : PS_NEG_INF;
          s_frag[ct][r] = sv;
          m_new[r] = fmaxf(m_new[r], sv);
        }
      }
      // row max across the 16 lanes of this g-group
#pragma unroll
      for (int r = 0; r < 4; r++) m_new[r] = ps_group_max<16>(m_new[r]);
#pragma unroll
      for (int r = 0; r < 4; r++) {
        row_corr[r] = __expf(m_run[r] - m_new[r]);
        l_run[r] *= row_corr[r];
        m_run[r] = m_new[r];
      }
#pragma unroll
      for (int ct = 0; ct < 2; ct++) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int lrow = wave * 16 + g * 4 + r;
          const int kv_pos = tok0 + ct * 16 + rc;
          const int q_pos = q_pos0 + lrow;
          const bool valid = (lrow < n_rows) && (kv_pos <= q_pos);
          const float p = valid ? __expf(s_frag[ct][r] - m_new[r]) : 0.f;
          p_vals[ct][r] = p;
        }
      }
      // row-sum of p across 16 lanes
#pragma unroll
      for (int r = 0; r < 4; r++) {
        float rs = p_vals[0][r] + p_vals[1][r];
        rs = ps_group_sum<16>(rs);
        l_run[r] += rs;
      }
      // rescale O
#pragma unroll
      for (int s = 0; s < 8; s++)
#pragma unroll
        for (int r = 0; r < 4; r++) o_acc[s][r] *= row_corr[r];
      // ---- write P to per-wave LDS (C-layout -> A-layout transpose) ----
#pragma unroll
      for (int ct = 0; ct < 2; ct++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          p_lds[wave][g * 4 + r][ct * 16 + rc] =
              ps_f32_to_bf16(p_vals[ct][r]);
      // ---- PV ----
      // A-frag: P[row rc][tokens g*8 .. g*8+8)
      ps_mbf16x8 p_frag =
          ps_as_mbf16(*(const ps_bf16x8*)(&p_lds[wave][rc][g * 8]));
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 8; s++) {
        // B-frag: V^T[dim s*16 + rc][tokens g*8 .. +8) (swizzled group)
        const int d = s * 16 + rc;
        ps_mbf16x8 v_frag = ps_as_mbf16(
            *(const ps_bf16x8*)(&v_t[d][(g ^ (d & 3)) * 8]));
        o_acc[s] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag, v_frag, o_acc[s], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();  // protect v_t before next chunk's staging
  }

  // ---- epilogue: normalize + store ----
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int lrow = wave * 16 + g * 4 + r;
    if (lrow >= n_rows) continue;
    const float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    unsigned short* orow = out + ((long)(q_tok0 + lrow) * QH + qh) * D;
#pragma unroll
    for (int s = 0; s < 8; s++)
      orow[s * 16 + rc] = ps_f32_to_bf16(o_acc[s][r] * inv);
  }
}

extern "C" {

int ps_paged_attn_prefill_mfma(void* out, const void* q, const void* k_cache,
                               const void* v_cache, const void* block_tables,
                               const void* tile_info, int num_tiles,
                               int num_q_heads, int max_blocks, float scale,
                               int KH, int GQ, int head_dim, long q_stride,
                               hipStream_t stream) {
  if (head_dim != 128) return -1;
  dim3 grid(num_tiles, num_q_heads);
  paged_attn_prefill_mfma_kernel<128><<<grid, 256, 0, stream>>>(
      (unsigned short*)out, (const unsigned short*)q,
      (const unsigned short*)k_cache, (const unsigned short*)v_cache,
      (const int*)block_tables, (const int*)tile_info, max_blocks, scale, KH,
      GQ, q_stride);
  return 0;
}

}  // extern "C"
