// MFMA chunked-prefill attention for gfx950 (head_dim 128, bf16, paged KV).
//
// Structure (MI355X-first; mfma_f32_16x16x32_bf16, per-wave tiles):
//   grid = (num_q_tiles, num_q_heads), block = 256 (4 waves)
//   Each workgroup owns one (q-tile of <=64 rows, q-head); wave w computes
//   rows [16w, 16w+16). KV is consumed in 64-token chunks (4 paged blocks):
//     QK^T: A = Q[16 x 128] (registers), B = K^T via direct global 16-B lane
//           loads (KV pages are L2-resident across the 4 waves x GQ heads
//           that re-read them -- no LDS staging for K, guide mistake #7).
//     softmax: online, per-row state in C-frag register layout
//           (row = (lane>>4)*4 + reg, col = lane&15  [HW-verified]); one
//           rescale per 64-token chunk.
//     PV:   P routed through a small per-wave LDS tile to convert
//           C-layout -> A-layout; V staged TRANSPOSED in shared LDS with an
//           8-token-group XOR swizzle so B-frags are single conflict-light
//           ds_read_b128s. V staging is software-pipelined (guide T14):
//           next chunk's global loads issue under this chunk's MFMA phase.
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

#define PS_CHUNK 64      // KV tokens per chunk (4 pages)
#define PS_PL_STRIDE 72  // P row stride in tokens (multiple of 8)

// tile_info: int4 per tile = (seq_row, q_token_start, q_pos_start, n_rows)
template <int HEAD_DIM, int WPS, typename KVT>
__global__ __launch_bounds__(256, WPS) void paged_attn_prefill_mfma_kernel(
    unsigned short* __restrict__ out,            // [T, QH, HD]
    const unsigned short* __restrict__ q,        // [T, QH, HD]
    const KVT* __restrict__ k_cache,             // [NB, KH, 16, HD]
    const KVT* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ tile_info,     // [NT, 4]
    int max_blocks, float scale, int KH, int GQ, long q_stride,
    int QH, int n_work, int window) {
  using KVTr = ps_kv_traits<KVT>;
  using kvec8 = typename KVTr::vec8;
  constexpr int D = HEAD_DIM;  // 128
  constexpr int BS = 16;       // page size in tokens
  constexpr int NK = D / 32;   // mfma k-steps over head dim (4)
  constexpr int NCT = PS_CHUNK / 16;  // QK col-tiles per chunk (4)
  constexpr int NKC = PS_CHUNK / 32;  // PV k-chunks per chunk (2)

  // T1 XCD swizzle: works ordered (kvh, gq, tile) and sliced into 8
  // contiguous chunks, one per XCD (dispatch round-robins linear id mod
  // NXCD) -> the GQ=4 heads of one kv-head plus neighboring tiles of the
  // same sequence hit the same XCD's L2 with the same KV pages.
  const int W = n_work;  // num_tiles * QH
  const int cpx = (W + 7) >> 3;
  const int w = (blockIdx.x & 7) * cpx + (blockIdx.x >> 3);
  if (w >= W) return;
  const int n_tiles = W / QH;
  const int tile = w % n_tiles;
  const int gq = (w / n_tiles) % GQ;
  const int kvh = w / (n_tiles * GQ);
  const int qh = kvh * GQ + gq;
  const int seq_row = tile_info[tile * 4 + 0];
  const int q_tok0 = tile_info[tile * 4 + 1];
  const int q_pos0 = tile_info[tile * 4 + 2];
  const int n_rows = tile_info[tile * 4 + 3];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = lane >> 4;   // 0..3
  const int rc = lane & 15;  // row (A/C) or col (B) index

  const int* bt = block_tables + (long)seq_row * max_blocks;
  const int ctx_limit = q_pos0 + n_rows;  // causal bound for the tile
  const int n_pages = (ctx_limit + BS - 1) / BS;
  const int n_chunks = (ctx_limit + PS_CHUNK - 1) / PS_CHUNK;

  // shared: K rows + V^T staging (both XOR-swizzled) + per-wave P tiles.
  // K is staged cooperatively ONCE per workgroup: the 4 waves all consume
  // the same 64-token K tile, and per-wave direct global loads both
  // quadruplicated HBM traffic and exposed load latency (the 128-VGPR
  // budget can't keep 16 loads in flight per wave).
  __shared__ __align__(16) unsigned short k_lds[PS_CHUNK][D];
  __shared__ __align__(16) unsigned short v_t[D][PS_CHUNK];
  __shared__ __align__(16) unsigned short p_lds[4][16][PS_PL_STRIDE];

  // ---- load Q fragments (row rc of this wave's 16) ----
  const int my_local_row = wave * 16 + rc;
  const int q_row_clamped = min(my_local_row, n_rows - 1);
  const unsigned short* qrow =
      q + (long)(q_tok0 + q_row_clamped) * q_stride + (long)qh * D;
  // scale*log2(e) folded into q so the softmax runs in the exp2 domain:
  // saves two v_mul per score element (scale + ln->log2 conversion)
  const float qmul = scale * 1.44269504f;
  ps_mbf16x8 q_frag[NK];
#pragma unroll
  for (int kk = 0; kk < NK; kk++) {
    ps_bf16x8 qv = *(const ps_bf16x8*)(qrow + kk * 32 + g * 8);
    ps_bf16x8 qs;
#pragma unroll
    for (int j = 0; j < 8; j++)
      qs[j] = ps_f32_to_bf16(ps_bf16_to_f32(qv[j]) * qmul);
    q_frag[kk] = ps_as_mbf16(qs);
  }

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
  // sliding window: the wave's largest-window row sets which chunks can
  // contribute at all; per-element masking handles the ragged edge
  const int wave_win_lo =
      window > 0 ? max(0, (q_pos0 + wave * 16) - window + 1) : 0;

  // Staging assignments:
  //   K: thread stages token tv, dims [wave*32, +32) as 4 x b128.
  //   V: thread (octet vo = tv>>3, slice vds = tv&7) stages tokens
  //      [vo*8, +8) x dims [d0 + vds*4, +4): an octet never crosses a
  //      16-token page, and the register-transposed write is 4 x b128
  //      rows of v_t instead of 32 scalar b16 stores (the phase probe
  //      put the scalar staging at 24% of kernel time).
  // Loads for chunk c+1 issue during chunk c's compute phase (T14).
  const int tv = tid & 63;
  const int d0 = wave * 32;
  const int vo = tv >> 3;
  const int vd0 = d0 + (tv & 7) * 4;
  using kvec4 = typename KVTr::vec4;
  auto v_oct_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK + vo * 8;
    const int pg_idx = min(tok / BS, n_pages - 1);
    const long pg = bt[pg_idx];
    return v_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D + vd0;
  };
  auto k_row_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK + tv;
    const int pg_idx = min(tok / BS, n_pages - 1);
    const long pg = bt[pg_idx];
    return k_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D;
  };
  kvec4 vstage[8];
  kvec8 kstage[4];
  {
    const KVT* vrow = v_oct_ptr(0);
    const KVT* krow = k_row_ptr(0);
#pragma unroll
    for (int i = 0; i < 8; i++) vstage[i] = *(const kvec4*)(vrow + i * D);
#pragma unroll
    for (int h = 0; h < 4; h++)
      kstage[h] = *(const kvec8*)(krow + d0 + h * 8);
  }

  for (int chunk = 0; chunk < n_chunks; chunk++) {
    const int tok0 = chunk * PS_CHUNK;
    // write the pre-fetched K rows (16-B slot index XOR-swizzled by the
    // token's low bits: row-major [64][128] would put all 16 lanes of a
    // B-frag read in one bank — guide Guideline 4) and V^T tile
#pragma unroll
    for (int h = 0; h < 4; h++) {
      const int slot = ((wave * 4 + h) ^ (tv & 7));
      *(ps_bf16x8*)(&k_lds[tv][slot * 8]) = KVTr::to_bf16x8(kstage[h]);
    }
#pragma unroll
    for (int dd = 0; dd < 4; dd++) {
      const int d = vd0 + dd;
      ps_bf16x8 row;
#pragma unroll
      for (int i = 0; i < 8; i++) row[i] = KVTr::to_bf16(vstage[i][dd]);
      *(ps_bf16x8*)(&v_t[d][(vo ^ (d & 7)) << 3]) = row;
    }
    __syncthreads();
    if (chunk + 1 < n_chunks) {
      const KVT* vrow = v_oct_ptr(chunk + 1);
      const KVT* krow = k_row_ptr(chunk + 1);
#pragma unroll
      for (int i = 0; i < 8; i++)
        vstage[i] = *(const kvec4*)(vrow + i * D);
#pragma unroll
      for (int h = 0; h < 4; h++)
        kstage[h] = *(const kvec8*)(krow + d0 + h * 8);
    }

    const bool wave_active = (wave * 16 < n_rows) &&
                             (tok0 <= wave_pos_max) &&
                             (tok0 + PS_CHUNK > wave_win_lo);
    if (wave_active) {
      // ---- QK^T over NCT 16-token column tiles ----
      ps_mf32x4 s_frag[NCT];
#pragma unroll
      for (int ct = 0; ct < NCT; ct++) {
        s_frag[ct] = {0.f, 0.f, 0.f, 0.f};
        const int trow = ct * 16 + rc;  // this lane's kv token row (col)
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < NK; kk++) {
          const int slot = ((kk * 4 + g) ^ (trow & 7));
          ps_mbf16x8 k_frag = ps_as_mbf16(
              *(const ps_bf16x8*)(&k_lds[trow][slot * 8]));
          s_frag[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[kk], k_frag, s_frag[ct], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      // ---- mask + online softmax (one rescale per chunk) ----
      // interior chunks (entirely below every row's diagonal, all rows
      // real) skip the per-element mask compare
      bool full_chunk = (tok0 + PS_CHUNK - 1 <= q_pos0 + wave * 16) &&
                        (wave * 16 + 15 < n_rows);
      if (window > 0)
        full_chunk = full_chunk && (tok0 >= wave_pos_max - window + 1);
      float m_new[4];
#pragma unroll
      for (int r = 0; r < 4; r++) m_new[r] = m_run[r];
      if (full_chunk) {
#pragma unroll
        for (int ct = 0; ct < NCT; ct++)
#pragma unroll
          for (int r = 0; r < 4; r++)
            m_new[r] = fmaxf(m_new[r], s_frag[ct][r]);
      } else {
#pragma unroll
        for (int ct = 0; ct < NCT; ct++) {
          const int kv_pos = tok0 + ct * 16 + rc;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const int lrow = wave * 16 + g * 4 + r;
            const int q_pos = q_pos0 + lrow;
            bool valid = (lrow < n_rows) && (kv_pos <= q_pos);
            if (window > 0) valid = valid && (kv_pos > q_pos - window);
            const float sv = valid ? s_frag[ct][r] : PS_NEG_INF;
            s_frag[ct][r] = sv;
            m_new[r] = fmaxf(m_new[r], sv);
          }
        }
      }
#pragma unroll
      for (int r = 0; r < 4; r++) m_new[r] = ps_group_max<16>(m_new[r]);
      // T13 defer-max: if no row's max grew by more than THR, keep the
      // old m (P is then bounded by exp(THR), fine in f32 accum) and skip
      // the whole O rescale pass. Wave-uniform so the branch is free.
      constexpr float PS_RESCALE_THR = 11.54f;  // 8 nats in log2 units
      bool grew = false;
#pragma unroll
      for (int r = 0; r < 4; r++)
        grew |= (m_new[r] > m_run[r] + PS_RESCALE_THR) ||
                (m_run[r] == PS_NEG_INF && m_new[r] > PS_NEG_INF);
      if (__any(grew)) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const float corr = exp2f(m_run[r] - m_new[r]);
          l_run[r] *= corr;
#pragma unroll
          for (int s = 0; s < 8; s++) o_acc[s][r] *= corr;
          m_run[r] = m_new[r];
        }
      }
      float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ct = 0; ct < NCT; ct++) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const float p = s_frag[ct][r] > PS_NEG_INF
                              ? exp2f(s_frag[ct][r] - m_run[r])
                              : 0.f;
          s_frag[ct][r] = p;
          psum[r] += p;
        }
      }
#pragma unroll
      for (int r = 0; r < 4; r++) {
        float rs = ps_group_sum<16>(psum[r]);
        l_run[r] += rs;
      }
      // ---- write P to per-wave LDS (C-layout -> A-layout transpose) ----
#pragma unroll
      for (int ct = 0; ct < NCT; ct++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          p_lds[wave][g * 4 + r][ct * 16 + rc] =
              ps_f32_to_bf16(s_frag[ct][r]);
      // ---- PV: o[16 x 128] += P[16 x 64] @ V[64 x 128] ----
#pragma unroll
      for (int kc = 0; kc < NKC; kc++) {
        // A-frag: P[row rc][tokens kc*32 + g*8 ..+8)
        ps_mbf16x8 p_frag = ps_as_mbf16(
            *(const ps_bf16x8*)(&p_lds[wave][rc][kc * 32 + g * 8]));
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int s = 0; s < 8; s++) {
          const int d = s * 16 + rc;
          const int pg2 = (kc * 4 + g) ^ (d & 7);
          ps_mbf16x8 v_frag =
              ps_as_mbf16(*(const ps_bf16x8*)(&v_t[d][pg2 << 3]));
          o_acc[s] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              p_frag, v_frag, o_acc[s], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
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

int ps_prefill_mfma32_splits(int num_tiles, int KH, int GQ);
int ps_paged_attn_prefill_mfma32(void* out, void* ws_o, void* ws_ml,
                                 long q_tokens, const void* q,
                                 const void* k_cache, const void* v_cache,
                                 const void* block_tables,
                                 const void* tile_info, int num_tiles,
                                 int num_q_heads, int max_blocks,
                                 float scale, int KH, int GQ, int head_dim,
                                 long q_stride, int kv_fp8, int window,
                                 hipStream_t stream);

int ps_paged_attn_prefill_mfma(void* out, const void* q, const void* k_cache,
                               const void* v_cache, const void* block_tables,
                               const void* tile_info, int num_tiles,
                               int num_q_heads, int max_blocks, float scale,
                               int KH, int GQ, int head_dim, long q_stride,
                               int variant, int kv_fp8, int window,
                               hipStream_t stream) {
  if (head_dim != 128) return -1;
  if (variant == 5) {  // 8-wave 32x32 swapped-QK^T kernel; the torch
    // binding passes workspace for split-KV — this raw entry runs S=1
    return ps_paged_attn_prefill_mfma32(
        out, nullptr, nullptr, 0, q, k_cache, v_cache,
        block_tables, tile_info, num_tiles,
        num_q_heads, max_blocks, scale, KH, GQ, head_dim, q_stride,
        kv_fp8, window, stream);
  }
  const int n_work = num_tiles * num_q_heads;
  dim3 grid(((n_work + 7) / 8) * 8);
#define PS_PREFILL_T(WPS, KVT)                                              \
  paged_attn_prefill_mfma_kernel<128, WPS, KVT><<<grid, 256, 0, stream>>>(  \
      (unsigned short*)out, (const unsigned short*)q,                       \
      (const KVT*)k_cache, (const KVT*)v_cache, (const int*)block_tables,   \
      (const int*)tile_info, max_blocks, scale, KH, GQ, q_stride,           \
      num_q_heads, n_work, window)
  if (kv_fp8) {
    if (variant == 3) PS_PREFILL_T(3, unsigned char);
    else PS_PREFILL_T(4, unsigned char);
  } else {
    if (variant == 3) PS_PREFILL_T(3, unsigned short);
    else PS_PREFILL_T(4, unsigned short);
  }
#undef PS_PREFILL_T
  return 0;
}

}  // extern "C"
