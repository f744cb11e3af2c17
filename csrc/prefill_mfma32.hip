// MFMA chunked-prefill attention v5 for gfx950: 8-wave 32x32 swapped-QK^T
// structure (head_dim 128, bf16/fp8 paged KV).
//
// Design follows the CDNA4 guide's fused-attention ladder
// (/opt/skills/guides/cdna_hip_programming.md §B "8-warp 32×32 ladder"):
//   grid = (num_q_tiles, num_q_heads), block = 512 (8 waves), tile = 256
//   q-rows; wave w owns rows [32w, 32w+32). KV consumed in 64-token chunks.
//   * swapped QK^T: S^T[kv,q] = mfma_32x32x16(K_frag, Q_frag) — each lane's
//     accumulator column is ONE q row (col=lane&31), so the online-softmax
//     row max/sum are lane-local scalars + one shfl_xor(·,32) across the
//     half-wave split (guide T12 precondition), replacing the 16-lane
//     group reduces + LDS P-transpose of the 16x16 kernel (v3/v4).
//   * K staged once per workgroup in XOR-swizzled LDS (guide T2);
//     V staged register-transposed to v_t[d][tok] with octet swizzle.
//   * register prefetch of chunk c+1's K/V under chunk c's MFMAs (T14).
//   * defer-max rescale threshold (T13), exp2-domain softmax.
//   * P routed lane->LDS->A-frag per wave (32x64 bf16, swizzled slots).
//   * 32x32x16 operand layouts HW-verified by csrc/tools/mfma32_probe.hip:
//     A: lane l = A[l&31][8*(l>>5)+i]; B: lane l = B[8*(l>>5)+i][l&31];
//     D: lane l reg r = D[(r&3)+8*(r>>2)+4*(l>>5)][l&31].
#include "ps_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 ps_mbf16x8;
typedef __attribute__((ext_vector_type(16))) float ps_mf32x16;

PS_DEV ps_mbf16x8 ps_as_mbf16_32(ps_bf16x8 u) {
  union {
    ps_bf16x8 u16;
    ps_mbf16x8 bf;
  } v;
  v.u16 = u;
  return v.bf;
}

#define PS_CHUNK32 64    // KV tokens per chunk (4 pages)
#define PS_TILE32 256    // q rows per workgroup (8 waves x 32)

// tile_info: int4 per tile = (seq_row, q_token_start, q_pos_start, n_rows)
// GQW = GQA q-heads processed per workgroup: the 8 waves split into
// GQW head-columns x (8/GQW) row-halves, so all GQW heads of a kv-head
// share ONE K/V staging pass (serving-shape prefill chunks are
// staging-bound: a 200-row continuation re-reads a 2000+-token context).
// Tile rows = 32 * (8/GQW).
// Split-KV (gridDim.y > 1): split s processes kv chunks s, s+S, s+2S...
// and writes UNNORMALIZED partials (o_acc, m, l) to the workspace; the
// combine kernel merges them. Strided chunk assignment balances the
// causal tail across splits. Raises the workgroup count for small
// serving launches (1-block/CU quantization measured ~25% idle at the
// 10-seq continuation shape).
template <int HEAD_DIM, int GQW, typename KVT>
__global__ __launch_bounds__(512, 2) void paged_attn_prefill_mfma32_kernel(
    unsigned short* __restrict__ out,            // [T, QH, HD]
    float* __restrict__ ws_o,    // [S, T, QH, HD] (splits > 1)
    float* __restrict__ ws_ml,   // [2, S, T, QH]
    long q_tokens,
    const unsigned short* __restrict__ q,        // [T, QH, HD]
    const KVT* __restrict__ k_cache,             // [NB, KH, 16, HD]
    const KVT* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ tile_info,     // [NT, 4]
    int max_blocks, float scale, int KH, int GQ, long q_stride,
    int QH, int n_work, int window) {
  const int split = blockIdx.y;
  const int n_splits = gridDim.y;
  using KVTr = ps_kv_traits<KVT>;
  using kvec8 = typename KVTr::vec8;
  using kvec4 = typename KVTr::vec4;
  constexpr int D = HEAD_DIM;   // 128
  constexpr int BS = 16;        // page size in tokens
  constexpr int NKS = D / 16;   // QK k-steps over head dim (8)
  constexpr int ROWH = 8 / GQW; // row-halves (32 rows each) per tile

  // T1 XCD swizzle (same scheme as v3/v4): contiguous work chunks per XCD
  const int W = n_work;  // n_tiles * KH * (GQ/GQW)
  const int cpx = (W + 7) >> 3;
  const int w = (blockIdx.x & 7) * cpx + (blockIdx.x >> 3);
  if (w >= W) return;
  const int n_grp = GQ / GQW;
  const int n_tiles = W / (n_grp * KH);
  const int tile = w % n_tiles;
  const int g_grp = (w / n_tiles) % n_grp;
  const int kvh = w / (n_tiles * n_grp);
  const int seq_row = tile_info[tile * 4 + 0];
  const int q_tok0 = tile_info[tile * 4 + 1];
  const int q_pos0 = tile_info[tile * 4 + 2];
  const int n_rows = tile_info[tile * 4 + 3];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;   // half-wave
  const int rc = lane & 31;   // q row (B/D col) or kv row (A row)
  // wave -> (head column, row half)
  const int qh = kvh * GQ + g_grp * GQW + (wave / ROWH);
  const int wq0 = (wave % ROWH) * 32;  // wave's first local q row

  const int* bt = block_tables + (long)seq_row * max_blocks;
  const int ctx_limit = q_pos0 + n_rows;
  const int n_pages = (ctx_limit + BS - 1) / BS;
  const int n_chunks = (ctx_limit + PS_CHUNK32 - 1) / PS_CHUNK32;

  __shared__ __align__(16) unsigned short k_lds[PS_CHUNK32][D];
  __shared__ __align__(16) unsigned short v_t[D][PS_CHUNK32];
  __shared__ __align__(16) unsigned short p_lds[8][32][64];

  // ---- Q fragments: B-operand, lane supplies Q[q=rc][d=ks*16+hi*8+i] ----
  const int q_row_clamped = min(wq0 + rc, n_rows - 1);
  const unsigned short* qrow =
      q + (long)(q_tok0 + q_row_clamped) * q_stride + (long)qh * D;
  const float qmul = scale * 1.44269504f;  // exp2-domain softmax
  ps_mbf16x8 q_frag[NKS];
#pragma unroll
  for (int ks = 0; ks < NKS; ks++) {
    ps_bf16x8 qv = *(const ps_bf16x8*)(qrow + ks * 16 + hi * 8);
    ps_bf16x8 qs;
#pragma unroll
    for (int j = 0; j < 8; j++)
      qs[j] = ps_f32_to_bf16(ps_bf16_to_f32(qv[j]) * qmul);
    q_frag[ks] = ps_as_mbf16_32(qs);
  }

  // online-softmax state for q row rc: lane-local scalars
  float m_run = PS_NEG_INF, l_run = 0.f;
  ps_mf32x16 o_acc[4];  // 4 d-tiles; lane holds O[roff(r,hi)][dt*32+rc]
#pragma unroll
  for (int dt = 0; dt < 4; dt++)
#pragma unroll
    for (int r = 16; r-- > 0;) o_acc[dt][r] = 0.f;

  const int q_pos = q_pos0 + wq0 + rc;  // this lane's q position
  const int wave_pos_max = q_pos0 + min(wq0 + 31, n_rows - 1);
  const int wave_win_lo =
      window > 0 ? max(0, (q_pos0 + wq0) - window + 1) : 0;

  // ---- staging assignments (512 threads) ----
  // K: 64 tokens x 16 slots of 8 dims = 1024 16-B segments -> each thread
  //    stages TWO segments (slots t&7 and (t&7)+8) of token t>>3.
  // V: thread t stages 4 tokens x 4 dims register-transposed:
  //    token quad tq = t&15, dim slice ds = t>>4.
  const int ktok = tid >> 3, kslot = tid & 7;
  const int vtq = tid & 15, vds = tid >> 4;
  auto k_seg_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK32 + ktok;
    const long pg = bt[min(tok / BS, n_pages - 1)];
    return k_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D +
           kslot * 8;
  };
  auto v_quad_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK32 + vtq * 4;
    const long pg = bt[min(tok / BS, n_pages - 1)];
    return v_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D + vds * 4;
  };
  kvec8 kstage[2];
  kvec4 vstage[4];
  {
    const KVT* ks = k_seg_ptr(split);
    const KVT* vq = v_quad_ptr(split);
    kstage[0] = *(const kvec8*)ks;
    kstage[1] = *(const kvec8*)(ks + 64);
#pragma unroll
    for (int i = 0; i < 4; i++) vstage[i] = *(const kvec4*)(vq + i * D);
  }

  for (int chunk = split; chunk < n_chunks; chunk += n_splits) {
    const int tok0 = chunk * PS_CHUNK32;
    // write prefetched K segments (16B-slot XOR swizzle by token low bits;
    // the XOR only permutes the low 3 slot bits, so slot+8 stays in 8..15)
    *(ps_bf16x8*)(&k_lds[ktok][(kslot ^ (ktok & 7)) * 8]) =
        KVTr::to_bf16x8(kstage[0]);
    *(ps_bf16x8*)(&k_lds[ktok][((kslot ^ (ktok & 7)) + 8) * 8]) =
        KVTr::to_bf16x8(kstage[1]);
    // write prefetched V quad transposed (octet swizzle by dim low bits)
    {
      const int vo = vtq >> 1;  // token octet
      const int off = (vtq & 1) * 4;
#pragma unroll
      for (int dd = 0; dd < 4; dd++) {
        const int d = vds * 4 + dd;
        unsigned short r4[4];
#pragma unroll
        for (int i = 0; i < 4; i++) r4[i] = KVTr::to_bf16(vstage[i][dd]);
        *(ps_bf16x4*)(&v_t[d][(vo ^ (d & 7)) * 8 + off]) =
            *(const ps_bf16x4*)r4;
      }
    }
    __syncthreads();
    if (chunk + n_splits < n_chunks) {  // T14: next loads under compute
      const KVT* ks = k_seg_ptr(chunk + n_splits);
      const KVT* vq = v_quad_ptr(chunk + n_splits);
      kstage[0] = *(const kvec8*)ks;
      kstage[1] = *(const kvec8*)(ks + 64);
#pragma unroll
      for (int i = 0; i < 4; i++) vstage[i] = *(const kvec4*)(vq + i * D);
    }

    const bool wave_active = (wq0 < n_rows) && (tok0 <= wave_pos_max) &&
                             (tok0 + PS_CHUNK32 > wave_win_lo);
    if (wave_active) {
      // ---- swapped QK^T: S^T[kv=kt*32+.., q] over 2 kv col-tiles ----
      ps_mf32x16 s_frag[2];
#pragma unroll
      for (int kt = 0; kt < 2; kt++) {
#pragma unroll
        for (int r = 16; r-- > 0;) s_frag[kt][r] = 0.f;
        const int ktok_row = kt * 32 + rc;  // A-row: kv token in chunk
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < NKS; ks++) {
          const int slot = ((ks * 2 + hi) ^ (ktok_row & 7));
          ps_mbf16x8 k_frag = ps_as_mbf16_32(
              *(const ps_bf16x8*)(&k_lds[ktok_row][slot * 8]));
          s_frag[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              k_frag, q_frag[ks], s_frag[kt], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      // ---- mask + lane-local online softmax ----
      bool full_chunk = (tok0 + PS_CHUNK32 - 1 <= q_pos0 + wq0) &&
                        (wq0 + 31 < n_rows);
      if (window > 0)
        full_chunk = full_chunk && (tok0 >= wave_pos_max - window + 1);
      float m_new = m_run;
      if (full_chunk) {
#pragma unroll
        for (int kt = 0; kt < 2; kt++)
#pragma unroll
          for (int r = 0; r < 16; r++)
            m_new = fmaxf(m_new, s_frag[kt][r]);
      } else {
        const bool row_real = (wq0 + rc) < n_rows;
#pragma unroll
        for (int kt = 0; kt < 2; kt++) {
#pragma unroll
          for (int r = 0; r < 16; r++) {
            const int kv_pos =
                tok0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            bool valid = row_real && (kv_pos <= q_pos);
            if (window > 0) valid = valid && (kv_pos > q_pos - window);
            const float sv = valid ? s_frag[kt][r] : PS_NEG_INF;
            s_frag[kt][r] = sv;
            m_new = fmaxf(m_new, sv);
          }
        }
      }
      m_new = fmaxf(m_new, __shfl_xor(m_new, 32));  // join half-waves
      // T13 defer-max: skip the O rescale while the row max grew < THR
      constexpr float PS_RESCALE_THR32 = 11.54f;  // 8 nats in log2 units
      const bool grew = (m_new > m_run + PS_RESCALE_THR32) ||
                        (m_run == PS_NEG_INF && m_new > PS_NEG_INF);
      if (__any(grew)) {
        const float corr = exp2f(m_run - m_new);
        l_run *= corr;
        m_run = m_new;
        // broadcast corr from the owner lane of each accumulator row
#pragma unroll
        for (int r = 0; r < 16; r++) {
          const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float c = __shfl(corr, row);
#pragma unroll
          for (int dt = 0; dt < 4; dt++) o_acc[dt][r] *= c;
        }
      }
      float psum = 0.f;
#pragma unroll
      for (int kt = 0; kt < 2; kt++) {
#pragma unroll
        for (int r = 0; r < 16; r++) {
          const float p = s_frag[kt][r] > PS_NEG_INF
                              ? exp2f(s_frag[kt][r] - m_run)
                              : 0.f;
          s_frag[kt][r] = p;
          psum += p;
        }
      }
      l_run += psum + __shfl_xor(psum, 32);
      // ---- P to per-wave LDS (S-layout -> A-frag slots, swizzled) ----
      // reg quad r=4qd..4qd+3 holds 4 consecutive kv tokens -> one b64
      // store per (kt, qd): 8 vector stores/lane. (An in-register
      // T12-style shfl_xor exchange measured 7% SLOWER than this path --
      // the vectorized LDS round-trip is cheap at this occupancy.)
#pragma unroll
      for (int kt = 0; kt < 2; kt++)
#pragma unroll
        for (int qd = 0; qd < 4; qd++) {
          unsigned short p4[4];
#pragma unroll
          for (int j = 0; j < 4; j++)
            p4[j] = ps_f32_to_bf16(s_frag[kt][qd * 4 + j]);
          const int slot = (kt * 4 + qd) ^ (rc & 7);
          *(ps_bf16x4*)(&p_lds[wave][rc][slot * 8 + 4 * hi]) =
              *(const ps_bf16x4*)p4;
        }
      // ---- PV: O[32q x 128d] += P[32q x 64kv] @ V[64kv x 128d] ----
#pragma unroll
      for (int kc = 0; kc < 4; kc++) {
        const int pslot = ((kc * 2 + hi) ^ (rc & 7));
        ps_mbf16x8 p_frag = ps_as_mbf16_32(
            *(const ps_bf16x8*)(&p_lds[wave][rc][pslot * 8]));
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int dt = 0; dt < 4; dt++) {
          const int d = dt * 32 + rc;
          const int vo = (kc * 2 + hi) ^ (d & 7);
          ps_mbf16x8 v_frag =
              ps_as_mbf16_32(*(const ps_bf16x8*)(&v_t[d][vo * 8]));
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              p_frag, v_frag, o_acc[dt], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }
    __syncthreads();  // protect k_lds/v_t before next chunk's staging
  }

  if (n_splits > 1) {
    // ---- epilogue: write unnormalized partials for the combiner ----
    if (hi == 0 && wq0 + rc < n_rows) {
      const long t = q_tok0 + wq0 + rc;
      ws_ml[((long)split * q_tokens + t) * QH + qh] = m_run;
      ws_ml[((long)(n_splits + split) * q_tokens + t) * QH + qh] = l_run;
    }
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int lrow = wq0 + row;
      if (lrow >= n_rows) continue;
      float* orow = ws_o +
          (((long)split * q_tokens + q_tok0 + lrow) * QH + qh) * D;
#pragma unroll
      for (int dt = 0; dt < 4; dt++)
        orow[dt * 32 + rc] = o_acc[dt][r];
    }
    return;
  }

  // ---- epilogue: normalize + store ----
  const float inv_own = l_run > 0.f ? 1.f / l_run : 0.f;
#pragma unroll
  for (int r = 0; r < 16; r++) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int lrow = wq0 + row;
    const float inv = __shfl(inv_own, row);
    if (lrow >= n_rows) continue;
    unsigned short* orow = out + ((long)(q_tok0 + lrow) * QH + qh) * D;
#pragma unroll
    for (int dt = 0; dt < 4; dt++)
      orow[dt * 32 + rc] = ps_f32_to_bf16(o_acc[dt][r] * inv);
  }
}

// merge the splits: out = sum_s w_s O_s / sum_s w_s l_s, w_s=exp2(m_s-M)
__global__ void prefill_split_combine_kernel(
    unsigned short* __restrict__ out,      // [T, QH, 128]
    const float* __restrict__ ws_o,        // [S, T, QH, 128]
    const float* __restrict__ ws_ml,       // [2, S, T, QH]
    long q_tokens, int QH, int n_splits) {
  const long th = blockIdx.x;  // t * QH + qh
  if (th >= q_tokens * QH) return;
  const int d = threadIdx.x;
  float M = PS_NEG_INF;
  for (int s = 0; s < n_splits; s++)
    M = fmaxf(M, ws_ml[(long)s * q_tokens * QH + th]);
  float denom = 0.f, acc = 0.f;
  for (int s = 0; s < n_splits; s++) {
    const float m = ws_ml[(long)s * q_tokens * QH + th];
    const float l = ws_ml[((long)(n_splits + s) * q_tokens) * QH + th];
    const float w = (m > PS_NEG_INF && l > 0.f) ? exp2f(m - M) : 0.f;
    denom += w * l;
    acc += w * ws_o[((long)s * q_tokens * QH + th) * 128 + d];
  }
  out[th * 128 + d] =
      ps_f32_to_bf16(denom > 0.f ? acc / denom : 0.f);
}

extern "C" {

// Split count the launcher will use for a shape (callers size ws with it)
int ps_prefill_mfma32_splits(int num_tiles, int KH, int GQ) {
  int gqw = 1;
  while (gqw < 4 && GQ % (gqw * 2) == 0) gqw *= 2;
  const int n_work = num_tiles * KH * (GQ / gqw);
  if (n_work >= 500) return 1;
  int s = (500 + n_work - 1) / n_work;
  return s > 8 ? 8 : s;
}

int ps_paged_attn_prefill_mfma32(void* out, void* ws_o, void* ws_ml,
                                 long q_tokens, const void* q,
                                 const void* k_cache, const void* v_cache,
                                 const void* block_tables,
                                 const void* tile_info, int num_tiles,
                                 int num_q_heads, int max_blocks,
                                 float scale, int KH, int GQ, int head_dim,
                                 long q_stride, int kv_fp8, int window,
                                 hipStream_t stream) {
  if (head_dim != 128) return -1;
  // GQW = largest power-of-2 divisor of GQ, capped at 4 so the tile row
  // capacity 32*(8/GQW) never drops below 64 (the engine and tests build
  // 64..256-row tiles via ops.prefill_tile_rows).
  int gqw = 1;
  while (gqw < 4 && GQ % (gqw * 2) == 0) gqw *= 2;
  const int n_work = num_tiles * KH * (GQ / gqw);
  int splits = ps_prefill_mfma32_splits(num_tiles, KH, GQ);
  if (ws_o == nullptr) splits = 1;
  dim3 grid(((n_work + 7) / 8) * 8, splits);
#define PS_PF32_LAUNCH(GQW, KVT)                                          \
  paged_attn_prefill_mfma32_kernel<128, GQW, KVT>                         \
      <<<grid, 512, 0, stream>>>(                                         \
          (unsigned short*)out, (float*)ws_o, (float*)ws_ml, q_tokens,    \
          (const unsigned short*)q,                                       \
          (const KVT*)k_cache, (const KVT*)v_cache,                       \
          (const int*)block_tables, (const int*)tile_info, max_blocks,    \
          scale, KH, GQ, q_stride, num_q_heads, n_work, window)
  if (kv_fp8) {
    if (gqw == 4) PS_PF32_LAUNCH(4, unsigned char);
    else if (gqw == 2) PS_PF32_LAUNCH(2, unsigned char);
    else PS_PF32_LAUNCH(1, unsigned char);
  } else {
    if (gqw == 4) PS_PF32_LAUNCH(4, unsigned short);
    else if (gqw == 2) PS_PF32_LAUNCH(2, unsigned short);
    else PS_PF32_LAUNCH(1, unsigned short);
  }
#undef PS_PF32_LAUNCH
  if (splits > 1) {
    prefill_split_combine_kernel<<<
        dim3((unsigned)(q_tokens * num_q_heads)), 128, 0, stream>>>(
        (unsigned short*)out, (const float*)ws_o, (const float*)ws_ml,
        q_tokens, num_q_heads, splits);
  }
  return 0;
}

}  // extern "C"
