// Paged attention kernels for MI355X (gfx950), bf16 KV cache.
//
// Behavioural parity target: the paged-attention op surface the reference
// stack exercises through its engine images —
// paged KV, GQA, causal chunked prefill, streaming decode.
//
// MI355X-first design (not a port):
//  * decode is HBM-bandwidth-bound on the KV read -> one workgroup per
//    (sequence, kv_head) so the GQA query group shares every KV byte read;
//    16-lane sub-groups each own one KV token (16 lanes x ushort8 = 128 dims)
//    so all loads are contiguous 16 B per lane, and the score reduction is a
//    4-step shfl_xor inside the sub-group (wave64-native, no LDS traffic in
//    the inner loop).
//  * online softmax (flash-decode) held entirely in registers; wave-level
//    merge via width-64 shuffles, workgroup merge via a small LDS exchange.
//  * KV cache layout [num_blocks, kv_heads, block_size, head_dim] keeps one
//    (block, head) tile contiguous (block_size x head_dim x 2 B = 4 KiB for
//    head_dim 128, block 16): a whole tile streams as perfectly coalesced
//    16 B lane loads.
#include "ps_common.h"

// ---------------------------------------------------------------------------
// Decode: one query token per sequence, flash-decode split-KV.
//   grid = (num_seqs, num_kv_heads, num_splits), block = NWAVES * 64
// With num_splits == 1 the result is written directly to `out`; otherwise
// each split writes a partial (m, l, acc) to the fp32 workspace and a small
// combine kernel merges the splits. Splitting exists because at decode batch
// sizes of 16-64 seqs, seqs*kv_heads workgroups cannot fill 256 CUs and the
// kernel runs at <10% of HBM bandwidth (measured, profiles/ run1).
// ---------------------------------------------------------------------------
template <int HEAD_DIM, int GQ, int BLOCK_SIZE, int NWAVES, int LOWREG,
          typename KVT>
// LOWREG: cap registers so 3 workgroups co-reside per SIMD (168-VGPR
// budget; the fp8 instantiation otherwise lands on 170 -> 176 alloc ->
// 2 waves and runs 36% slower despite half the KV bytes).
__global__ __launch_bounds__(NWAVES * 64, LOWREG ? 3 : 1) void paged_attn_decode_kernel(
    unsigned short* __restrict__ out,            // [S, QH, HEAD_DIM]
    float* __restrict__ ws_acc,   // [S, QH, SPLITS, HD] (splits > 1)
    float* __restrict__ ws_ml,    // [S, QH, SPLITS, 2]
    const unsigned short* __restrict__ q,        // [S, QH, HEAD_DIM]
    const KVT* __restrict__ k_cache,             // [NB, KH, BS, HD]
    const KVT* __restrict__ v_cache,             // [NB, KH, BS, HD]
    const int* __restrict__ block_tables,        // [S, max_blocks]
    const int* __restrict__ seq_lens,            // [S]
    int max_blocks, float scale, int KH, long q_stride, int window) {
  using KVTr = ps_kv_traits<KVT>;
  using kvec8 = typename KVTr::vec8;
  constexpr int D = HEAD_DIM;
  constexpr int LPG = D / 8;       // lanes per token sub-group (16 @ D=128)
  constexpr int TPW = 64 / LPG;    // tokens per wave per iteration (4)
  static_assert(BLOCK_SIZE % TPW == 0, "block size must divide");

  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int ctx = seq_lens[seq];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int sg = lane / LPG;   // sub-group id within wave
  const int sl = lane % LPG;   // lane within sub-group -> dims [sl*8, sl*8+8)

  // Q for the whole GQA group, kept PACKED bf16 (8 dims per lane per head
  // = 4 VGPRs/head instead of 8 fp32): at GQ=4 this moves the kernel from
  // 2 to 3 waves/SIMD occupancy, which matters more than the extra
  // convert in the dot (the kernel is HBM-latency-bound). The softmax
  // scale is applied to the reduced score instead of to q.
  ps_bf16x8 qp[GQ];
  const unsigned short* qbase = q + (long)seq * q_stride + (long)kvh * GQ * D;
#pragma unroll
  for (int g = 0; g < GQ; g++)
    qp[g] = *(const ps_bf16x8*)(qbase + g * D + sl * 8);
  float qf[GQ][8];
  if constexpr (!LOWREG) {
#pragma unroll
    for (int g = 0; g < GQ; g++)
#pragma unroll
      for (int j = 0; j < 8; j++) qf[g][j] = ps_bf16_to_f32(qp[g][j]);
  }

  float m[GQ], l[GQ], acc[GQ][8];
#pragma unroll
  for (int g = 0; g < GQ; g++) {
    m[g] = PS_NEG_INF;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; j++) acc[g][j] = 0.f;
  }

  const int nblocks = (ctx + BLOCK_SIZE - 1) / BLOCK_SIZE;
  // sliding window (Mistral-style): only keys with pos >= win_lo attend
  const int win_lo = (window > 0 && ctx > window) ? ctx - window : 0;
  const int first_block = win_lo / BLOCK_SIZE;
  // split-KV: this workgroup handles blocks [b_begin, b_end)
  const int nsplit = gridDim.z;
  const int split = blockIdx.z;
  const int live_blocks = nblocks - first_block;
  const int per_split = (live_blocks + nsplit - 1) / nsplit;
  const int b_begin = first_block + split * per_split;
  const int b_end = min(nblocks, b_begin + per_split);
  const int* bt = block_tables + (long)seq * max_blocks;
  // Per paged block: score all of this sub-group's tokens first
  // (independent dot products -> independent exps), then ONE online-softmax
  // rescale per block. The naive per-token update serializes
  // exp->mul->fma chains 4x deeper and left the kernel latency-bound at
  // <50% of HBM bandwidth (profiles/ attn_bench).
  constexpr int TPB = BLOCK_SIZE / TPW;  // tokens per sub-group per block (4)
  for (int b = b_begin + wave; b < b_end; b += NWAVES) {
    const long blk = bt[b];
    const KVT* kb = k_cache + ((blk * KH + kvh) * BLOCK_SIZE) * D;
    const KVT* vb = v_cache + ((blk * KH + kvh) * BLOCK_SIZE) * D;
    const int valid_tokens = min(BLOCK_SIZE, ctx - b * BLOCK_SIZE);
    float sc[GQ][TPB];
    kvec8 vv[TPB];
    if constexpr (LOWREG) {
      // keep q packed: block the compiler from hoisting converted copies
#pragma unroll
      for (int g = 0; g < GQ; g++)
        asm volatile("" : "+v"(qp[g]));
    }
#pragma unroll
    for (int tt = 0; tt < TPB; tt++) {
      const int tok = tt * TPW + sg;
      const bool valid =
          tok < valid_tokens && b * BLOCK_SIZE + tok >= win_lo;
      kvec8 kv = *(const kvec8*)(kb + tok * D + sl * 8);
      if constexpr (!LOWREG)
        vv[tt] = *(const kvec8*)(vb + tok * D + sl * 8);
      float kf[8];
      KVTr::to_f32x8(kv, kf);  // packed v_cvt_pk_f32_fp8 on fp8 caches
#pragma unroll
      for (int g = 0; g < GQ; g++) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < 8; j++) {
          if constexpr (LOWREG)
            s += ps_bf16_to_f32(qp[g][j]) * kf[j];
          else
            s += qf[g][j] * kf[j];
        }
        s = ps_group_sum<LPG>(s) * scale;
        sc[g][tt] = valid ? s : PS_NEG_INF;
      }
    }
    if constexpr (LOWREG) {
#pragma unroll
      for (int tt = 0; tt < TPB; tt++)
        vv[tt] = *(const kvec8*)(vb + (tt * TPW + sg) * D + sl * 8);
    }
#pragma unroll
    for (int g = 0; g < GQ; g++) {
      float bmax = sc[g][0];
#pragma unroll
      for (int tt = 1; tt < TPB; tt++) bmax = fmaxf(bmax, sc[g][tt]);
      const float mnew = fmaxf(m[g], bmax);
      const float corr = __expf(m[g] - mnew);  // <=1; 1 when both -inf
      float p[TPB];
      float psum = 0.f;
#pragma unroll
      for (int tt = 0; tt < TPB; tt++) {
        p[tt] = sc[g][tt] > PS_NEG_INF ? __expf(sc[g][tt] - mnew) : 0.f;
        psum += p[tt];
      }
      l[g] = l[g] * corr + psum;
#pragma unroll
      for (int j = 0; j < 8; j++) acc[g][j] *= corr;
#pragma unroll
      for (int tt = 0; tt < TPB; tt++) {
        float vf[8];
        KVTr::to_f32x8(vv[tt], vf);
#pragma unroll
        for (int j = 0; j < 8; j++) acc[g][j] += p[tt] * vf[j];
      }
      m[g] = mnew;
    }
  }

  // Merge the TPW sub-groups of this wave (lanes sl, sl+LPG, ... hold
  // partial state for the same dims).
#pragma unroll
  for (int off = LPG; off < 64; off <<= 1) {
#pragma unroll
    for (int g = 0; g < GQ; g++) {
      const float mo = __shfl_xor(m[g], off, 64);
      const float lo = __shfl_xor(l[g], off, 64);
      const float mnew = fmaxf(m[g], mo);
      const float c1 = __expf(m[g] - mnew);
      const float c2 = __expf(mo - mnew);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float ao = __shfl_xor(acc[g][j], off, 64);
        acc[g][j] = acc[g][j] * c1 + ao * c2;
      }
      l[g] = l[g] * c1 + lo * c2;
      m[g] = mnew;
    }
  }

  // Merge the NWAVES waves via LDS.
  __shared__ float sm[NWAVES][GQ];
  __shared__ float slse[NWAVES][GQ];
  __shared__ float sacc[NWAVES][GQ][D];
  if (sg == 0) {
#pragma unroll
    for (int g = 0; g < GQ; g++) {
      if (sl == 0) {
        sm[wave][g] = m[g];
        slse[wave][g] = l[g];
      }
#pragma unroll
      for (int j = 0; j < 8; j++) sacc[wave][g][sl * 8 + j] = acc[g][j];
    }
  }
  __syncthreads();
  if (wave == 0 && sg == 0) {
#pragma unroll
    for (int g = 0; g < GQ; g++) {
      float M = PS_NEG_INF, L = 0.f, A[8];
#pragma unroll
      for (int j = 0; j < 8; j++) A[j] = 0.f;
#pragma unroll
      for (int wv = 0; wv < NWAVES; wv++) {
        const float mw = sm[wv][g];
        const float mnew = fmaxf(M, mw);
        const float c1 = __expf(M - mnew);
        const float c2 = __expf(mw - mnew);
        L = L * c1 + slse[wv][g] * c2;
#pragma unroll
        for (int j = 0; j < 8; j++)
          A[j] = A[j] * c1 + sacc[wv][g][sl * 8 + j] * c2;
        M = mnew;
      }
      const long head = (long)seq * KH * GQ + (long)kvh * GQ + g;
      if (nsplit == 1) {
        const float inv = L > 0.f ? 1.f / L : 0.f;
        ps_bf16x8 ov;
#pragma unroll
        for (int j = 0; j < 8; j++) ov[j] = ps_f32_to_bf16(A[j] * inv);
        *(ps_bf16x8*)(out + head * D + sl * 8) = ov;
      } else {
        float* wa = ws_acc + (head * nsplit + split) * D + sl * 8;
#pragma unroll
        for (int j = 0; j < 8; j++) wa[j] = A[j];
        if (sl == 0) {
          float* wm = ws_ml + (head * nsplit + split) * 2;
          wm[0] = M;
          wm[1] = L;
        }
      }
    }
  }
}

// Combine the per-split partials: grid = (S * QH), block = HEAD_DIM/8 lanes
// padded to a wave.
template <int HEAD_DIM>
__global__ __launch_bounds__(64) void paged_attn_combine_kernel(
    unsigned short* __restrict__ out,   // [S*QH, HD]
    const float* __restrict__ ws_acc,   // [S*QH, SPLITS, HD]
    const float* __restrict__ ws_ml,    // [S*QH, SPLITS, 2]
    int nsplit) {
  constexpr int D = HEAD_DIM;
  constexpr int LPG = D / 8;
  const long head = blockIdx.x;
  const int sl = threadIdx.x;
  if (sl >= LPG) return;
  float M = PS_NEG_INF, L = 0.f, A[8];
#pragma unroll
  for (int j = 0; j < 8; j++) A[j] = 0.f;
  for (int s = 0; s < nsplit; s++) {
    const float mw = ws_ml[(head * nsplit + s) * 2 + 0];
    const float lw = ws_ml[(head * nsplit + s) * 2 + 1];
    const float mnew = fmaxf(M, mw);
    const float c1 = __expf(M - mnew);
    const float c2 = __expf(mw - mnew);
    const float* wa = ws_acc + (head * nsplit + s) * D + sl * 8;
#pragma unroll
    for (int j = 0; j < 8; j++) A[j] = A[j] * c1 + wa[j] * c2;
    L = L * c1 + lw * c2;
    M = mnew;
  }
  const float inv = L > 0.f ? 1.f / L : 0.f;
  ps_bf16x8 ov;
#pragma unroll
  for (int j = 0; j < 8; j++) ov[j] = ps_f32_to_bf16(A[j] * inv);
  *(ps_bf16x8*)(out + head * D + sl * 8) = ov;
}

// ---------------------------------------------------------------------------
// Chunked prefill: a batch of query tokens (possibly many sequences, variable
// lengths) attends causally to the paged KV cache, which already contains
// every query token's K/V (appended before the call).
//   grid = (num_q_tokens, num_q_heads), block = 64 (one wave per (token, head))
// Compute-bound VALU v1; MFMA tile version is the planned upgrade.
// ---------------------------------------------------------------------------
template <int HEAD_DIM, int BLOCK_SIZE>
__global__ __launch_bounds__(64) void paged_attn_prefill_kernel(
    unsigned short* __restrict__ out,            // [T, QH, HD]
    const unsigned short* __restrict__ q,        // [T, QH, HD]
    const unsigned short* __restrict__ k_cache,  // [NB, KH, BS, HD]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ token_seq,     // [T] sequence index per q token
    const int* __restrict__ token_pos,     // [T] absolute position per q token
    int max_blocks, float scale, int KH, int GQ, long q_stride,
    int window) {
  constexpr int D = HEAD_DIM;
  constexpr int LPG = D / 8;
  constexpr int TPW = 64 / LPG;

  const int tokid = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / GQ;
  const int seq = token_seq[tokid];
  const int pos = token_pos[tokid];
  const int ctx = pos + 1;  // causal: attend to [0, pos]
  const int lane = threadIdx.x & 63;
  const int sg = lane / LPG;
  const int sl = lane % LPG;
  const int QH = KH * GQ;

  float qf[8];
  {
    ps_bf16x8 qv =
        *(const ps_bf16x8*)(q + (long)tokid * q_stride + (long)qh * D + sl * 8);
#pragma unroll
    for (int j = 0; j < 8; j++) qf[j] = ps_bf16_to_f32(qv[j]) * scale;
  }

  float m = PS_NEG_INF, l = 0.f, acc[8];
#pragma unroll
  for (int j = 0; j < 8; j++) acc[j] = 0.f;

  const int nblocks = (ctx + BLOCK_SIZE - 1) / BLOCK_SIZE;
  const int win_lo = (window > 0 && ctx > window) ? ctx - window : 0;
  const int* bt = block_tables + (long)seq * max_blocks;
  for (int b = win_lo / BLOCK_SIZE; b < nblocks; b++) {
    const long blk = bt[b];
    const unsigned short* kb = k_cache + ((blk * KH + kvh) * BLOCK_SIZE) * D;
    const unsigned short* vb = v_cache + ((blk * KH + kvh) * BLOCK_SIZE) * D;
    const int valid_tokens = min(BLOCK_SIZE, ctx - b * BLOCK_SIZE);
#pragma unroll
    for (int tt = 0; tt < BLOCK_SIZE / TPW; tt++) {
      const int tok = tt * TPW + sg;
      const bool valid =
          tok < valid_tokens && b * BLOCK_SIZE + tok >= win_lo;
      ps_bf16x8 kv = *(const ps_bf16x8*)(kb + tok * D + sl * 8);
      ps_bf16x8 vv = *(const ps_bf16x8*)(vb + tok * D + sl * 8);
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; j++) s += qf[j] * ps_bf16_to_f32(kv[j]);
      s = ps_group_sum<LPG>(s);
      const float sv = valid ? s : PS_NEG_INF;
      const float mnew = fmaxf(m, sv);
      const float corr = __expf(m - mnew);
      const float p = valid ? __expf(sv - mnew) : 0.f;
      l = l * corr + p;
#pragma unroll
      for (int j = 0; j < 8; j++)
        acc[j] = acc[j] * corr + p * ps_bf16_to_f32(vv[j]);
      m = mnew;
    }
  }

  // merge sub-groups within the wave
#pragma unroll
  for (int off = LPG; off < 64; off <<= 1) {
    const float mo = __shfl_xor(m, off, 64);
    const float lo = __shfl_xor(l, off, 64);
    const float mnew = fmaxf(m, mo);
    const float c1 = __expf(m - mnew);
    const float c2 = __expf(mo - mnew);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const float ao = __shfl_xor(acc[j], off, 64);
      acc[j] = acc[j] * c1 + ao * c2;
    }
    l = l * c1 + lo * c2;
    m = mnew;
  }
  if (sg == 0) {
    const float inv = l > 0.f ? 1.f / l : 0.f;
    ps_bf16x8 ov;
#pragma unroll
    for (int j = 0; j < 8; j++) ov[j] = ps_f32_to_bf16(acc[j] * inv);
    *(ps_bf16x8*)(out + ((long)tokid * QH + qh) * D + sl * 8) = ov;
  }
}

// ---------------------------------------------------------------------------
// KV append (reshape_and_cache): scatter the new tokens' K/V into the paged
// cache according to slot_mapping (slot = block_id * BLOCK_SIZE + offset).
// ---------------------------------------------------------------------------
template <typename KVT>
__global__ void reshape_and_cache_kernel(
    const unsigned short* __restrict__ k,  // [T, KH*HD]
    const unsigned short* __restrict__ v,  // [T, KH*HD]
    KVT* __restrict__ k_cache,             // [NB, KH, BS, HD]
    KVT* __restrict__ v_cache,
    const long* __restrict__ slot_mapping,  // [T]
    int KH, int HD, int BS, long total /* = T * KH * HD / 8 */) {
  using KVTr = ps_kv_traits<KVT>;
  using kvec8 = typename KVTr::vec8;
  const long stride = (long)gridDim.x * blockDim.x;
  const int row = KH * HD;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += stride) {
    const long elem = idx * 8;
    const long t = elem / row;
    const int r = (int)(elem % row);
    const int h = r / HD;
    const int d = r % HD;
    const long slot = slot_mapping[t];
    if (slot < 0) continue;
    const long blk = slot / BS;
    const int off = (int)(slot % BS);
    const long dst = ((blk * KH + h) * BS + off) * (long)HD + d;
    ps_bf16x8 kv = *(const ps_bf16x8*)(k + elem);
    ps_bf16x8 vv = *(const ps_bf16x8*)(v + elem);
    kvec8 ko, vo;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      ko[j] = KVTr::from_f32(ps_bf16_to_f32(kv[j]));
      vo[j] = KVTr::from_f32(ps_bf16_to_f32(vv[j]));
    }
    *(kvec8*)(k_cache + dst) = ko;
    *(kvec8*)(v_cache + dst) = vo;
  }
}

// ---------------------------------------------------------------------------
// Greedy sampling: argmax over the vocab per row.
//   grid = (num_rows), block = 256
// ---------------------------------------------------------------------------
__global__ void greedy_sample_kernel(long* __restrict__ out,  // [R]
                                     const unsigned short* __restrict__ logits,
                                     int V) {
  const long r = blockIdx.x;
  const unsigned short* row = logits + r * (long)V;
  float best = PS_NEG_INF;
  int besti = 0;
  for (int i = threadIdx.x * 8; i + 8 <= V; i += blockDim.x * 8) {
    ps_bf16x8 xv = *(const ps_bf16x8*)(row + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const float f = ps_bf16_to_f32(xv[j]);
      if (f > best) {
        best = f;
        besti = i + j;
      }
    }
  }
  // tail (V not multiple of 8*block)
  const int tail_start = (V / 8) * 8;
  for (int i = tail_start + threadIdx.x; i < V; i += blockDim.x) {
    const float f = ps_bf16_to_f32(row[i]);
    if (f > best) {
      best = f;
      besti = i;
    }
  }
  // wave reduce keeping the smallest index among ties
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, 64);
    const int oi = __shfl_xor(besti, off, 64);
    if (ob > best || (ob == best && oi < besti)) {
      best = ob;
      besti = oi;
    }
  }
  __shared__ float wbest[4];
  __shared__ int wbesti[4];
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) {
    wbest[wave] = best;
    wbesti[wave] = besti;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int wv = 1; wv < (int)(blockDim.x >> 6); wv++) {
      if (wbest[wv] > wbest[0] ||
          (wbest[wv] == wbest[0] && wbesti[wv] < wbesti[0])) {
        wbest[0] = wbest[wv];
        wbesti[0] = wbesti[wv];
      }
    }
    out[r] = wbesti[0];
  }
}

// ---------------------------------------------------------------------------
// C launchers
// ---------------------------------------------------------------------------
extern "C" {

int ps_paged_attn_decode(void* out, void* ws_acc, void* ws_ml, const void* q,
                         const void* k_cache, const void* v_cache,
                         const void* block_tables, const void* seq_lens,
                         int num_seqs, int max_blocks, float scale, int KH,
                         int GQ, int head_dim, int block_size, int num_splits,
                         long q_stride, int variant, int kv_fp8,
                         int window, hipStream_t stream) {
  dim3 grid(num_seqs, KH, num_splits);
  constexpr int NW = 4;
  dim3 block(NW * 64);
#define PS_DECODE_T(HD, G, BS, LR, KVT)                                      \
  paged_attn_decode_kernel<HD, G, BS, NW, LR, KVT>                           \
      <<<grid, block, 0, stream>>>(                                          \
          (unsigned short*)out, (float*)ws_acc, (float*)ws_ml,               \
          (const unsigned short*)q, (const KVT*)k_cache,                     \
          (const KVT*)v_cache, (const int*)block_tables,                     \
          (const int*)seq_lens, max_blocks, scale, KH, q_stride, window)
#define PS_DISPATCH_DECODE(HD, G, BS)                                        \
  do {                                                                       \
    if (kv_fp8) {                                                            \
      if (variant == 1)                                                      \
        PS_DECODE_T(HD, G, BS, 1, unsigned char);                            \
      else                                                                   \
        PS_DECODE_T(HD, G, BS, 0, unsigned char);                            \
    } else {                                                                 \
      if (variant == 1)                                                      \
        PS_DECODE_T(HD, G, BS, 1, unsigned short);                           \
      else                                                                   \
        PS_DECODE_T(HD, G, BS, 0, unsigned short);                           \
    }                                                                        \
    if (num_splits > 1)                                                      \
      paged_attn_combine_kernel<HD>                                          \
          <<<dim3(num_seqs * KH * G), 64, 0, stream>>>(                      \
              (unsigned short*)out, (const float*)ws_acc,                    \
              (const float*)ws_ml, num_splits);                              \
  } while (0)
  if (head_dim == 128 && block_size == 16) {
    switch (GQ) {
      case 1: PS_DISPATCH_DECODE(128, 1, 16); return 0;
      case 2: PS_DISPATCH_DECODE(128, 2, 16); return 0;
      case 3: PS_DISPATCH_DECODE(128, 3, 16); return 0;
      case 4: PS_DISPATCH_DECODE(128, 4, 16); return 0;
      case 6: PS_DISPATCH_DECODE(128, 6, 16); return 0;
      case 7: PS_DISPATCH_DECODE(128, 7, 16); return 0;
      case 8: PS_DISPATCH_DECODE(128, 8, 16); return 0;
      default: return -1;
    }
  }
  if (head_dim == 64 && block_size == 16 && !kv_fp8) {
    switch (GQ) {
      case 1: PS_DISPATCH_DECODE(64, 1, 16); return 0;
      case 2: PS_DISPATCH_DECODE(64, 2, 16); return 0;
      case 4: PS_DISPATCH_DECODE(64, 4, 16); return 0;
      case 8: PS_DISPATCH_DECODE(64, 8, 16); return 0;
      default: return -1;
    }
  }
  return -1;
#undef PS_DECODE_T
#undef PS_DISPATCH_DECODE
}

int ps_paged_attn_prefill(void* out, const void* q, const void* k_cache,
                          const void* v_cache, const void* block_tables,
                          const void* token_seq, const void* token_pos,
                          int num_tokens, int num_q_heads, int max_blocks,
                          float scale, int KH, int GQ, int head_dim,
                          int block_size, long q_stride, int window,
                          hipStream_t stream) {
  dim3 grid(num_tokens, num_q_heads);
  dim3 block(64);
#define PS_DISPATCH_PREFILL(HD, BS)                                         \
  paged_attn_prefill_kernel<HD, BS><<<grid, block, 0, stream>>>(            \
      (unsigned short*)out, (const unsigned short*)q,                       \
      (const unsigned short*)k_cache, (const unsigned short*)v_cache,       \
      (const int*)block_tables, (const int*)token_seq,                      \
      (const int*)token_pos, max_blocks, scale, KH, GQ, q_stride, window)
  if (head_dim == 128 && block_size == 16) {
    PS_DISPATCH_PREFILL(128, 16);
    return 0;
  }
  if (head_dim == 64 && block_size == 16) {
    PS_DISPATCH_PREFILL(64, 16);
    return 0;
  }
  return -1;
#undef PS_DISPATCH_PREFILL
}

void ps_reshape_and_cache(const void* k, const void* v, void* k_cache,
                          void* v_cache, const void* slot_mapping, long T,
                          int KH, int HD, int BS, int kv_fp8,
                          hipStream_t stream) {
  const long total = T * (long)KH * HD / 8;
  const int block = 256;
  const long want = (total + block - 1) / block;
  const int grid = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  if (kv_fp8)
    reshape_and_cache_kernel<unsigned char><<<grid, block, 0, stream>>>(
        (const unsigned short*)k, (const unsigned short*)v,
        (unsigned char*)k_cache, (unsigned char*)v_cache,
        (const long*)slot_mapping, KH, HD, BS, total);
  else
    reshape_and_cache_kernel<unsigned short><<<grid, block, 0, stream>>>(
        (const unsigned short*)k, (const unsigned short*)v,
        (unsigned short*)k_cache, (unsigned short*)v_cache,
        (const long*)slot_mapping, KH, HD, BS, total);
}

void ps_greedy_sample(void* out, const void* logits, long R, int V,
                      hipStream_t stream) {
  greedy_sample_kernel<<<dim3((unsigned)R), 256, 0, stream>>>(
      (long*)out, (const unsigned short*)logits, V);
}

}  // extern "C"
