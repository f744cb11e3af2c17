// Phase-cost probe for the MFMA chunked-prefill kernel (gfx950).
//
// Compiles the production kernel body with phases compiled out (PROBE level)
// to attribute time: 0=full, 1=no V^T staging, 2=no softmax (raw scores as
// P), 3=QK^T MFMA only. Numerics are wrong for levels>0 by design — this is
// a TIMING instrument (guide §5 methodology: measure, don't guess).
//
// Build: hipcc -O3 --offload-arch=gfx950 prefill_phase_probe.hip -o probe
// Run:   ./probe [ctx] [heads]
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../ps_common.h"

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

#define PS_CHUNK 64
#define PS_PL_STRIDE 72

template <int PROBE>
__global__ __launch_bounds__(256, 3) void probe_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k_cache,
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ tile_info,
    int max_blocks, float scale, int KH, int GQ, long q_stride, int QH,
    int n_work) {
  constexpr int D = 128, BS = 16, NK = 4, NCT = 4, NKC = 2;
  const int W = n_work;
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
  const int wave = tid >> 6, lane = tid & 63, g = lane >> 4, rc = lane & 15;
  const int* bt = block_tables + (long)seq_row * max_blocks;
  const int ctx_limit = q_pos0 + n_rows;
  const int n_pages = (ctx_limit + BS - 1) / BS;
  const int n_chunks = (ctx_limit + PS_CHUNK - 1) / PS_CHUNK;
  __shared__ __align__(16) unsigned short k_lds[PS_CHUNK][D];
  __shared__ __align__(16) unsigned short v_t[D][PS_CHUNK];
  __shared__ __align__(16) unsigned short p_lds[4][16][PS_PL_STRIDE];
  const int my_local_row = wave * 16 + rc;
  const int q_row_clamped = min(my_local_row, n_rows - 1);
  const unsigned short* qrow =
      q + (long)(q_tok0 + q_row_clamped) * q_stride + (long)qh * D;
  ps_mbf16x8 q_frag[NK];
#pragma unroll
  for (int kk = 0; kk < NK; kk++)
    q_frag[kk] = ps_as_mbf16(*(const ps_bf16x8*)(qrow + kk * 32 + g * 8));
  float m_run[4], l_run[4];
  ps_mf32x4 o_acc[8];
#pragma unroll
  for (int r = 0; r < 4; r++) {
    m_run[r] = PS_NEG_INF;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int s = 0; s < 8; s++) o_acc[s] = {0.f, 0.f, 0.f, 0.f};
  const int wave_pos_max = q_pos0 + min(wave * 16 + 15, n_rows - 1);
  const int tv = tid & 63;
  const int d0 = wave * 32;
  const int lg = tv >> 3, tl = tv & 7;
  auto v_row_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK + tv;
    const int pg_idx = min(tok / BS, n_pages - 1);
    const long pg = bt[pg_idx];
    return v_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D;
  };
  auto k_row_ptr = [&](int chunk) {
    const int tok = chunk * PS_CHUNK + tv;
    const int pg_idx = min(tok / BS, n_pages - 1);
    const long pg = bt[pg_idx];
    return k_cache + ((pg * KH + kvh) * BS + (tok & (BS - 1))) * D;
  };
  ps_bf16x8 vstage[4], kstage[4];
  {
    const unsigned short* vrow = v_row_ptr(0);
    const unsigned short* krow = k_row_ptr(0);
#pragma unroll
    for (int h = 0; h < 4; h++) {
      vstage[h] = *(const ps_bf16x8*)(vrow + d0 + h * 8);
      kstage[h] = *(const ps_bf16x8*)(krow + d0 + h * 8);
    }
  }
  for (int chunk = 0; chunk < n_chunks; chunk++) {
    const int tok0 = chunk * PS_CHUNK;
#pragma unroll
    for (int h = 0; h < 4; h++) {
      const int slot = ((wave * 4 + h) ^ (tv & 7));
      *(ps_bf16x8*)(&k_lds[tv][slot * 8]) = kstage[h];
      if constexpr (PROBE < 1) {
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const int d = d0 + h * 8 + j;
          v_t[d][((lg ^ (d & 7)) << 3) + tl] = vstage[h][j];
        }
      }
    }
    __syncthreads();
    if (chunk + 1 < n_chunks) {
      const unsigned short* vrow = v_row_ptr(chunk + 1);
      const unsigned short* krow = k_row_ptr(chunk + 1);
#pragma unroll
      for (int h = 0; h < 4; h++) {
        vstage[h] = *(const ps_bf16x8*)(vrow + d0 + h * 8);
        kstage[h] = *(const ps_bf16x8*)(krow + d0 + h * 8);
      }
    }
    const bool wave_active = (wave * 16 < n_rows) && (tok0 <= wave_pos_max);
    if (wave_active) {
      ps_mf32x4 s_frag[NCT];
#pragma unroll
      for (int ct = 0; ct < NCT; ct++) {
        s_frag[ct] = {0.f, 0.f, 0.f, 0.f};
        const int trow = ct * 16 + rc;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < NK; kk++) {
          const int slot = ((kk * 4 + g) ^ (trow & 7));
          ps_mbf16x8 k_frag =
              ps_as_mbf16(*(const ps_bf16x8*)(&k_lds[trow][slot * 8]));
          s_frag[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[kk], k_frag, s_frag[ct], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      if constexpr (PROBE < 2) {
        float m_new[4];
#pragma unroll
        for (int r = 0; r < 4; r++) m_new[r] = m_run[r];
#pragma unroll
        for (int ct = 0; ct < NCT; ct++) {
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
#pragma unroll
        for (int r = 0; r < 4; r++) m_new[r] = ps_group_max<16>(m_new[r]);
        constexpr float THR = 8.f;
        bool grew = false;
#pragma unroll
        for (int r = 0; r < 4; r++)
          grew |= (m_new[r] > m_run[r] + THR) ||
                  (m_run[r] == PS_NEG_INF && m_new[r] > PS_NEG_INF);
        if (__any(grew)) {
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const float corr = __expf(m_run[r] - m_new[r]);
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
                                ? __expf(s_frag[ct][r] - m_run[r])
                                : 0.f;
            s_frag[ct][r] = p;
            psum[r] += p;
          }
        }
#pragma unroll
        for (int r = 0; r < 4; r++) l_run[r] += ps_group_sum<16>(psum[r]);
      }
      if constexpr (PROBE < 3) {
#pragma unroll
        for (int ct = 0; ct < NCT; ct++)
#pragma unroll
          for (int r = 0; r < 4; r++)
            p_lds[wave][g * 4 + r][ct * 16 + rc] =
                ps_f32_to_bf16(s_frag[ct][r]);
#pragma unroll
        for (int kc = 0; kc < NKC; kc++) {
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
      } else {
        // keep s_frag live so QK^T isn't dead-code-eliminated
#pragma unroll
        for (int ct = 0; ct < NCT; ct++)
#pragma unroll
          for (int r = 0; r < 4; r++) o_acc[ct][r] += s_frag[ct][r];
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int lrow = wave * 16 + g * 4 + r;
    if (lrow >= n_rows) continue;
    const float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 1.f;
    unsigned short* orow = out + ((long)(q_tok0 + lrow) * QH + qh) * D;
#pragma unroll
    for (int s = 0; s < 8; s++)
      orow[s * 16 + rc] = ps_f32_to_bf16(o_acc[s][r] * inv);
  }
}

#define HIP_CHECK(x)                                              \
  do {                                                            \
    hipError_t e = (x);                                           \
    if (e != hipSuccess) {                                        \
      fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e), \
              __LINE__);                                          \
      exit(1);                                                    \
    }                                                             \
  } while (0)

int main(int argc, char** argv) {
  const int ctx = argc > 1 ? atoi(argv[1]) : 4096;
  const int QH = argc > 2 ? atoi(argv[2]) : 32;
  const int KH = 8, GQ = QH / KH, D = 128, BS = 16;
  const int T = ctx;                   // q tokens (full self-attn prefill)
  const int n_tiles = (T + 63) / 64;   // 64-row q tiles
  const int n_pages = (ctx + BS - 1) / BS;
  unsigned short *q, *k, *v, *out;
  int *bt, *ti;
  HIP_CHECK(hipMalloc(&q, (size_t)T * QH * D * 2));
  HIP_CHECK(hipMalloc(&out, (size_t)T * QH * D * 2));
  HIP_CHECK(hipMalloc(&k, (size_t)n_pages * KH * BS * D * 2));
  HIP_CHECK(hipMalloc(&v, (size_t)n_pages * KH * BS * D * 2));
  HIP_CHECK(hipMalloc(&bt, n_pages * 4));
  HIP_CHECK(hipMalloc(&ti, n_tiles * 4 * 4));
  std::vector<int> h_bt(n_pages), h_ti(n_tiles * 4);
  for (int i = 0; i < n_pages; i++) h_bt[i] = i;
  for (int t = 0; t < n_tiles; t++) {
    h_ti[t * 4 + 0] = 0;
    h_ti[t * 4 + 1] = t * 64;
    h_ti[t * 4 + 2] = t * 64;
    h_ti[t * 4 + 3] = std::min(64, T - t * 64);
  }
  HIP_CHECK(hipMemcpy(bt, h_bt.data(), n_pages * 4, hipMemcpyHostToDevice));
  HIP_CHECK(
      hipMemcpy(ti, h_ti.data(), n_tiles * 16, hipMemcpyHostToDevice));
  const int n_work = n_tiles * QH;
  dim3 grid(((n_work + 7) / 8) * 8);
  const float scale = 0.088388f;
  // causal flops: sum over rows of (pos+1) ~ ctx^2/2 per head, x2 (QK+PV)
  const double flops = 2.0 * 2.0 * (double)ctx * ctx / 2 * QH * D;
  auto run = [&](auto tag, const char* name) {
    constexpr int P = decltype(tag)::value;
    for (int i = 0; i < 3; i++)
      probe_kernel<P><<<grid, 256>>>(out, q, k, v, bt, ti, n_pages, scale,
                                     KH, GQ, (long)QH * D, QH, n_work);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t a, b2;
    hipEventCreate(&a);
    hipEventCreate(&b2);
    hipEventRecord(a);
    const int iters = 20;
    for (int i = 0; i < iters; i++)
      probe_kernel<P><<<grid, 256>>>(out, q, k, v, bt, ti, n_pages, scale,
                                     KH, GQ, (long)QH * D, QH, n_work);
    hipEventRecord(b2);
    HIP_CHECK(hipEventSynchronize(b2));
    float ms;
    hipEventElapsedTime(&ms, a, b2);
    ms /= iters;
    printf("%-28s %8.3f ms  %7.1f TF\n", name, ms, flops / ms / 1e9);
  };
  run(std::integral_constant<int, 0>{}, "full");
  run(std::integral_constant<int, 1>{}, "no V-staging");
  run(std::integral_constant<int, 2>{}, "no V-staging, no softmax");
  run(std::integral_constant<int, 3>{}, "QK MFMA only");
  return 0;
}
