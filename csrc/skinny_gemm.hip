// Skinny-batch GEMM for decode steps: out[M, N] = x[M, K] @ W[N, K]^T,
// bf16 inputs, fp32 accumulation, M <= 256.
//
// Why it exists: hipBLASLt's tile selection for the decode-shaped GEMMs
// (qkv [M,4096]x[6144,4096], o, down) streams weights at only 1.4-2 TB/s at
// M in 16..128 (profiles/gemm_plain.log) when the op is purely
// weight-bandwidth-bound (~6 TB/s available). This kernel is built around
// streaming W exactly once:
//   grid = (N/64, SPLITS): workgroup owns 64 N-rows x a K-range.
//   4 waves x 16 N-rows each; B-frags are direct 16-B lane loads from W
//   (lane's col = its N-row, contiguous dims) -- W bytes are read once
//   across the whole grid.
//   x (tiny) is staged per 64-wide K-chunk in XOR-swizzled LDS and consumed
//   as A-frags for ceil(M/16) m-tiles; per-k-chunk MFMA count scales with
//   m-tiles while W traffic stays fixed.
//   Split-K partials combine via fp32 atomicAdd into the pre-zeroed output
//   (device-scope; SPLITS atomics per element).
// mfma_f32_16x16x32_bf16; operand k-pattern as verified in
// csrc/tools/mfma_probe.hip.
#include "ps_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 ps_gbf16x8;
typedef __attribute__((ext_vector_type(4))) float ps_gf32x4;

PS_DEV ps_gbf16x8 ps_as_gbf16(ps_bf16x8 u) {
  union {
    ps_bf16x8 u16;
    ps_gbf16x8 bf;
  } v;
  v.u16 = u;
  return v.bf;
}

template <int M_TILES>
__global__ __launch_bounds__(256, 4) void skinny_gemm_kernel(
    unsigned short* __restrict__ out_bf16,  // [M, N] (splits == 1)
    float* __restrict__ ws,                 // [S, M, N] (splits > 1)
    const unsigned short* __restrict__ x,   // [M, K]
    const unsigned short* __restrict__ w,   // [N, K]
    int M, int N, int K, long x_stride /* row stride of x, elems */) {
  constexpr int MP = M_TILES * 16;  // padded M
  const int n0 = blockIdx.x * 64;
  const int splits = gridDim.y;
  const int kchunks = K / 64;
  const int per_split = (kchunks + splits - 1) / splits;
  const int kc_begin = blockIdx.y * per_split;
  const int kc_end = min(kchunks, kc_begin + per_split);
  if (kc_begin >= kc_end) return;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = lane >> 4;
  const int rc = lane & 15;
  const int nrow = n0 + wave * 16 + rc;  // this lane's W row (B col)

  // x chunk [MP][64] staged with 16-B slot XOR swizzle (slot ^= row&7)
  __shared__ __align__(16) unsigned short x_lds[MP][64];

  ps_gf32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; mt++) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  for (int kc = kc_begin; kc < kc_end; kc++) {
    // ---- cooperative x staging: MP*8 16-B slots over 256 threads ----
    for (int u = tid; u < MP * 8; u += 256) {
      const int row = u >> 3;
      const int slot = u & 7;
      ps_bf16x8 v = {};
      if (row < M)
        v = *(const ps_bf16x8*)(x + (long)row * x_stride + kc * 64 +
                                slot * 8);
      *(ps_bf16x8*)(&x_lds[row][(slot ^ (row & 7)) * 8]) = v;
    }
    __syncthreads();
    // ---- B-frags: W rows streamed once ----
    const unsigned short* wr = w + (long)nrow * K + kc * 64;
    ps_gbf16x8 b0 = ps_as_gbf16(*(const ps_bf16x8*)(wr + g * 8));
    ps_gbf16x8 b1 = ps_as_gbf16(*(const ps_bf16x8*)(wr + 32 + g * 8));
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < M_TILES; mt++) {
      const int row = mt * 16 + rc;
      ps_gbf16x8 a0 = ps_as_gbf16(
          *(const ps_bf16x8*)(&x_lds[row][(g ^ (row & 7)) * 8]));
      ps_gbf16x8 a1 = ps_as_gbf16(
          *(const ps_bf16x8*)(&x_lds[row][((4 + g) ^ (row & 7)) * 8]));
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[mt],
                                                        0, 0, 0);
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[mt],
                                                        0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // ---- epilogue (C layout: row=(g*4+r), col=rc of each 16x16 tile).
  // splits==1 writes bf16 output directly; otherwise each split streams
  // its fp32 partial to ws[split][M][N] and a combine kernel reduces —
  // device atomicAdd RMW traffic (M*N*splits) was the measured 1/M
  // throughput wall of the earlier design.
#pragma unroll
  for (int mt = 0; mt < M_TILES; mt++) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int m = mt * 16 + g * 4 + r;
      if (m >= M) continue;
      if (gridDim.y == 1) {
        out_bf16[(long)m * N + nrow] = ps_f32_to_bf16(acc[mt][r]);
      } else {
        ws[((long)blockIdx.y * M + m) * N + nrow] = acc[mt][r];
      }
    }
  }
}

__global__ __launch_bounds__(256) void skinny_combine_kernel(
    unsigned short* __restrict__ out,  // [M, N] bf16
    const float* __restrict__ ws,      // [S, M, N] fp32
    long MN, int splits) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= MN) return;
  float acc = 0.f;
  for (int s = 0; s < splits; s++) acc += ws[(long)s * MN + i];
  out[i] = ps_f32_to_bf16(acc);
}



extern "C" {

// ws == nullptr forces splits = 1. ps_skinny_gemm_splits reports the
// split depth the heuristic will pick for a shape so the caller can size
// the fp32 workspace.
int ps_skinny_gemm_splits(int M, int N, int K) {
  if (N % 64 != 0 || K % 64 != 0 || M > 256) return -1;
  const int kchunks = K / 64;
  const int target_wgs = 1024;
  int splits = target_wgs / (N / 64);
  if (splits < 1) splits = 1;
  if (splits > kchunks) splits = kchunks;
  return splits;
}

int ps_skinny_gemm(void* out_bf16, void* ws, const void* x, const void* w,
                   int M, int N, int K, long x_stride, hipStream_t stream) {
  int splits = ps_skinny_gemm_splits(M, N, K);
  if (splits < 0) return -1;
  if (ws == nullptr) splits = 1;
  dim3 grid(N / 64, splits);
  dim3 block(256);
#define PS_SG(MT)                                                           \
  skinny_gemm_kernel<MT><<<grid, block, 0, stream>>>(                       \
      (unsigned short*)out_bf16, (float*)ws, (const unsigned short*)x,      \
      (const unsigned short*)w, M, N, K, x_stride)
  if (M <= 16) PS_SG(1);
  else if (M <= 32) PS_SG(2);
  else if (M <= 64) PS_SG(4);
  else if (M <= 128) PS_SG(8);
  else PS_SG(16);
#undef PS_SG
  if (splits > 1) {
    const long MN = (long)M * N;
    skinny_combine_kernel<<<dim3((MN + 255) / 256), 256, 0, stream>>>(
        (unsigned short*)out_bf16, (const float*)ws, MN, splits);
  }
  return 0;
}

}  // extern "C"
