// Fused normalisation / activation / rotary kernels for the MI355X engine.
//
// Parity targets (behavioural, not code): the external vLLM engine's
// rms_norm / fused_add_rms_norm / silu_and_mul / rotary_embedding op surface
// that /root/reference drives through its helm engine flags
// (see SURVEY.md section 2.8).  All kernels are memory-bound: the design
// point is HBM3E bandwidth (~6.3 TB/s achievable), so every bf16 access is a
// 16-byte ushort8 vector and each tensor element is touched exactly once
// where possible.
#include "ps_common.h"

// ---------------------------------------------------------------------------
// RMSNorm:  out[t, :] = x[t, :] * rsqrt(mean(x^2) + eps) * w
// One workgroup per token row; row kept in registers between the two passes.
// ---------------------------------------------------------------------------
template <int BLOCK, int MAX_VPT>  // MAX_VPT = max ushort8 vectors per thread
__global__ __launch_bounds__(BLOCK) void rms_norm_kernel(
    unsigned short* __restrict__ out,      // [T, D]
    const unsigned short* __restrict__ x,  // [T, D]
    const unsigned short* __restrict__ w,  // [D]
    float eps, int D) {
  const long t = blockIdx.x;
  const unsigned short* xr = x + t * (long)D;
  unsigned short* orow = out + t * (long)D;

  float vals[MAX_VPT][8];
  int nvec = 0;
  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 xv = *(const ps_bf16x8*)(xr + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = ps_bf16_to_f32(xv[j]);
      vals[nvec][j] = f;
      ss += f * f;
    }
    nvec++;
  }
  // reduce ss across the block
  __shared__ float red[BLOCK / 64];
  ss = ps_group_sum<64>(ss);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tot = 0.f;
#pragma unroll
    for (int wv = 0; wv < BLOCK / 64; wv++) tot += red[wv];
    red[0] = rsqrtf(tot / (float)D + eps);
  }
  __syncthreads();
  const float inv = red[0];

  nvec = 0;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 wv = *(const ps_bf16x8*)(w + i);
    ps_bf16x8 ov;
#pragma unroll
    for (int j = 0; j < 8; j++)
      ov[j] = ps_f32_to_bf16(vals[nvec][j] * inv * ps_bf16_to_f32(wv[j]));
    *(ps_bf16x8*)(orow + i) = ov;
    nvec++;
  }
}

// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm (the per-layer hot path):
//   residual[t,:] += x[t,:];  x[t,:] = rmsnorm(residual[t,:]) * w
// Keeps the summed row in registers: 2 reads + 2 writes per element.
// ---------------------------------------------------------------------------
template <int BLOCK, int MAX_VPT>
__global__ __launch_bounds__(BLOCK) void fused_add_rms_norm_kernel(
    unsigned short* __restrict__ x,         // [T, D] in: delta, out: normed
    unsigned short* __restrict__ residual,  // [T, D] in/out: accumulated
    const unsigned short* __restrict__ w,   // [D]
    float eps, int D) {
  const long t = blockIdx.x;
  unsigned short* xr = x + t * (long)D;
  unsigned short* rr = residual + t * (long)D;

  float vals[MAX_VPT][8];
  int nvec = 0;
  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 xv = *(const ps_bf16x8*)(xr + i);
    ps_bf16x8 rv = *(const ps_bf16x8*)(rr + i);
    ps_bf16x8 sv;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = ps_bf16_to_f32(xv[j]) + ps_bf16_to_f32(rv[j]);
      vals[nvec][j] = f;
      ss += f * f;
      sv[j] = ps_f32_to_bf16(f);
    }
    *(ps_bf16x8*)(rr + i) = sv;
    nvec++;
  }
  __shared__ float red[BLOCK / 64];
  ss = ps_group_sum<64>(ss);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tot = 0.f;
#pragma unroll
    for (int wv = 0; wv < BLOCK / 64; wv++) tot += red[wv];
    red[0] = rsqrtf(tot / (float)D + eps);
  }
  __syncthreads();
  const float inv = red[0];

  nvec = 0;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 wv = *(const ps_bf16x8*)(w + i);
    ps_bf16x8 ov;
#pragma unroll
    for (int j = 0; j < 8; j++)
      ov[j] = ps_f32_to_bf16(vals[nvec][j] * inv * ps_bf16_to_f32(wv[j]));
    *(ps_bf16x8*)(xr + i) = ov;
    nvec++;
  }
}

// ---------------------------------------------------------------------------
// SiLU-and-mul: out[t, d] = silu(x[t, d]) * x[t, D + d], D = out feature dim.
// ---------------------------------------------------------------------------
__global__ void silu_and_mul_kernel(unsigned short* __restrict__ out,
                                    const unsigned short* __restrict__ x,
                                    int D, long total /* = T * D / 8 */) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += stride) {
    const long elem = idx * 8;
    const long t = elem / D;
    const long d = elem % D;
    ps_bf16x8 g = *(const ps_bf16x8*)(x + t * 2 * (long)D + d);
    ps_bf16x8 u = *(const ps_bf16x8*)(x + t * 2 * (long)D + D + d);
    ps_bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = ps_bf16_to_f32(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = ps_f32_to_bf16(s * ps_bf16_to_f32(u[j]));
    }
    *(ps_bf16x8*)(out + elem) = o;
  }
}

// ---------------------------------------------------------------------------
// Rotary embedding, GPT-NeoX style (Llama): rotate (r, r + D/2) pairs in the
// leading rot_dim dims of every q / k head.  cos/sin are precomputed on the
// host (guide App. B: never evaluate trig on-device for RoPE).
// cos_sin layout: [max_pos, rot_dim]  (cos[rot/2] || sin[rot/2]), fp32.
// ---------------------------------------------------------------------------
__global__ void rope_kernel(const int* __restrict__ positions,  // [T]
                            unsigned short* __restrict__ q,     // [T, QH*HD]
                            unsigned short* __restrict__ k,     // [T, KH*HD]
                            const float* __restrict__ cos_sin,
                            int QH, int KH, int HD, int ROT) {
  const long t = blockIdx.x;
  const float* cs = cos_sin + (long)positions[t] * ROT;
  const int half = ROT / 2;
  const int total = (QH + KH) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int r = idx % half;
    unsigned short* base = (h < QH) ? (q + t * (long)QH * HD + (long)h * HD)
                                    : (k + t * (long)KH * HD + (long)(h - QH) * HD);
    const float c = cs[r];
    const float s = cs[half + r];
    const float x1 = ps_bf16_to_f32(base[r]);
    const float x2 = ps_bf16_to_f32(base[r + half]);
    base[r] = ps_f32_to_bf16(x1 * c - x2 * s);
    base[r + half] = ps_f32_to_bf16(x2 * c + x1 * s);
  }
}

// ---------------------------------------------------------------------------
// C launchers
// ---------------------------------------------------------------------------
extern "C" {

void ps_rms_norm(void* out, const void* x, const void* w, float eps, long T,
                 int D, hipStream_t stream) {
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)T);
  if (D <= BLOCK * 8 * 2)
    rms_norm_kernel<BLOCK, 2><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)out, (const unsigned short*)x,
        (const unsigned short*)w, eps, D);
  else if (D <= BLOCK * 8 * 4)
    rms_norm_kernel<BLOCK, 4><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)out, (const unsigned short*)x,
        (const unsigned short*)w, eps, D);
  else
    rms_norm_kernel<BLOCK, 16><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)out, (const unsigned short*)x,
        (const unsigned short*)w, eps, D);
}

void ps_fused_add_rms_norm(void* x, void* residual, const void* w, float eps,
                           long T, int D, hipStream_t stream) {
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)T);
  if (D <= BLOCK * 8 * 2)
    fused_add_rms_norm_kernel<BLOCK, 2><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)x, (unsigned short*)residual,
        (const unsigned short*)w, eps, D);
  else if (D <= BLOCK * 8 * 4)
    fused_add_rms_norm_kernel<BLOCK, 4><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)x, (unsigned short*)residual,
        (const unsigned short*)w, eps, D);
  else
    fused_add_rms_norm_kernel<BLOCK, 16><<<grid, BLOCK, 0, stream>>>(
        (unsigned short*)x, (unsigned short*)residual,
        (const unsigned short*)w, eps, D);
}

void ps_silu_and_mul(void* out, const void* x, long T, int D,
                     hipStream_t stream) {
  const long total = T * (long)D / 8;
  const int block = 256;
  const long want = (total + block - 1) / block;
  const int grid = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  silu_and_mul_kernel<<<grid, block, 0, stream>>>(
      (unsigned short*)out, (const unsigned short*)x, D, total);
}

void ps_rope(const void* positions, void* q, void* k, const void* cos_sin,
             long T, int QH, int KH, int HD, int ROT, hipStream_t stream) {
  const int block = 256;
  rope_kernel<<<dim3((unsigned)T), block, 0, stream>>>(
      (const int*)positions, (unsigned short*)q, (unsigned short*)k,
      (const float*)cos_sin, QH, KH, HD, ROT);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Fused RoPE + KV append operating IN-PLACE on the packed qkv tensor
// ([T, (QH+2*KH)*HD] with arbitrary row stride): rotates q and k, then
// scatters the rotated k and v rows into the paged cache. Replaces
// rope_kernel + reshape_and_cache_kernel + three .contiguous() copies.
// ---------------------------------------------------------------------------
template <typename KVT>
__global__ void fused_rope_cache_kernel(
    unsigned short* __restrict__ qkv, const int* __restrict__ positions,
    const float* __restrict__ cos_sin, const long* __restrict__ slot_mapping,
    KVT* __restrict__ k_cache, KVT* __restrict__ v_cache,
    int QH, int KH, int HD, int ROT, long qkv_stride, int BS) {
  using KVTr = ps_kv_traits<KVT>;
  using kvec8 = typename KVTr::vec8;
  const long t = blockIdx.x;
  unsigned short* base = qkv + t * qkv_stride;
  const float* cs = cos_sin + (long)positions[t] * ROT;
  const int half = ROT / 2;
  const int total = (QH + KH) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int r = idx % half;
    unsigned short* hp = base + (long)h * HD;  // q heads then k heads
    const float c = cs[r];
    const float s = cs[half + r];
    const float x1 = ps_bf16_to_f32(hp[r]);
    const float x2 = ps_bf16_to_f32(hp[r + half]);
    hp[r] = ps_f32_to_bf16(x1 * c - x2 * s);
    hp[r + half] = ps_f32_to_bf16(x2 * c + x1 * s);
  }
  __syncthreads();
  const long slot = slot_mapping[t];
  if (slot < 0) return;
  const long blk = slot / BS;
  const int off = (int)(slot % BS);
  const int kvelems = KH * HD;
  const unsigned short* ksrc = base + (long)QH * HD;
  const unsigned short* vsrc = ksrc + kvelems;
  for (int i = threadIdx.x * 8; i < 2 * kvelems; i += blockDim.x * 8) {
    const bool isv = i >= kvelems;
    const int r = isv ? i - kvelems : i;
    const int h = r / HD;
    const int d = r % HD;
    const long dst = ((blk * KH + h) * BS + off) * (long)HD + d;
    KVT* cache = isv ? v_cache : k_cache;
    const unsigned short* src = (isv ? vsrc : ksrc) + r;
    ps_bf16x8 sv = *(const ps_bf16x8*)src;
    kvec8 ov;
#pragma unroll
    for (int j = 0; j < 8; j++)
      ov[j] = KVTr::from_f32(ps_bf16_to_f32(sv[j]));
    *(kvec8*)(cache + dst) = ov;
  }
}

extern "C" void ps_fused_rope_cache(void* qkv, const void* positions,
                                    const void* cos_sin,
                                    const void* slot_mapping, void* k_cache,
                                    void* v_cache, long T, int QH, int KH,
                                    int HD, int ROT, long qkv_stride, int BS,
                                    int kv_fp8, hipStream_t stream) {
  if (kv_fp8)
    fused_rope_cache_kernel<unsigned char>
        <<<dim3((unsigned)T), 256, 0, stream>>>(
            (unsigned short*)qkv, (const int*)positions,
            (const float*)cos_sin, (const long*)slot_mapping,
            (unsigned char*)k_cache, (unsigned char*)v_cache, QH, KH, HD,
            ROT, qkv_stride, BS);
  else
    fused_rope_cache_kernel<unsigned short>
        <<<dim3((unsigned)T), 256, 0, stream>>>(
            (unsigned short*)qkv, (const int*)positions,
            (const float*)cos_sin, (const long*)slot_mapping,
            (unsigned short*)k_cache, (unsigned short*)v_cache, QH, KH, HD,
            ROT, qkv_stride, BS);
}

// ---------------------------------------------------------------------------
// Row-wise int8 KV quantization (CacheGen-style serde for the offload /
// transfer tiers): each row of `hd` elements gets one fp32 scale
// (max|x| / 127). Halves PCIe/host-pool traffic vs bf16 at ~0.4% RMS error.
// ---------------------------------------------------------------------------
__global__ void kv_quant_kernel(signed char* __restrict__ out,
                                float* __restrict__ scales,
                                const unsigned short* __restrict__ in,
                                int hd, long rows) {
  const long r = (long)blockIdx.x * blockDim.y + threadIdx.y;
  if (r >= rows) return;
  const unsigned short* row = in + r * hd;
  signed char* orow = out + r * hd;
  const int lane = threadIdx.x;  // 64 lanes per row
  float amax = 0.f;
  for (int i = lane * 8; i < hd; i += 64 * 8) {
    ps_bf16x8 v = *(const ps_bf16x8*)(row + i);
#pragma unroll
    for (int j = 0; j < 8; j++) amax = fmaxf(amax, fabsf(ps_bf16_to_f32(v[j])));
  }
  amax = ps_group_max<64>(amax);
  const float scale = amax > 0.f ? amax / 127.f : 1.f;
  const float inv = 1.f / scale;
  if (lane == 0) scales[r] = scale;
  for (int i = lane * 8; i < hd; i += 64 * 8) {
    ps_bf16x8 v = *(const ps_bf16x8*)(row + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = ps_bf16_to_f32(v[j]) * inv;
      orow[i + j] = (signed char)lrintf(fminf(fmaxf(f, -127.f), 127.f));
    }
  }
}

__global__ void kv_dequant_kernel(unsigned short* __restrict__ out,
                                  const signed char* __restrict__ in,
                                  const float* __restrict__ scales,
                                  int hd, long rows) {
  const long r = (long)blockIdx.x * blockDim.y + threadIdx.y;
  if (r >= rows) return;
  const signed char* row = in + r * hd;
  unsigned short* orow = out + r * hd;
  const float scale = scales[r];
  for (int i = threadIdx.x * 8; i < hd; i += 64 * 8) {
    ps_bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; j++)
      v[j] = ps_f32_to_bf16((float)row[i + j] * scale);
    *(ps_bf16x8*)(orow + i) = v;
  }
}

extern "C" {
void ps_kv_quant(void* out, void* scales, const void* in, long rows, int hd,
                 hipStream_t stream) {
  dim3 block(64, 4);
  dim3 grid((unsigned)((rows + 3) / 4));
  kv_quant_kernel<<<grid, block, 0, stream>>>(
      (signed char*)out, (float*)scales, (const unsigned short*)in, hd, rows);
}
void ps_kv_dequant(void* out, const void* in, const void* scales, long rows,
                   int hd, hipStream_t stream) {
  dim3 block(64, 4);
  dim3 grid((unsigned)((rows + 3) / 4));
  kv_dequant_kernel<<<grid, block, 0, stream>>>(
      (unsigned short*)out, (const signed char*)in, (const float*)scales, hd,
      rows);
}
}

// ---------------------------------------------------------------------------
// fp8 activation-quantizing variants (fuse the dynamic
// activation quantization into the producing kernel instead of a separate
// amax + cast pass). Each emits OCP fp8-e4m3 rows plus a per-ROW scale
// (amax/448) for rowwise-scaled fp8 GEMMs. Three phases per row: reduce
// sum-of-squares; compute normed values in registers + reduce row amax;
// quantize + store fp8.
// ---------------------------------------------------------------------------
template <int BLOCK, int MAX_VPT, bool FUSED_ADD>
__global__ __launch_bounds__(BLOCK) void rms_norm_fp8_kernel(
    unsigned char* __restrict__ out_q,      // [T, D] fp8
    float* __restrict__ out_scale,          // [T]
    unsigned short* __restrict__ x,         // [T, D] (delta when fused)
    unsigned short* __restrict__ residual,  // [T, D] in/out (fused only)
    const unsigned short* __restrict__ w,   // [D]
    float eps, int D) {
  const long t = blockIdx.x;
  unsigned short* xr = x + t * (long)D;
  unsigned short* rr = FUSED_ADD ? residual + t * (long)D : nullptr;
  unsigned char* orow = out_q + t * (long)D;

  float vals[MAX_VPT][8];
  int nvec = 0;
  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 xv = *(const ps_bf16x8*)(xr + i);
    ps_bf16x8 sv;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = ps_bf16_to_f32(xv[j]);
      if (FUSED_ADD) {
        f += ps_bf16_to_f32(((const ps_bf16x8*)(rr + i))[0][j]);
        sv[j] = ps_f32_to_bf16(f);
      }
      vals[nvec][j] = f;
      ss += f * f;
    }
    if (FUSED_ADD) *(ps_bf16x8*)(rr + i) = sv;
    nvec++;
  }
  __shared__ float red[BLOCK / 64];
  ss = ps_group_sum<64>(ss);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tot = 0.f;
#pragma unroll
    for (int wv = 0; wv < BLOCK / 64; wv++) tot += red[wv];
    red[0] = rsqrtf(tot / (float)D + eps);
  }
  __syncthreads();
  const float inv = red[0];

  // normed values + row amax
  float amax = 1e-6f;
  nvec = 0;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 wv = *(const ps_bf16x8*)(w + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = vals[nvec][j] * inv * ps_bf16_to_f32(wv[j]);
      vals[nvec][j] = f;
      amax = fmaxf(amax, fabsf(f));
    }
    nvec++;
  }
  __shared__ float reda[BLOCK / 64];
  amax = ps_group_max<64>(amax);
  if ((threadIdx.x & 63) == 0) reda[threadIdx.x >> 6] = amax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = 0.f;
#pragma unroll
    for (int wv = 0; wv < BLOCK / 64; wv++) m = fmaxf(m, reda[wv]);
    reda[0] = m / 448.0f;
    out_scale[t] = m / 448.0f;
  }
  __syncthreads();
  const float rs = 1.0f / reda[0];

  nvec = 0;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_fp8x8 qv;
#pragma unroll
    for (int j = 0; j < 8; j++)
      qv[j] = ps_f32_to_fp8(vals[nvec][j] * rs);
    *(ps_fp8x8*)(orow + i) = qv;
    nvec++;
  }
}

template <int BLOCK, int MAX_VPT>
__global__ __launch_bounds__(BLOCK) void silu_and_mul_fp8_kernel(
    unsigned char* __restrict__ out_q,  // [T, D] fp8
    float* __restrict__ out_scale,      // [T]
    const unsigned short* __restrict__ x,  // [T, 2D]
    int D) {
  const long t = blockIdx.x;
  const unsigned short* g = x + t * (long)(2 * D);
  const unsigned short* u = g + D;
  unsigned char* orow = out_q + t * (long)D;

  float vals[MAX_VPT][8];
  int nvec = 0;
  float amax = 1e-6f;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_bf16x8 gv = *(const ps_bf16x8*)(g + i);
    ps_bf16x8 uv = *(const ps_bf16x8*)(u + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float a = ps_bf16_to_f32(gv[j]);
      float f = a / (1.0f + __expf(-a)) * ps_bf16_to_f32(uv[j]);
      vals[nvec][j] = f;
      amax = fmaxf(amax, fabsf(f));
    }
    nvec++;
  }
  __shared__ float reda[BLOCK / 64];
  amax = ps_group_max<64>(amax);
  if ((threadIdx.x & 63) == 0) reda[threadIdx.x >> 6] = amax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = 0.f;
#pragma unroll
    for (int wv = 0; wv < BLOCK / 64; wv++) m = fmaxf(m, reda[wv]);
    reda[0] = m / 448.0f;
    out_scale[t] = m / 448.0f;
  }
  __syncthreads();
  const float rs = 1.0f / reda[0];
  nvec = 0;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
    ps_fp8x8 qv;
#pragma unroll
    for (int j = 0; j < 8; j++)
      qv[j] = ps_f32_to_fp8(vals[nvec][j] * rs);
    *(ps_fp8x8*)(orow + i) = qv;
    nvec++;
  }
}

extern "C" {

void ps_rms_norm_fp8(void* out_q, void* out_scale, void* x, void* residual,
                     const void* w, float eps, long T, int D, int fused,
                     hipStream_t stream) {
  dim3 grid((unsigned)T);
#define PS_RNF(B, V, F)                                                   \
  rms_norm_fp8_kernel<B, V, F><<<grid, B, 0, stream>>>(                    \
      (unsigned char*)out_q, (float*)out_scale, (unsigned short*)x,       \
      (unsigned short*)residual, (const unsigned short*)w, eps, D)
  if (fused) {
    if (D <= 4096) PS_RNF(512, 1, true);
    else PS_RNF(1024, 2, true);
  } else {
    if (D <= 4096) PS_RNF(512, 1, false);
    else PS_RNF(1024, 2, false);
  }
#undef PS_RNF
}

void ps_silu_and_mul_fp8(void* out_q, void* out_scale, const void* x, long T,
                         int D, hipStream_t stream) {
  dim3 grid((unsigned)T);
  if (D <= 4096)
    silu_and_mul_fp8_kernel<512, 1><<<grid, 512, 0, stream>>>(
        (unsigned char*)out_q, (float*)out_scale,
        (const unsigned short*)x, D);
  else
    silu_and_mul_fp8_kernel<1024, 4><<<grid, 1024, 0, stream>>>(
        (unsigned char*)out_q, (float*)out_scale,
        (const unsigned short*)x, D);
}

}  // extern "C"
