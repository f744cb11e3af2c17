// PyTorch bindings for the MI355X serving kernels (production_stack_amd._C).
//
// The kernels themselves live in *.hip translation units behind a plain C
// ABI (raw pointers + hipStream_t); this file only validates tensors and
// forwards to them on the current HIP stream.
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <algorithm>

extern "C" {
void ps_rms_norm(void* out, const void* x, const void* w, float eps, long T,
                 int D, hipStream_t stream);
void ps_fused_add_rms_norm(void* x, void* residual, const void* w, float eps,
                           long T, int D, hipStream_t stream);
void ps_rms_norm_fp8(void* out_q, void* out_scale, void* x, void* residual,
                     const void* w, float eps, long T, int D, int fused,
                     hipStream_t stream);
void ps_silu_and_mul_fp8(void* out_q, void* out_scale, const void* x, long T,
                         int D, hipStream_t stream);
void ps_silu_and_mul(void* out, const void* x, long T, int D,
                     hipStream_t stream);
void ps_rope(const void* positions, void* q, void* k, const void* cos_sin,
             long T, int QH, int KH, int HD, int ROT, hipStream_t stream);
void ps_fused_rope_cache(void* qkv, const void* positions, const void* cos_sin,
                         const void* slot_mapping, void* k_cache,
                         void* v_cache, long T, int QH, int KH, int HD,
                         int ROT, long qkv_stride, int BS, int kv_fp8,
                         hipStream_t stream);
int ps_paged_attn_decode(void* out, void* ws_acc, void* ws_ml, const void* q,
                         const void* k_cache, const void* v_cache,
                         const void* block_tables, const void* seq_lens,
                         int num_seqs, int max_blocks, float scale, int KH,
                         int GQ, int head_dim, int block_size, int num_splits,
                         long q_stride, int variant, int kv_fp8, int window,
                         hipStream_t stream);
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
                               int variant, int kv_fp8, int window, hipStream_t stream);
int ps_paged_attn_prefill(void* out, const void* q, const void* k_cache,
                          const void* v_cache, const void* block_tables,
                          const void* token_seq, const void* token_pos,
                          int num_tokens, int num_q_heads, int max_blocks,
                          float scale, int KH, int GQ, int head_dim,
                          int block_size, long q_stride, int window,
                          hipStream_t stream);
void ps_reshape_and_cache(const void* k, const void* v, void* k_cache,
                          void* v_cache, const void* slot_mapping, long T,
                          int KH, int HD, int BS, int kv_fp8,
                          hipStream_t stream);
void ps_greedy_sample(void* out, const void* logits, long R, int V,
                      hipStream_t stream);
void ps_kv_quant(void* out, void* scales, const void* in, long rows, int hd,
                 hipStream_t stream);
void ps_kv_dequant(void* out, const void* in, const void* scales, long rows,
                   int hd, hipStream_t stream);
int ps_skinny_gemm_splits(int M, int N, int K);
int ps_skinny_gemm(void* out_bf16, void* ws, const void* x, const void* w,
                   int M, int N, int K, long x_stride, hipStream_t stream);
int ps_gemm8p_splits(int M, int N, int K);
int ps_gemm8p(void* out_bf16, void* ws, const void* x, const void* w, int M,
              int N, int K, long x_stride, hipStream_t stream);
int ps_lora_bgmv(void* out, const void* x, const void* A, const void* B,
                 const void* scale, const void* idx, int T, int IN, int W,
                 int R, long out_stride, long x_stride, int col_off,
                 hipStream_t stream);
}

at::Tensor cachegen_encode(at::Tensor q);
at::Tensor cachegen_decode(at::Tensor blob, int64_t hd);

namespace {

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_GPU_BF16(t)                                             \
  TORCH_CHECK((t).is_cuda(), #t " must be on the GPU");               \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

#define CHECK_GPU_DTYPE(t, ty)                                  \
  TORCH_CHECK((t).is_cuda(), #t " must be on the GPU");         \
  TORCH_CHECK((t).scalar_type() == (ty), #t " dtype mismatch"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// KV caches are bf16 or OCP fp8 e4m3; returns 1 for fp8
int cache_fp8(const at::Tensor& c) {
  TORCH_CHECK(c.is_cuda() && c.is_contiguous(), "cache must be GPU-contig");
  if (c.scalar_type() == at::kBFloat16) return 0;
  if (c.scalar_type() == at::kFloat8_e4m3fn) return 1;
  TORCH_CHECK(false, "KV cache must be bf16 or float8_e4m3fn");
  return 0;
}

// q may be a row-strided view into the packed qkv tensor
long q_row_stride(const at::Tensor& q, int HD) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "q must be bf16 on GPU");
  TORCH_CHECK(q.dim() == 3 && q.stride(2) == 1 && q.stride(1) == HD,
              "q must be [T, H, HD] with contiguous heads");
  return q.stride(0);
}

void rms_norm(at::Tensor out, at::Tensor x, at::Tensor w, double eps) {
  CHECK_GPU_BF16(out);
  CHECK_GPU_BF16(x);
  CHECK_GPU_BF16(w);
  const long T = x.numel() / x.size(-1);
  const int D = (int)x.size(-1);
  TORCH_CHECK(D % 8 == 0, "hidden dim must be a multiple of 8");
  ps_rms_norm(out.data_ptr(), x.data_ptr(), w.data_ptr(), (float)eps, T, D,
              current_stream());
}

void fused_add_rms_norm(at::Tensor x, at::Tensor residual, at::Tensor w,
                        double eps) {
  CHECK_GPU_BF16(x);
  CHECK_GPU_BF16(residual);
  CHECK_GPU_BF16(w);
  const long T = x.numel() / x.size(-1);
  const int D = (int)x.size(-1);
  TORCH_CHECK(D % 8 == 0, "hidden dim must be a multiple of 8");
  ps_fused_add_rms_norm(x.data_ptr(), residual.data_ptr(), w.data_ptr(),
                        (float)eps, T, D, current_stream());
}

void silu_and_mul(at::Tensor out, at::Tensor x) {
  CHECK_GPU_BF16(out);
  CHECK_GPU_BF16(x);
  const int D = (int)out.size(-1);
  TORCH_CHECK(x.size(-1) == 2 * D, "x last dim must be 2*out last dim");
  TORCH_CHECK(D % 8 == 0, "feature dim must be a multiple of 8");
  const long T = out.numel() / D;
  ps_silu_and_mul(out.data_ptr(), x.data_ptr(), T, D, current_stream());
}

void rotary_embedding(at::Tensor positions, at::Tensor q, at::Tensor k,
                      at::Tensor cos_sin, long head_dim) {
  CHECK_GPU_DTYPE(positions, at::kInt);
  CHECK_GPU_BF16(q);
  CHECK_GPU_BF16(k);
  CHECK_GPU_DTYPE(cos_sin, at::kFloat);
  const long T = positions.size(0);
  const int HD = (int)head_dim;
  const int ROT = (int)cos_sin.size(-1);
  const int QH = (int)(q.numel() / T / HD);
  const int KH = (int)(k.numel() / T / HD);
  ps_rope(positions.data_ptr(), q.data_ptr(), k.data_ptr(),
          cos_sin.data_ptr(), T, QH, KH, HD, ROT, current_stream());
}

void paged_attn_decode(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor block_tables,
                       at::Tensor seq_lens, double scale, int64_t num_splits,
                       int64_t variant, int64_t window) {
  CHECK_GPU_BF16(out);
  const int kv_fp8 = cache_fp8(k_cache);
  cache_fp8(v_cache);
  CHECK_GPU_DTYPE(block_tables, at::kInt);
  CHECK_GPU_DTYPE(seq_lens, at::kInt);
  const int S = (int)q.size(0);
  const int QH = (int)q.size(1);
  const int HD = (int)q.size(2);
  const int KH = (int)k_cache.size(1);
  const int BS = (int)k_cache.size(2);
  TORCH_CHECK(QH % KH == 0, "GQA group mismatch");
  const int GQ = QH / KH;
  const int max_blocks = (int)block_tables.size(1);
  if (num_splits <= 0) {
    // fill ~3 workgroups per CU, bounded by useful split granularity
    const int target = 2048;
    num_splits = std::min<int64_t>(
        32, std::max<int64_t>(1, target / std::max(1, S * KH)));
  }
  at::Tensor ws_acc, ws_ml;
  void *acc_p = nullptr, *ml_p = nullptr;
  if (num_splits > 1) {
    auto opts = q.options().dtype(at::kFloat);
    ws_acc = at::empty({(long)S * QH * num_splits * HD}, opts);
    ws_ml = at::empty({(long)S * QH * num_splits * 2}, opts);
    acc_p = ws_acc.data_ptr();
    ml_p = ws_ml.data_ptr();
  }
  int rc = ps_paged_attn_decode(
      out.data_ptr(), acc_p, ml_p, q.data_ptr(), k_cache.data_ptr(),
      v_cache.data_ptr(), block_tables.data_ptr(), seq_lens.data_ptr(), S,
      max_blocks, (float)scale, KH, GQ, HD, BS, (int)num_splits,
      q_row_stride(q, HD), (int)variant, kv_fp8, (int)window,
      current_stream());
  TORCH_CHECK(rc == 0, "unsupported decode config: head_dim=", HD,
              " block_size=", BS, " gqa=", GQ);
}

void paged_attn_prefill(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                        at::Tensor v_cache, at::Tensor block_tables,
                        at::Tensor token_seq, at::Tensor token_pos,
                        double scale, int64_t window) {
  CHECK_GPU_BF16(out);
  CHECK_GPU_BF16(k_cache);
  CHECK_GPU_BF16(v_cache);
  CHECK_GPU_DTYPE(block_tables, at::kInt);
  CHECK_GPU_DTYPE(token_seq, at::kInt);
  CHECK_GPU_DTYPE(token_pos, at::kInt);
  const int T = (int)q.size(0);
  const int QH = (int)q.size(1);
  const int HD = (int)q.size(2);
  const int KH = (int)k_cache.size(1);
  const int BS = (int)k_cache.size(2);
  TORCH_CHECK(QH % KH == 0, "GQA group mismatch");
  const int GQ = QH / KH;
  const int max_blocks = (int)block_tables.size(1);
  int rc = ps_paged_attn_prefill(
      out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr(), token_seq.data_ptr(), token_pos.data_ptr(), T,
      QH, max_blocks, (float)scale, KH, GQ, HD, BS, q_row_stride(q, HD),
      (int)window, current_stream());
  TORCH_CHECK(rc == 0, "unsupported prefill config: head_dim=", HD,
              " block_size=", BS);
}

void paged_attn_prefill_mfma(at::Tensor out, at::Tensor q,
                             at::Tensor k_cache, at::Tensor v_cache,
                             at::Tensor block_tables, at::Tensor tile_info,
                             double scale, int64_t variant, int64_t window) {
  CHECK_GPU_BF16(out);
  const int kv_fp8 = cache_fp8(k_cache);
  cache_fp8(v_cache);
  CHECK_GPU_DTYPE(block_tables, at::kInt);
  CHECK_GPU_DTYPE(tile_info, at::kInt);
  const int QH = (int)q.size(1);
  const int HD = (int)q.size(2);
  const int KH = (int)k_cache.size(1);
  TORCH_CHECK(QH % KH == 0, "GQA group mismatch");
  TORCH_CHECK(k_cache.size(2) == 16, "mfma prefill needs block_size 16");
  const int GQ = QH / KH;
  const int NT = (int)tile_info.size(0);
  int rc;
  if (variant == 5) {
    // v5 with split-KV: allocate fp32 partial workspace when the launch
    // is small enough that the splits heuristic fires
    const int splits = ps_prefill_mfma32_splits(NT, KH, QH / KH);
    const long T = q.size(0);
    void *ws_o_p = nullptr, *ws_ml_p = nullptr;
    at::Tensor ws_o, ws_ml;
    if (splits > 1) {
      ws_o = at::empty({(long)splits * T * QH * HD},
                       q.options().dtype(at::kFloat));
      ws_ml = at::empty({2L * splits * T * QH},
                        q.options().dtype(at::kFloat));
      ws_o_p = ws_o.data_ptr();
      ws_ml_p = ws_ml.data_ptr();
    }
    rc = ps_paged_attn_prefill_mfma32(
        out.data_ptr(), ws_o_p, ws_ml_p, T, q.data_ptr(),
        k_cache.data_ptr(), v_cache.data_ptr(), block_tables.data_ptr(),
        tile_info.data_ptr(), NT, QH, (int)block_tables.size(1),
        (float)scale, KH, QH / KH, HD, q_row_stride(q, HD), kv_fp8,
        (int)window, current_stream());
  } else {
    rc = ps_paged_attn_prefill_mfma(
        out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
        v_cache.data_ptr(), block_tables.data_ptr(), tile_info.data_ptr(),
        NT, QH, (int)block_tables.size(1), (float)scale, KH, GQ, HD,
        q_row_stride(q, HD), (int)variant, kv_fp8, (int)window,
        current_stream());
  }
  TORCH_CHECK(rc == 0, "unsupported mfma prefill config: head_dim=", HD);
}

void fused_rope_cache(at::Tensor qkv, at::Tensor positions,
                      at::Tensor cos_sin, at::Tensor slot_mapping,
                      at::Tensor k_cache, at::Tensor v_cache,
                      int64_t q_heads, int64_t head_dim) {
  CHECK_GPU_BF16(qkv);
  CHECK_GPU_DTYPE(positions, at::kInt);
  CHECK_GPU_DTYPE(cos_sin, at::kFloat);
  CHECK_GPU_DTYPE(slot_mapping, at::kLong);
  const int kv_fp8 = cache_fp8(k_cache);
  cache_fp8(v_cache);
  const long T = positions.size(0);
  const int KH = (int)k_cache.size(1);
  const int BS = (int)k_cache.size(2);
  const int HD = (int)head_dim;
  const int ROT = (int)cos_sin.size(-1);
  TORCH_CHECK(
      qkv.size(-1) == (q_heads + 2 * KH) * HD,
      "qkv width mismatch");
  ps_fused_rope_cache(qkv.data_ptr(), positions.data_ptr(),
                      cos_sin.data_ptr(), slot_mapping.data_ptr(),
                      k_cache.data_ptr(), v_cache.data_ptr(), T,
                      (int)q_heads, KH, HD, ROT, qkv.stride(0), BS, kv_fp8,
                      current_stream());
}

void reshape_and_cache(at::Tensor k, at::Tensor v, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slot_mapping) {
  CHECK_GPU_BF16(k);
  CHECK_GPU_BF16(v);
  const int kv_fp8 = cache_fp8(k_cache);
  cache_fp8(v_cache);
  CHECK_GPU_DTYPE(slot_mapping, at::kLong);
  const long T = slot_mapping.size(0);
  const int KH = (int)k_cache.size(1);
  const int BS = (int)k_cache.size(2);
  const int HD = (int)k_cache.size(3);
  TORCH_CHECK(k.numel() == T * KH * HD, "k shape mismatch");
  ps_reshape_and_cache(k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
                       v_cache.data_ptr(), slot_mapping.data_ptr(), T, KH, HD,
                       BS, kv_fp8, current_stream());
}

void skinny_gemm(at::Tensor out, at::Tensor x, at::Tensor w) {
  CHECK_GPU_BF16(out);
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "x must be bf16");
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1, "x must be 2D row-major");
  CHECK_GPU_BF16(w);
  const int M = (int)x.size(0);
  const int K = (int)x.size(1);
  const int N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  const int splits = ps_skinny_gemm_splits(M, N, K);
  TORCH_CHECK(splits > 0, "unsupported skinny gemm shape M=", M, " N=", N,
              " K=", K);
  void* ws_p = nullptr;
  at::Tensor ws;
  if (splits > 1) {
    ws = at::empty({(long)splits * M * N},
                   x.options().dtype(at::kFloat));
    ws_p = ws.data_ptr();
  }
  int rc = ps_skinny_gemm(out.data_ptr(), ws_p, x.data_ptr(), w.data_ptr(),
                          M, N, K, x.stride(0), current_stream());
  TORCH_CHECK(rc == 0, "skinny gemm launch failed");
}

void rms_norm_fp8(at::Tensor out_q, at::Tensor scales, at::Tensor x,
                  at::Tensor residual, at::Tensor w, double eps,
                  int64_t fused) {
  CHECK_GPU_BF16(x);
  CHECK_GPU_BF16(w);
  CHECK_GPU_DTYPE(scales, at::kFloat);
  TORCH_CHECK(out_q.scalar_type() == at::kFloat8_e4m3fn ||
                  out_q.scalar_type() == at::kByte,
              "out_q must be fp8/byte");
  const long T = x.size(0);
  const int D = (int)x.size(1);
  ps_rms_norm_fp8(out_q.data_ptr(), scales.data_ptr(), x.data_ptr(),
                  fused ? residual.data_ptr() : nullptr, w.data_ptr(),
                  (float)eps, T, D, (int)fused, current_stream());
}

void silu_and_mul_fp8(at::Tensor out_q, at::Tensor scales, at::Tensor x) {
  CHECK_GPU_BF16(x);
  CHECK_GPU_DTYPE(scales, at::kFloat);
  const long T = x.size(0);
  const int D = (int)x.size(1) / 2;
  ps_silu_and_mul_fp8(out_q.data_ptr(), scales.data_ptr(), x.data_ptr(), T,
                      D, current_stream());
}

void gemm8p(at::Tensor out, at::Tensor x, at::Tensor w) {
  CHECK_GPU_BF16(out);
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "x must be bf16");
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1, "x must be 2D row-major");
  CHECK_GPU_BF16(w);
  const int M = (int)x.size(0);
  const int K = (int)x.size(1);
  const int N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  const int splits = ps_gemm8p_splits(M, N, K);
  TORCH_CHECK(splits > 0, "unsupported gemm8p shape M=", M, " N=", N,
              " K=", K);
  void* ws_p = nullptr;
  at::Tensor ws;
  if (splits > 1) {
    ws = at::empty({(long)splits * M * N}, x.options().dtype(at::kFloat));
    ws_p = ws.data_ptr();
  }
  int rc = ps_gemm8p(out.data_ptr(), ws_p, x.data_ptr(), w.data_ptr(), M, N,
                     K, x.stride(0), current_stream());
  TORCH_CHECK(rc == 0, "gemm8p launch failed");
}

void lora_bgmv(at::Tensor out, at::Tensor x, at::Tensor A, at::Tensor B,
               at::Tensor scale, at::Tensor idx, long col_off) {
  CHECK_GPU_BF16(out);
  CHECK_GPU_BF16(x);
  CHECK_GPU_BF16(A);
  CHECK_GPU_BF16(B);
  CHECK_GPU_DTYPE(scale, at::kFloat);
  CHECK_GPU_DTYPE(idx, at::kInt);
  TORCH_CHECK(out.dim() == 2 && out.stride(1) == 1, "out 2D row-major");
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1, "x 2D row-major");
  TORCH_CHECK(A.dim() == 3 && A.is_contiguous(), "A [S,R,IN] contiguous");
  TORCH_CHECK(B.dim() == 3 && B.is_contiguous(), "B [S,W,R] contiguous");
  const int T = (int)x.size(0);
  const int IN = (int)x.size(1);
  const int R = (int)A.size(1);
  const int W = (int)B.size(1);
  TORCH_CHECK(A.size(2) == IN, "A IN mismatch");
  TORCH_CHECK(B.size(2) == R, "B R mismatch");
  TORCH_CHECK(out.size(0) == T && idx.numel() == T, "T mismatch");
  TORCH_CHECK(col_off + W <= out.size(1), "col range out of bounds");
  int rc = ps_lora_bgmv(out.data_ptr(), x.data_ptr(), A.data_ptr(),
                        B.data_ptr(), scale.data_ptr(), idx.data_ptr(), T,
                        IN, W, R, out.stride(0), x.stride(0), (int)col_off,
                        current_stream());
  TORCH_CHECK(rc == 0, "lora_bgmv unsupported (R>64?) R=", R);
}

void kv_quant(at::Tensor out, at::Tensor scales, at::Tensor in) {
  CHECK_GPU_DTYPE(out, at::kChar);
  CHECK_GPU_DTYPE(scales, at::kFloat);
  CHECK_GPU_BF16(in);
  const int hd = (int)in.size(-1);
  TORCH_CHECK(hd % 8 == 0, "row width must be multiple of 8");
  const long rows = in.numel() / hd;
  ps_kv_quant(out.data_ptr(), scales.data_ptr(), in.data_ptr(), rows, hd,
              current_stream());
}

void kv_dequant(at::Tensor out, at::Tensor in, at::Tensor scales) {
  CHECK_GPU_BF16(out);
  CHECK_GPU_DTYPE(in, at::kChar);
  CHECK_GPU_DTYPE(scales, at::kFloat);
  const int hd = (int)out.size(-1);
  const long rows = out.numel() / hd;
  ps_kv_dequant(out.data_ptr(), in.data_ptr(), scales.data_ptr(), rows, hd,
                current_stream());
}

void greedy_sample(at::Tensor out, at::Tensor logits) {
  CHECK_GPU_DTYPE(out, at::kLong);
  CHECK_GPU_BF16(logits);
  const long R = logits.size(0);
  const int V = (int)logits.size(1);
  ps_greedy_sample(out.data_ptr(), logits.data_ptr(), R, V, current_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm,
        "Fused residual add + RMSNorm (bf16, in-place)");
  m.def("silu_and_mul", &silu_and_mul, "SiLU-and-mul (bf16)");
  m.def("rotary_embedding", &rotary_embedding,
        "Neox-style rotary embedding (bf16, in-place)");
  m.def("paged_attn_decode", &paged_attn_decode,
        "Paged attention, decode phase (bf16 KV, split-KV)",
        pybind11::arg("out"), pybind11::arg("q"), pybind11::arg("k_cache"),
        pybind11::arg("v_cache"), pybind11::arg("block_tables"),
        pybind11::arg("seq_lens"), pybind11::arg("scale"),
        pybind11::arg("num_splits") = 0, pybind11::arg("variant") = 0,
        pybind11::arg("window") = 0);
  m.def("paged_attn_prefill", &paged_attn_prefill,
        "Paged attention, chunked prefill (bf16 KV)");
  m.def("lora_bgmv", &lora_bgmv,
        "batched-grouped LoRA matvec (per-token adapter slots)");
  m.def("paged_attn_prefill_mfma", &paged_attn_prefill_mfma,
        "Paged attention, chunked prefill via MFMA tiles (head_dim 128)",
        pybind11::arg("out"), pybind11::arg("q"), pybind11::arg("k_cache"),
        pybind11::arg("v_cache"), pybind11::arg("block_tables"),
        pybind11::arg("tile_info"), pybind11::arg("scale"),
        pybind11::arg("variant") = 4, pybind11::arg("window") = 0);
  m.def("fused_rope_cache", &fused_rope_cache,
        "Fused RoPE + paged KV append on the packed qkv tensor");
  m.def("reshape_and_cache", &reshape_and_cache,
        "Append K/V for new tokens into the paged cache");
  m.def("greedy_sample", &greedy_sample, "Per-row argmax over vocab");
  m.def("kv_quant", &kv_quant, "Row-wise int8 KV quantization");
  m.def("rms_norm_fp8", &rms_norm_fp8,
        "RMSNorm (optionally fused residual add) emitting fp8 + row scales");
  m.def("silu_and_mul_fp8", &silu_and_mul_fp8,
        "SiLU-mul emitting fp8 + row scales");
  m.def("cachegen_encode", &cachegen_encode,
        "CacheGen-style adaptive range encode of int8 KV (CPU)");
  m.def("cachegen_decode", &cachegen_decode,
        "CacheGen-style range decode -> int8 (CPU)");
  m.def("gemm8p", &gemm8p,
        "8-phase deep-pipelined bf16 GEMM (out = x @ w.T)");
  m.def("skinny_gemm", &skinny_gemm,
        "Split-K MFMA GEMM for decode-shaped (M<=128) projections");
  m.def("kv_dequant", &kv_dequant, "Row-wise int8 KV dequantization");
}
