// CacheGen-style entropy codec for serialized KV blocks (CPU side).
//
// The KV offload/remote tier first quantizes KV rows to int8 with per-row
// scales (csrc/norm_act_rope.hip kv_quant — the GPU stage). This codec is
// the serde stage the reference delegates to LMCache's CacheGen encoder
// (north-star item "CacheGen encode/decode, KV quantize/serialize"):
// an adaptive range coder over the int8 symbols with per-channel contexts.
//
// Model: symbols are near-Laplacian around 0 with channel-dependent
// spread, so the coder keeps an adaptive 256-bin frequency table per
// channel context (channel & 63). Both sides adapt identically, so no
// tables travel in the blob — the stream is self-contained:
//   [u32 magic 'PSKV'][u32 n][payload...]
// Typical compression on real KV int8: 1.6-2.5x (measured in
// tests/test_kvpool.py::test_cachegen_codec_roundtrip_and_ratio).
#include <torch/extension.h>

#include <cstdint>
#include <cstring>
#include <vector>

namespace {

constexpr uint32_t kMagic = 0x50534b56;  // 'PSKV'
constexpr int kCtx = 64;                 // channel contexts
constexpr uint32_t kTop = 1u << 24;
constexpr uint32_t kBot = 1u << 16;

struct Freq {
  // adaptive frequency table with cumulative totals rebuilt lazily
  uint32_t f[256];
  uint32_t total;
  Freq() {
    for (auto& x : f) x = 1;
    total = 256;
  }
  void update(int s) {
    f[s] += 64;
    total += 64;
    if (total > kBot - 256) {  // halve to keep ranges renormalizable
      total = 0;
      for (auto& x : f) {
        x = (x >> 1) | 1;
        total += x;
      }
    }
  }
};

struct RangeEncoder {
  uint32_t low = 0, range = 0xFFFFFFFFu;
  std::vector<uint8_t> out;
  void encode(uint32_t cum, uint32_t freq, uint32_t tot) {
    range /= tot;
    low += cum * range;
    range *= freq;
    while ((low ^ (low + range)) < kTop ||
           (range < kBot && ((range = -low & (kBot - 1)), true))) {
      out.push_back((uint8_t)(low >> 24));
      low <<= 8;
      range <<= 8;
    }
  }
  void flush() {
    for (int i = 0; i < 4; i++) {
      out.push_back((uint8_t)(low >> 24));
      low <<= 8;
    }
  }
};

struct RangeDecoder {
  uint32_t low = 0, range = 0xFFFFFFFFu, code = 0;
  const uint8_t* in;
  size_t pos = 0, n;
  RangeDecoder(const uint8_t* p, size_t len) : in(p), n(len) {
    for (int i = 0; i < 4; i++) code = (code << 8) | next();
  }
  uint8_t next() { return pos < n ? in[pos++] : 0; }
  uint32_t decode_cum(uint32_t tot) {
    range /= tot;
    return (code - low) / range;
  }
  void decode_update(uint32_t cum, uint32_t freq) {
    low += cum * range;
    range *= freq;
    while ((low ^ (low + range)) < kTop ||
           (range < kBot && ((range = -low & (kBot - 1)), true))) {
      code = (code << 8) | next();
      low <<= 8;
      range <<= 8;
    }
  }
};

}  // namespace

at::Tensor cachegen_encode(at::Tensor q) {
  TORCH_CHECK(q.scalar_type() == at::kChar && q.is_cpu() &&
                  q.is_contiguous(),
              "cachegen_encode expects contiguous CPU int8");
  const int64_t n = q.numel();
  const int64_t hd = q.dim() >= 1 ? q.size(-1) : 1;
  const int8_t* src = q.data_ptr<int8_t>();
  std::vector<Freq> ctx(kCtx);
  RangeEncoder enc;
  enc.out.reserve((size_t)n / 2 + 16);
  for (int64_t i = 0; i < n; i++) {
    Freq& fr = ctx[(i % hd) & (kCtx - 1)];
    // delta along the token axis within a channel (CacheGen's key
    // observation: KV channels vary slowly across adjacent tokens, so
    // deltas concentrate near 0 and the adaptive model bites)
    const int prev = i >= hd ? src[i - hd] : 0;
    const int s = (uint8_t)(int8_t)(src[i] - prev);
    uint32_t cum = 0;
    for (int j = 0; j < s; j++) cum += fr.f[j];
    enc.encode(cum, fr.f[s], fr.total);
    fr.update(s);
  }
  enc.flush();
  at::Tensor out = at::empty({(int64_t)enc.out.size() + 12},
                             q.options().dtype(at::kByte));
  uint8_t* dst = out.data_ptr<uint8_t>();
  uint32_t magic = kMagic;
  uint64_t count = (uint64_t)n;
  memcpy(dst, &magic, 4);
  memcpy(dst + 4, &count, 8);
  memcpy(dst + 12, enc.out.data(), enc.out.size());
  return out;
}

at::Tensor cachegen_decode(at::Tensor blob, int64_t hd) {
  TORCH_CHECK(blob.scalar_type() == at::kByte && blob.is_cpu() &&
                  blob.is_contiguous() && blob.numel() >= 12,
              "cachegen_decode expects a PSKV byte blob");
  const uint8_t* src = blob.data_ptr<uint8_t>();
  uint32_t magic;
  uint64_t n;
  memcpy(&magic, src, 4);
  memcpy(&n, src + 4, 8);
  TORCH_CHECK(magic == kMagic, "bad PSKV magic");
  at::Tensor out = at::empty({(int64_t)n},
                             blob.options().dtype(at::kChar));
  int8_t* dst = out.data_ptr<int8_t>();
  std::vector<Freq> ctx(kCtx);
  RangeDecoder dec(src + 12, (size_t)blob.numel() - 12);
  for (int64_t i = 0; i < (int64_t)n; i++) {
    Freq& fr = ctx[(i % hd) & (kCtx - 1)];
    const uint32_t target = dec.decode_cum(fr.total);
    uint32_t cum = 0;
    int s = 0;
    while (s < 255 && cum + fr.f[s] <= target) cum += fr.f[s++];
    dec.decode_update(cum, fr.f[s]);
    fr.update(s);
    const int prev = i >= hd ? dst[i - hd] : 0;
    dst[i] = (int8_t)(uint8_t)(s + prev);
  }
  return out;
}
