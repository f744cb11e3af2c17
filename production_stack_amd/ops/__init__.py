"""Op dispatch layer.

On a GPU (any tensor on `cuda`), every op REQUIRES the in-tree HIP extension
production_stack_amd._C and raises if it is missing -- there is no silent
eager fallback on MI355X. On CPU tensors the pure-PyTorch references run so
the engine and its tests work in GPU-less CI.
"""

from __future__ import annotations

import os

import torch

from production_stack_amd.ops import reference

try:
    from production_stack_amd import _C  # type: ignore[attr-defined]
except ImportError:  # pragma: no cover - exercised only when not built
    _C = None

HAVE_EXT = _C is not None


def _require_ext() -> None:
    if _C is None:
        raise RuntimeError(
            "production_stack_amd._C HIP extension is not built; run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
            "Refusing to fall back to eager PyTorch on GPU."
        )


def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        _C.rms_norm(out, x, w, eps)
        return out
    return reference.rms_norm(x, w, eps)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """In-place on GPU: x <- normed(residual + x); residual <- residual + x."""
    if x.is_cuda:
        _require_ext()
        _C.fused_add_rms_norm(x, residual, w, eps)
        return x, residual
    normed, summed = reference.fused_add_rms_norm(x, residual, w, eps)
    return normed, summed


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        d = x.shape[-1] // 2
        out = torch.empty(x.shape[:-1] + (d,), dtype=x.dtype, device=x.device)
        _C.silu_and_mul(out, x)
        return out
    return reference.silu_and_mul(x)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    cos_sin: torch.Tensor,
    head_dim: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """In-place on GPU. positions int32, q/k bf16 [T, H*HD], cos_sin fp32."""
    if q.is_cuda:
        _require_ext()
        _C.rotary_embedding(positions, q, k, cos_sin, head_dim)
        return q, k
    return reference.rotary_embedding(positions, q, k, cos_sin, head_dim)


def fused_rope_cache(
    qkv: torch.Tensor,
    positions: torch.Tensor,
    cos_sin: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    q_heads: int,
    head_dim: int,
) -> None:
    """Fused in-place RoPE on packed qkv + paged KV append (GPU only)."""
    _require_ext()
    _C.fused_rope_cache(
        qkv, positions, cos_sin, slot_mapping, k_cache, v_cache, q_heads,
        head_dim,
    )


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if k_cache.is_cuda:
        _require_ext()
        _C.reshape_and_cache(
            k.view(k.shape[0], -1), v.view(v.shape[0], -1), k_cache, v_cache,
            slot_mapping,
        )
        return
    reference.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def paged_attn_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    if q.is_cuda:
        _require_ext()
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        # variant 1 = low-register packed-q kernel (3 waves/SIMD): measured
        # +17-26% over variant 0 across batch/ctx (profiles/attn_bench).
        _C.paged_attn_decode(
            out, q, k_cache, v_cache, block_tables, seq_lens, scale, 0, 1,
            window
        )
        return out
    return reference.paged_attn_decode(
        q, k_cache, v_cache, block_tables, seq_lens, scale, window
    )


def paged_attn_prefill(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    token_seq: torch.Tensor,
    token_pos: torch.Tensor,
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    if q.is_cuda:
        _require_ext()
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _C.paged_attn_prefill(
            out, q, k_cache, v_cache, block_tables, token_seq, token_pos,
            scale, window
        )
        return out
    return reference.paged_attn_prefill(
        q, k_cache, v_cache, block_tables, token_seq, token_pos, scale,
        window
    )


# Default MFMA prefill variant: 5 = 8-wave 32x32 swapped-QK^T kernel
# (csrc/prefill_mfma32.hip); 3/4 = the 4-wave 16x16 kernel (64-row
# tiles). Overridable for A/B via PS_PREFILL_VARIANT.
PREFILL_VARIANT = int(os.environ.get("PS_PREFILL_VARIANT", "5"))


def prefill_tile_rows(num_q_heads: int, num_kv_heads: int) -> int:
    """Prefill tile row count for the active kernel. v5 splits its 8 waves
    into GQW head-columns x row-halves (GQW = largest pow2 divisor of the
    GQA ratio, <=4), so tiles carry 32*(8/GQW) rows (64..256)."""
    if PREFILL_VARIANT != 5:
        return 64
    gq = max(1, num_q_heads // max(1, num_kv_heads))
    gqw = 1
    while gqw < 4 and gq % (gqw * 2) == 0:
        gqw *= 2
    return 32 * (8 // gqw)


def paged_attn_prefill_mfma(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    tile_info: torch.Tensor,
    scale: float,
    window: int = 0,
    variant: int = None,
) -> torch.Tensor:
    """MFMA-tiled chunked prefill (GPU, head_dim 128 only). tile_info rows
    must be built with n_rows <= prefill_tile_rows() for the active
    variant."""
    _require_ext()
    out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
    # v5's split-KV (up to 8 kv-splits) keeps CUs filled at small
    # serving launches, where it now beats v3 at the canonical shapes
    # (12-seq x 200-row continuation at hist 4600: 305 vs 263 TF; fresh
    # 900-row prefill: 400 vs 306) as well as at large ones — so v5 is
    # the default everywhere (PS_PREFILL_VARIANT=3 restores the 4-wave
    # kernel for A/B).
    v = PREFILL_VARIANT if variant is None else variant
    _C.paged_attn_prefill_mfma(
        out, q, k_cache, v_cache, block_tables, tile_info, scale, v, window
    )
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        _require_ext()
        if logits.dtype != torch.bfloat16:  # penalty-adjusted fp32 rows
            return logits.float().argmax(dim=-1)
        out = torch.empty(
            logits.shape[0], dtype=torch.long, device=logits.device
        )
        _C.greedy_sample(out, logits)
        return out
    return reference.greedy_sample(logits)


def kv_quant(x: torch.Tensor) -> tuple:
    """Row-wise int8 KV quantization (CacheGen-style serde)."""
    if x.is_cuda:
        _require_ext()
        hd = x.shape[-1]
        out = torch.empty(x.shape, dtype=torch.int8, device=x.device)
        scales = torch.empty(
            x.numel() // hd, dtype=torch.float32, device=x.device
        )
        _C.kv_quant(out, scales, x)
        return out, scales
    return reference.kv_quant(x)


def kv_dequant(
    q: torch.Tensor, scales: torch.Tensor, dtype=torch.bfloat16
) -> torch.Tensor:
    if q.is_cuda:
        _require_ext()
        out = torch.empty(q.shape, dtype=dtype, device=q.device)
        _C.kv_dequant(out, q, scales)
        return out
    return reference.kv_dequant(q, scales, dtype)


def skinny_gemm(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """out = x @ w.T for decode-shaped M<=128 (weight-stream-bound GEMMs
    where hipBLASLt tiles poorly — see csrc/skinny_gemm.hip). GPU only."""
    _require_ext()
    out = torch.empty(
        (x.shape[0], w.shape[0]), dtype=torch.bfloat16, device=x.device
    )
    _C.skinny_gemm(out, x, w)  # split-K partials + combine (no atomics)
    return out


def cachegen_encode(q: torch.Tensor) -> torch.Tensor:
    """Entropy-encode int8 KV (CPU adaptive range coder, per-channel
    contexts). Falls back to a pure-Python reference when the extension
    is unavailable (CPU CI)."""
    if _C is not None and hasattr(_C, "cachegen_encode"):
        return _C.cachegen_encode(q.contiguous().cpu())
    return reference.cachegen_encode(q)


def cachegen_decode(blob: torch.Tensor, hd: int) -> torch.Tensor:
    if _C is not None and hasattr(_C, "cachegen_decode"):
        return _C.cachegen_decode(blob.contiguous().cpu(), hd)
    return reference.cachegen_decode(blob, hd)


def gemm8p(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """out = x @ w.T via the 8-phase deep-pipelined MFMA GEMM
    (csrc/gemm8p.hip; 256x256 tiles + split-K for skinny M). GPU only."""
    _require_ext()
    out = torch.empty(
        (x.shape[0], w.shape[0]), dtype=torch.bfloat16, device=x.device
    )
    _C.gemm8p(out, x, w)
    return out


def lora_bgmv(
    out: torch.Tensor,
    x: torch.Tensor,
    A: torch.Tensor,
    B: torch.Tensor,
    scale: torch.Tensor,
    idx: torch.Tensor,
    col_off: int = 0,
) -> None:
    """out[t, col_off:+W] += scale[idx[t]] * B[idx[t]] @ (A[idx[t]] @ x[t]);
    idx[t] < 0 skips the row. hipGraph-safe (static shapes, no host sync)."""
    if out.is_cuda:
        _require_ext()
        _C.lora_bgmv(out, x, A, B, scale, idx, col_off)
        return
    reference.lora_bgmv(out, x, A, B, scale, idx, col_off)


def fp8_linear(
    x: torch.Tensor, w_q: torch.Tensor, w_scale: torch.Tensor
) -> torch.Tensor:
    """out = x @ (w_q * w_scale)^T with fp8 tensor cores.

    Weights are stored OCP fp8-e4m3 with a per-tensor scale; activations
    quantize dynamically per call (per-tensor amax / 448). On gfx950 this
    runs through hipBLASLt's fp8 MFMA path (torch._scaled_mm) at ~1.8x
    the bf16 GEMM rate for decode shapes. CPU fallback dequantizes.
    """
    if x.is_cuda:
        sx = (x.abs().amax().clamp(min=1e-6) / 448.0).to(torch.float32)
        xq = (x / sx).to(torch.float8_e4m3fn)
        if w_scale.dim() == 1:  # per-channel weight scales
            return fp8_linear_rowwise(
                xq, sx.expand(x.shape[0]), w_q, w_scale)
        return torch._scaled_mm(
            xq, w_q.t(), scale_a=sx, scale_b=w_scale,
            out_dtype=torch.bfloat16,
        )
    wf = w_q.float() * (
        w_scale.float()[:, None] if w_scale.dim() == 1
        else w_scale.float()
    )
    return torch.nn.functional.linear(x.float(), wf).to(x.dtype)


def fp8_quantize_weight(w: torch.Tensor):
    """Per-tensor weight quantization: returns (w_q fp8, scale f32)."""
    scale = (w.abs().amax().float().clamp(min=1e-6) / 448.0)
    w_q = (w.float() / scale).clamp(-448, 448).to(torch.float8_e4m3fn)
    return w_q, scale


def fp8_quantize_weight_rowwise(w: torch.Tensor):
    """Per-output-channel weight quantization: (w_q fp8 [N,K], scales
    [N] f32) — finer than per-tensor, same GEMM cost with rowwise-scaled
    fp8 matmul."""
    scales = (w.abs().amax(dim=1).float().clamp(min=1e-6) / 448.0)
    w_q = (w.float() / scales[:, None]).clamp(-448, 448).to(
        torch.float8_e4m3fn)
    return w_q, scales


def rms_norm_fp8(x, w, eps, residual=None):
    """Fused (add-)RMSNorm emitting fp8-e4m3 rows + per-row scales. When
    `residual` is given it is updated in place (residual += x) like
    fused_add_rms_norm. GPU only."""
    _require_ext()
    T, D = x.shape
    out_q = torch.empty((T, D), dtype=torch.float8_e4m3fn, device=x.device)
    scales = torch.empty(T, dtype=torch.float32, device=x.device)
    _C.rms_norm_fp8(out_q, scales, x,
                    residual if residual is not None else x, w, eps,
                    1 if residual is not None else 0)
    return out_q, scales


def silu_and_mul_fp8(x):
    _require_ext()
    T, D2 = x.shape
    out_q = torch.empty((T, D2 // 2), dtype=torch.float8_e4m3fn,
                        device=x.device)
    scales = torch.empty(T, dtype=torch.float32, device=x.device)
    _C.silu_and_mul_fp8(out_q, scales, x)
    return out_q, scales


_ROWWISE_SCALED_MM = None  # probed once: does _scaled_mm accept [M,1]x[1,N]?


def fp8_rowwise_supported(device="cuda") -> bool:
    """Whether this build's _scaled_mm takes rowwise scales. When it does
    not, the fused per-row act-quant path would need a slow fp32-out
    fallback, so callers keep the per-tensor fp8 pipeline instead."""
    global _ROWWISE_SCALED_MM
    if _ROWWISE_SCALED_MM is None:
        try:
            torch._scaled_mm(
                torch.zeros(16, 32, dtype=torch.float8_e4m3fn,
                            device=device),
                torch.zeros(32, 16, dtype=torch.float8_e4m3fn,
                            device=device),
                scale_a=torch.ones(16, 1, device=device),
                scale_b=torch.ones(1, 16, device=device),
                out_dtype=torch.bfloat16,
            )
            _ROWWISE_SCALED_MM = True
        except (RuntimeError, TypeError):
            _ROWWISE_SCALED_MM = False
    return _ROWWISE_SCALED_MM


def fp8_linear_rowwise(xq, sx, w_q, sw):
    """out[M,N] = (xq*sx[:,None]) @ (w_q*sw[None,:]).T in fp8 tensor
    cores with per-row activation + per-channel weight scales. Uses
    _scaled_mm's rowwise scaling when this build supports it, else a
    unit-scale fp8 GEMM with the outer-product scale applied after."""
    if fp8_rowwise_supported(xq.device):
        return torch._scaled_mm(
            xq, w_q.t(), scale_a=sx[:, None].contiguous(),
            scale_b=sw[None, :].contiguous(), out_dtype=torch.bfloat16,
        )
    one = torch.ones((), dtype=torch.float32, device=xq.device)
    out = torch._scaled_mm(xq, w_q.t(), scale_a=one, scale_b=one,
                           out_dtype=torch.float32)
    return (out * sx[:, None] * sw[None, :]).to(torch.bfloat16)
