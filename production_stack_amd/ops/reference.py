"""Pure-PyTorch reference implementations of the HIP kernels.

These serve two purposes:
  1. the fp32 ground truth that every GPU kernel unit test compares against
     (tests/test_kernels_gpu.py);
  2. the CPU execution path, so the whole engine runs (slowly) without a GPU
     for the CPU test tier (fake-engine/CPU tier first).

They intentionally mirror the kernel semantics exactly (paged cache layout
[num_blocks, kv_heads, block_size, head_dim], slot mapping, causal masking).
"""

from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * w.float()
    return out.to(x.dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """Returns (normed, new_residual). Matches the in-place kernel: the
    residual accumulator is stored in bf16 (summed in fp32, rounded once)."""
    summed = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(summed, w, eps), summed


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    gate = x[..., :d].float()
    up = x[..., d:].float()
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    cos_sin: torch.Tensor,
    head_dim: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Neox-style rotation of the leading rot_dim dims of each head.

    q: [T, QH*HD] (or [T, QH, HD]); k analogous. cos_sin: [max_pos, rot_dim]
    laid out as cos[rot/2] || sin[rot/2]. Returns rotated copies.
    """
    T = positions.shape[0]
    rot = cos_sin.shape[-1]
    half = rot // 2
    cs = cos_sin[positions.long()]  # [T, rot]
    cos = cs[:, :half].view(T, 1, half)
    sin = cs[:, half:].view(T, 1, half)

    def rotate(t: torch.Tensor) -> torch.Tensor:
        shape = t.shape
        t = t.view(T, -1, head_dim)
        x1 = t[..., :half].float()
        x2 = t[..., half:rot].float()
        r1 = x1 * cos - x2 * sin
        r2 = x2 * cos + x1 * sin
        out = t.clone()
        out[..., :half] = r1.to(t.dtype)
        out[..., half:rot] = r2.to(t.dtype)
        return out.view(shape)

    return rotate(q), rotate(k)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    """k/v: [T, KH, HD]; caches: [NB, KH, BS, HD]; slot_mapping: [T] int64."""
    nb, kh, bs, hd = k_cache.shape
    mask = slot_mapping >= 0
    slots = slot_mapping[mask]
    blk = torch.div(slots, bs, rounding_mode="floor").long()
    off = (slots % bs).long()
    k_cache[blk, :, off] = k.view(-1, kh, hd)[mask].to(k_cache.dtype)
    v_cache[blk, :, off] = v.view(-1, kh, hd)[mask].to(v_cache.dtype)


def _gather_kv(
    cache: torch.Tensor, block_table: torch.Tensor, ctx: int
) -> torch.Tensor:
    """cache: [NB, KH, BS, HD] -> [ctx, KH, HD] for one sequence."""
    nb, kh, bs, hd = cache.shape
    nblocks = (ctx + bs - 1) // bs
    blocks = block_table[:nblocks].long()
    kv = cache[blocks]  # [nblocks, KH, BS, HD]
    kv = kv.permute(0, 2, 1, 3).reshape(nblocks * bs, kh, hd)
    return kv[:ctx]


def paged_attn_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """q: [S, QH, HD] -> out [S, QH, HD] (fp32 math, cast back).
    window > 0 = sliding-window attention over the last `window` keys."""
    S, QH, HD = q.shape
    KH = k_cache.shape[1]
    GQ = QH // KH
    out = torch.empty_like(q)
    for s in range(S):
        ctx = int(seq_lens[s])
        k = _gather_kv(k_cache, block_tables[s], ctx).float()  # [ctx, KH, HD]
        v = _gather_kv(v_cache, block_tables[s], ctx).float()
        if window and ctx > window:
            k = k[ctx - window:]
            v = v[ctx - window:]
        for h in range(QH):
            kvh = h // GQ
            qs = q[s, h].float()
            logits = (k[:, kvh] @ qs) * scale  # [ctx]
            p = torch.softmax(logits, dim=-1)
            out[s, h] = (p @ v[:, kvh]).to(q.dtype)
    return out


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
    """q: [T, QH, HD]; each token t attends to cache[token_seq[t]][0..pos]
    (or the last `window` positions when window > 0)."""
    T, QH, HD = q.shape
    KH = k_cache.shape[1]
    GQ = QH // KH
    out = torch.empty_like(q)
    for t in range(T):
        s = int(token_seq[t])
        ctx = int(token_pos[t]) + 1
        k = _gather_kv(k_cache, block_tables[s], ctx).float()
        v = _gather_kv(v_cache, block_tables[s], ctx).float()
        if window and ctx > window:
            k = k[ctx - window:]
            v = v[ctx - window:]
        for h in range(QH):
            kvh = h // GQ
            qs = q[t, h].float()
            logits = (k[:, kvh] @ qs) * scale
            p = torch.softmax(logits, dim=-1)
            out[t, h] = (p @ v[:, kvh]).to(q.dtype)
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    return logits.float().argmax(dim=-1)


def kv_quant(x: torch.Tensor) -> tuple:
    """Row-wise int8 quantization over the last dim. Returns (int8, scales)."""
    hd = x.shape[-1]
    rows = x.reshape(-1, hd).float()
    amax = rows.abs().amax(dim=-1)
    scales = torch.where(amax > 0, amax / 127.0, torch.ones_like(amax))
    q = torch.clamp(
        torch.round(rows / scales[:, None]), -127, 127
    ).to(torch.int8)
    return q.view(x.shape), scales


def kv_dequant(q: torch.Tensor, scales: torch.Tensor,
               dtype=torch.bfloat16) -> torch.Tensor:
    hd = q.shape[-1]
    rows = q.reshape(-1, hd).float() * scales[:, None]
    return rows.to(dtype).view(q.shape)


def lora_bgmv(out, x, A, B, scale, idx, col_off):
    """Reference BGMV: out[t, col_off:+W] += scale[s] * B[s] @ (A[s] @ x[t])
    for s = idx[t] >= 0."""
    import torch

    W = B.shape[1]
    for t in range(x.shape[0]):
        s = int(idx[t])
        if s < 0:
            continue
        y = (A[s].float() @ x[t].float()) * float(scale[s])
        out[t, col_off : col_off + W] += (B[s].float() @ y).to(out.dtype)
