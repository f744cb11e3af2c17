#!/usr/bin/env python3
"""Standalone prefill/decode attention kernel benchmark on MI355X.

Reports achieved TFLOP/s for the MFMA chunked-prefill kernel and effective
HBM GB/s for decode, at Llama-3-8B shapes.
"""
import sys
import time

import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
import torch

from production_stack_amd import _C


def bench_prefill(ctx_len=4096, qh=32, kh=8, hd=128, iters=20, variant=4):
    from production_stack_amd import ops
    bs = 16
    tile = ops.prefill_tile_rows(qh, kh) if variant == 5 else 64
    nblocks = ctx_len // bs + 1
    k = torch.randn(nblocks + 1, kh, bs, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn_like(k)
    bt = torch.arange(1, nblocks + 1, dtype=torch.int32, device="cuda").reshape(1, -1)
    T = ctx_len
    q = torch.randn(T, qh, hd, dtype=torch.bfloat16, device="cuda")
    tiles = []
    for t0 in range(0, T, tile):
        tiles.append([0, t0, t0, min(tile, T - t0)])
    tiles = torch.tensor(tiles, dtype=torch.int32, device="cuda")
    out = torch.empty_like(q)
    scale = hd ** -0.5
    for _ in range(3):
        _C.paged_attn_prefill_mfma(out, q, k, v, bt, tiles, scale, variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _C.paged_attn_prefill_mfma(out, q, k, v, bt, tiles, scale, variant)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # causal flops: sum over q of ctx(q) = T*(T+1)/2 per head pair (QK+PV)
    flops = 4 * hd * qh * (T * (T + 1) / 2)
    print(f"prefill v{variant} ctx={ctx_len}: {dt*1e3:.2f} ms  "
          f"{flops/dt/1e12:.1f} TF")


def bench_decode(batch=64, ctx=1024, qh=32, kh=8, hd=128, iters=50,
                 variant=0, fp8=False):
    bs = 16
    per = ctx // bs
    nb = batch * per + 1
    k = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn_like(k)
    if fp8:
        k = k.to(torch.float8_e4m3fn)
        v = v.to(torch.float8_e4m3fn)
    bt = torch.arange(1, batch * per + 1, dtype=torch.int32, device="cuda").reshape(batch, per)
    sl = torch.full((batch,), ctx, dtype=torch.int32, device="cuda")
    q = torch.randn(batch, qh, hd, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    scale = hd ** -0.5
    for _ in range(3):
        _C.paged_attn_decode(out, q, k, v, bt, sl, scale, 0, variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _C.paged_attn_decode(out, q, k, v, bt, sl, scale, 0, variant)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    eb = 1 if fp8 else 2
    bytes_kv = batch * ctx * kh * hd * 2 * eb
    print(f"decode v{variant}{' fp8' if fp8 else ''} b={batch} ctx={ctx}: "
          f"{dt*1e6:.1f} us  {bytes_kv/dt/1e9:.0f} GB/s")


if __name__ == "__main__":
    for v in (3, 4):
        for ctx in (1024, 2048, 4096):
            bench_prefill(ctx, variant=v)
    for v in (0, 1):
        for b in (16, 64, 256):
            bench_decode(batch=b, variant=v)
        for b in (64, 256):
            bench_decode(batch=b, ctx=2048, variant=v)
    for b in (64, 256):
        bench_decode(batch=b, ctx=2048, variant=1, fp8=True)


def bench_prefill_serving(nseq=10, hist=2000, rows=200, qh=32, kh=8, hd=128,
                          iters=20, variant=5):
    """Serving-shaped chunked continuation: nseq sequences each prefilling
    `rows` new tokens on top of `hist` cached tokens (multi-round-QA shape)."""
    from production_stack_amd import ops
    bs = 16
    tile = ops.prefill_tile_rows(qh, kh) if variant == 5 else 64
    ctx = hist + rows
    per = (ctx + bs - 1) // bs
    nb = nseq * per + 1
    k = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn_like(k)
    bt = torch.arange(1, nseq * per + 1, dtype=torch.int32,
                      device="cuda").reshape(nseq, per)
    T = nseq * rows
    q = torch.randn(T, qh, hd, dtype=torch.bfloat16, device="cuda")
    tiles = []
    flat = 0
    for s in range(nseq):
        for t0 in range(0, rows, tile):
            tiles.append([s, flat + t0, hist + t0, min(tile, rows - t0)])
        flat += rows
    tiles = torch.tensor(tiles, dtype=torch.int32, device="cuda")
    out = torch.empty_like(q)
    scale = hd ** -0.5
    for _ in range(3):
        _C.paged_attn_prefill_mfma(out, q, k, v, bt, tiles, scale, variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _C.paged_attn_prefill_mfma(out, q, k, v, bt, tiles, scale, variant)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # flops: per row, attended kv = hist + row_idx + 1
    att = sum(hist + r + 1 for r in range(rows)) * nseq
    flops = 4 * hd * qh * att
    print(f"serving v{variant} nseq={nseq} hist={hist} rows={rows}: "
          f"{dt*1e3:.3f} ms  {flops/dt/1e12:.1f} TF")
