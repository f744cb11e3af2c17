#!/usr/bin/env python3
"""v5 vs v3 and vs reference: error structure on the exact failing configs."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch

from production_stack_amd import ops
from production_stack_amd.ops import reference


def build_tiles(chunks, tile):
    tiles, token_seq, token_pos = [], [], []
    flat = 0
    for row, (r, start, n) in enumerate(chunks):
        for t0 in range(0, n, tile):
            tiles.append([r, flat + t0, start + t0, min(tile, n - t0)])
        for p in range(start, start + n):
            token_seq.append(r)
            token_pos.append(p)
        flat += n
    return (torch.tensor(tiles, dtype=torch.int32),
            torch.tensor(token_seq, dtype=torch.int32),
            torch.tensor(token_pos, dtype=torch.int32))


def make_cache(nb, kh, bs, hd):
    k = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16, device="cuda") / 4
    v = torch.randn_like(k) / 4
    return k, v


def analyze(name, got, want, tq=None):
    err = (got.float().cpu() - want.float().cpu()).abs()
    rel = err / (want.float().cpu().abs() + 2e-2)
    bad = err > 2e-2 + 2e-2 * want.float().cpu().abs()
    print(f"{name}: maxerr={err.max():.4f} bad={int(bad.sum())}/{err.numel()}")
    if bad.any():
        rows = bad.any(dim=2).any(dim=1).nonzero().flatten().tolist()
        heads = bad.any(dim=2).any(dim=0).nonzero().flatten().tolist()
        print(f"  bad flat-rows ({len(rows)}): {rows[:24]}")
        print(f"  bad heads ({len(heads)}): {heads[:16]}")
        if tq is not None:
            print(f"  bad row pos: {[int(tq[r]) for r in rows[:24]]}")


# exact config of test_prefill_mfma32_v5_matches_v3
torch.manual_seed(41)
qh, kh, hd = 32, 8, 128
chunks = [(0, 2048, 1024)]
max_blocks = 192
k_cache, v_cache = make_cache(max_blocks + 1, kh, 16, hd)
bt = torch.arange(1, max_blocks + 1, dtype=torch.int32).reshape(1, -1)
t5, _, tpos = build_tiles(chunks, 64)
t3, _, _ = build_tiles(chunks, 64)
q = torch.randn((1024, qh, hd), dtype=torch.bfloat16, device="cuda")
a = ops.paged_attn_prefill_mfma(q, k_cache, v_cache, bt.cuda(), t5.cuda(),
                                0.0883883, variant=5)
b = ops.paged_attn_prefill_mfma(q, k_cache, v_cache, bt.cuda(), t3.cuda(),
                                0.0883883, variant=3)
analyze("v5 vs v3 (ctx 2048+1024)", a, b, tpos)
w = reference.paged_attn_prefill(
    q.cpu(), k_cache.cpu(), v_cache.cpu(), bt, torch.zeros(1024, dtype=torch.int32),
    tpos, 0.0883883)
analyze("v5 vs ref", a, w, tpos)
analyze("v3 vs ref", b, w, tpos)

# windowed config
torch.manual_seed(43)
W = 160
ctx = 640
nb = ctx // 16 + 1
k2 = torch.randn(nb, 2, 16, hd, dtype=torch.bfloat16, device="cuda") / 4
v2 = torch.randn_like(k2) / 4
bt2 = torch.arange(1, nb, dtype=torch.int32, device="cuda").view(1, -1)
q2 = torch.randn(ctx, 8, hd, dtype=torch.bfloat16, device="cuda") / 4
t5w, _, _ = build_tiles([(0, 0, ctx)], 64)
gw = ops.paged_attn_prefill_mfma(q2, k2, v2, bt2, t5w.cuda(), 0.0883,
                                 window=W, variant=5)
ww = reference.paged_attn_prefill(
    q2.cpu(), k2.cpu(), v2.cpu(), bt2.cpu(), torch.zeros(ctx, dtype=torch.int32),
    torch.arange(ctx, dtype=torch.int32), 0.0883, window=W)
analyze("v5 windowed vs ref", gw, ww, torch.arange(ctx))
