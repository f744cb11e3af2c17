#!/usr/bin/env python3
"""Bisect v5 NaN failure: exact failing pytest config, then vary one axis
at a time (tile size, heads, chunks, scale)."""
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


def case(name, chunks, qh, kh, tile, nseq, scale_div=1, seed=4):
    torch.manual_seed(seed)
    hd, bs = 128, 16
    max_blocks = 32
    nb = nseq * max_blocks + 1
    k_cache = torch.randn((nb, kh, bs, hd), dtype=torch.bfloat16,
                          device="cuda") / scale_div
    v_cache = torch.randn((nb, kh, bs, hd), dtype=torch.bfloat16,
                          device="cuda") / scale_div
    block_tables = torch.arange(
        1, nseq * max_blocks + 1, dtype=torch.int32).reshape(nseq, max_blocks)
    tiles, token_seq, token_pos = build_tiles(chunks, tile)
    T = token_seq.shape[0]
    q = torch.randn((T, qh, hd), dtype=torch.bfloat16,
                    device="cuda") / scale_div
    want = reference.paged_attn_prefill(
        q.cpu(), k_cache.cpu(), v_cache.cpu(), block_tables, token_seq,
        token_pos, 1.0 / hd ** 0.5)
    got = ops.paged_attn_prefill_mfma(
        q, k_cache, v_cache, block_tables.cuda(), tiles.cuda(),
        1.0 / hd ** 0.5, variant=5).cpu().float()
    w = want.float()
    nanmask = got.isnan()
    bad = ((got - w).abs() > 2e-2 + 2e-2 * w.abs()) | nanmask
    print(f"{name}: bad={int(bad.sum())}/{bad.numel()} "
          f"nan={int(nanmask.sum())}")
    if bad.any():
        rows = bad.any(dim=2).any(dim=1).nonzero().flatten().tolist()
        nrows = nanmask.any(dim=2).any(dim=1).nonzero().flatten().tolist()
        heads = bad.any(dim=2).any(dim=0).nonzero().flatten().tolist()
        print(f"  bad rows({len(rows)}): {rows[:20]}")
        print(f"  nan rows({len(nrows)}): {nrows[:20]}")
        print(f"  bad heads({len(heads)}): {heads[:12]}")


FULL = [(0, 0, 200), (1, 128, 100), (2, 0, 1), (3, 31, 64)]
case("exact failing", FULL, 32, 8, 64, 4)
case("tile=128gq1", FULL, 8, 8, 128, 4)
case("qh=kh=8", FULL, 8, 8, 64, 4)
case("qh=kh=1", FULL, 1, 1, 64, 4)
case("seq0 only", [(0, 0, 200)], 32, 8, 64, 1)
case("seq3 only", [(0, 31, 64)], 32, 8, 64, 1)
case("seq1 only", [(0, 128, 100)], 32, 8, 64, 1)
case("seq0 qh1", [(0, 0, 200)], 1, 1, 64, 1)
case("seq0 div4", [(0, 0, 200)], 32, 8, 64, 1, scale_div=4)
case("seq0 t256gq1", [(0, 0, 200)], 8, 8, 256, 1)
