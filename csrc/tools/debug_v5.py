#!/usr/bin/env python3
"""Debug harness for prefill v5: tiny structured cases that isolate which
stage (QK, softmax/mask, PV, epilogue) is wrong, printing error structure."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch

from production_stack_amd import ops
from production_stack_amd.ops import reference


def run_case(name, T, qh=1, kh=1, hd=128, kzero=False, vones=False,
             start=0, seed=0):
    torch.manual_seed(seed)
    bs = 16
    ctx = start + T
    nb = (ctx + bs - 1) // bs + 1
    k = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16, device="cuda") / 4
    v = torch.randn_like(k) / 4
    if kzero:
        k.zero_()
    if vones:
        v.fill_(1.0)
    bt = torch.arange(1, nb, dtype=torch.int32, device="cuda").view(1, -1)
    q = torch.randn(T, qh, hd, dtype=torch.bfloat16, device="cuda") / 4
    tiles = []
    TR = ops.prefill_tile_rows(qh, kh)
    for t0 in range(0, T, TR):
        tiles.append([0, t0, start + t0, min(TR, T - t0)])
    tiles = torch.tensor(tiles, dtype=torch.int32, device="cuda")
    token_seq = torch.zeros(T, dtype=torch.int32)
    token_pos = torch.arange(start, start + T, dtype=torch.int32)
    scale = hd ** -0.5
    want = reference.paged_attn_prefill(
        q.cpu(), k.cpu(), v.cpu(), bt.cpu(), token_seq, token_pos, scale)
    got = ops.paged_attn_prefill_mfma(q, k, v, bt, tiles, scale,
                                      variant=5).cpu()
    err = (got.float() - want.float()).abs()
    bad = err > 0.02 + 0.02 * want.float().abs()
    nbad = int(bad.sum())
    print(f"{name}: T={T} qh={qh} kh={kh} start={start} "
          f"maxerr={err.max():.4f} bad={nbad}/{err.numel()}")
    if nbad:
        # error structure: which rows/dims are bad
        rows = bad.any(dim=2).any(dim=1).nonzero().flatten().tolist()
        dims = bad.any(dim=0).any(dim=0).nonzero().flatten().tolist()
        print(f"  bad rows: {rows[:16]}{'...' if len(rows) > 16 else ''} "
              f"({len(rows)} rows)")
        print(f"  bad dims: {dims[:16]}{'...' if len(dims) > 16 else ''} "
              f"({len(dims)} dims)")
        r = rows[0]
        print(f"  row {r} got : {got[r, 0, :8].tolist()}")
        print(f"  row {r} want: {want[r, 0, :8].tolist()}")
    return nbad == 0


ok = True
ok &= run_case("A row1", 1)                    # single row, 1 chunk
ok &= run_case("B rows32", 32)                 # one wave fully
ok &= run_case("C rows64", 64)                 # chunk=rows
ok &= run_case("D rows256", 256)               # full tile
ok &= run_case("E rows300", 300)               # 2 tiles, ragged
ok &= run_case("F cont", 64, start=128)        # continuation (history)
ok &= run_case("G kzero", 128, kzero=True)     # S=0: uniform softmax -> PV
ok &= run_case("H vones", 128, vones=True)     # O must be exactly 1
ok &= run_case("I gqa", 128, qh=4, kh=2)       # GQA mapping
ok &= run_case("J long", 1024)                 # many chunks, defer-max
print("ALL OK" if ok else "FAILURES")
