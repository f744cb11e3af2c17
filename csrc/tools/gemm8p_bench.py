#!/usr/bin/env python3
"""Refcheck + A/B bench of the 8-phase GEMM vs hipBLASLt (F.linear) at the
Llama-3-8B serving shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch
import torch.nn.functional as F

from production_stack_amd import ops

SHAPES = [(6144, 4096, "qkv"), (4096, 4096, "o"),
          (28672, 4096, "gate_up"), (4096, 14336, "down")]


def refcheck():
    torch.manual_seed(3)
    ok = True
    for (M, N, K) in [(256, 256, 128), (64, 256, 256), (384, 512, 384),
                      (300, 768, 640), (1024, 1536, 896), (1, 256, 128)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") / 4
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") / 4
        got = ops.gemm8p(x, w).float().cpu()
        want = F.linear(x, w).float().cpu()
        err = (got - want).abs()
        tol = 2e-2 + 2e-2 * want.abs()
        bad = int((err > tol).sum())
        print(f"refcheck M={M} N={N} K={K}: maxerr={err.max():.4f} "
              f"bad={bad}/{err.numel()} {'OK' if not bad else 'FAIL'}")
        ok &= bad == 0
    # race screen: repeated runs must be bit-identical
    x = torch.randn(384, 4096, dtype=torch.bfloat16, device="cuda") / 4
    w = torch.randn(6144, 4096, dtype=torch.bfloat16, device="cuda") / 4
    first = ops.gemm8p(x, w)
    stable = all(torch.equal(ops.gemm8p(x, w), first) for _ in range(20))
    print(f"race screen (20 runs bit-identical): "
          f"{'OK' if stable else 'FAIL'}")
    return ok and stable


def bench(M, iters=100):
    for (N, K, name) in SHAPES:
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        res = {}
        for label, fn in [("blas", lambda: F.linear(x, w)),
                          ("8p", lambda: ops.gemm8p(x, w))]:
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                fn()
            torch.cuda.synchronize()
            res[label] = (time.perf_counter() - t0) / iters
        tf = 2 * M * N * K
        print(f"M={M:5d} {name:8s}: blas {res['blas']*1e6:7.1f}us "
              f"({tf/res['blas']/1e12:6.1f} TF)  8p {res['8p']*1e6:7.1f}us "
              f"({tf/res['8p']/1e12:6.1f} TF)  "
              f"{'8P WINS' if res['8p'] < res['blas'] else ''}")


if __name__ == "__main__":
    if refcheck():
        for M in (256, 384, 512, 1024, 1664, 2048):
            bench(M)
    else:
        print("REFCHECK FAILED — skipping bench")
