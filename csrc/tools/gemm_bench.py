#!/usr/bin/env python3
"""Decode-batch GEMM efficiency probe (hipBLASLt via torch; optionally
TunableOp when PYTORCH_TUNABLEOP_ENABLED=1)."""
import os
import time

import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
import torch

SHAPES = [  # (out_features, in_features) for Llama-3-8B layer GEMMs
    (6144, 4096, "qkv"),
    (4096, 4096, "o"),
    (28672, 4096, "gate_up"),
    (4096, 14336, "down"),
]


def bench(M, iters=200):
    total = 0.0
    for (N, K, name) in SHAPES:
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        for _ in range(10):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        gb = N * K * 2 / 1e9
        total += dt
        print(f"M={M:4d} {name:8s}: {dt*1e6:7.1f} us  {gb/dt:6.0f} GB/s")
    print(f"M={M:4d} layer total: {total*1e6:.1f} us "
          f"(weights roofline ~{(sum(n*k for n,k,_ in SHAPES)*2/6.3e12)*1e6:.1f} us)")


def bench_skinny(M, iters=200):
    from production_stack_amd import ops

    for (N, K, name) in SHAPES:
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        for _ in range(10):
            ops.skinny_gemm(x, w)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            ops.skinny_gemm(x, w)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        gb = N * K * 2 / 1e9
        print(f"skinny M={M:4d} {name:8s}: {dt*1e6:7.1f} us  {gb/dt:6.0f} GB/s")


if __name__ == "__main__":
    print("TunableOp:", os.environ.get("PYTORCH_TUNABLEOP_ENABLED", "0"))
    for M in (16, 32, 64, 128, 256):
        bench(M)
    for M in (16, 32, 64, 128, 256):
        bench_skinny(M)
