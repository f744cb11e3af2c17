"""Per-shape GEMM dispatch: hipBLASLt (torch F.linear) vs the in-house
skinny split-K MFMA kernel.

hipBLASLt's decode-shaped tiles stream weights at 0.8-2 TB/s for several
(M<=128, N, K) projections while csrc/skinny_gemm.hip reaches 2.4-5 TB/s at
M<=32 (profiles/gemm_skinny.log) — but loses at larger M. Instead of a
hardcoded table, the engine autotunes each (decode bucket M, projection
shape) pair once at startup on the actual device and the winner is what the
hipGraph capture records.
"""

from __future__ import annotations

import logging
import time
from typing import Dict, Tuple

import torch
import torch.nn.functional as F

logger = logging.getLogger("ops.gemm_policy")

# (M, N, K) -> winning impl: 0 = hipBLASLt, 1 = skinny split-K, 2 = 8-phase
_POLICY: Dict[Tuple[int, int, int], int] = {}


def _eligible(M: int, N: int, K: int) -> bool:
    return M <= 256 and N % 64 == 0 and K % 64 == 0


def _eligible_8p(M: int, N: int, K: int) -> bool:
    return N % 256 == 0 and K % 128 == 0


def _time_fn(fn, iters: int = 20) -> float:
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def tune(weights, buckets, device) -> None:
    """weights: iterable of weight tensors [N, K]; buckets: decode batch
    sizes to tune for."""
    from production_stack_amd import ops

    shapes = sorted({(int(w.shape[0]), int(w.shape[1])) for w in weights})
    w_by_shape = {}
    for w in weights:
        w_by_shape.setdefault((int(w.shape[0]), int(w.shape[1])), w)
    for M in buckets:
        for (N, K) in shapes:
            w = w_by_shape[(N, K)]
            x = torch.randn(M, K, dtype=torch.bfloat16, device=device)
            cands = [(0, _time_fn(lambda: F.linear(x, w)))]
            if _eligible(M, N, K):
                cands.append((1, _time_fn(lambda: ops.skinny_gemm(x, w))))
            if _eligible_8p(M, N, K):
                cands.append((2, _time_fn(lambda: ops.gemm8p(x, w))))
            impl, t = min(cands, key=lambda c: c[1])
            _POLICY[(M, N, K)] = impl
            if impl:
                logger.info(
                    "gemm M=%d N=%d K=%d: impl %s wins at %.1fus",
                    M, N, K, ("blas", "skinny", "8p")[impl], t * 1e6,
                )
    logger.info(
        "gemm autotune done: %d/%d cells use an in-house kernel",
        sum(1 for v in _POLICY.values() if v), len(_POLICY),
    )


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """F.linear with autotuned dispatch (no bias; Llama has none)."""
    if x.is_cuda and _POLICY:
        key = (x.shape[0], w.shape[0], w.shape[1])
        impl = _POLICY.get(key, 0)
        if impl:
            from production_stack_amd import ops

            return (ops.skinny_gemm if impl == 1 else ops.gemm8p)(x, w)
    return F.linear(x, w)


def reset() -> None:
    _POLICY.clear()
