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

# (M, N, K) -> True if the skinny kernel won
_POLICY: Dict[Tuple[int, int, int], bool] = {}


def _eligible(M: int, N: int, K: int) -> bool:
    return M <= 256 and N % 64 == 0 and K % 64 == 0


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
            if not _eligible(M, N, K):
                _POLICY[(M, N, K)] = False
                continue
            w = w_by_shape[(N, K)]
            x = torch.randn(M, K, dtype=torch.bfloat16, device=device)
            t_blas = _time_fn(lambda: F.linear(x, w))
            t_skinny = _time_fn(lambda: ops.skinny_gemm(x, w))
            _POLICY[(M, N, K)] = t_skinny < t_blas
            if t_skinny < t_blas:
                logger.info(
                    "gemm M=%d N=%d K=%d: skinny %.1fus < blas %.1fus",
                    M, N, K, t_skinny * 1e6, t_blas * 1e6,
                )
    logger.info(
        "gemm autotune done: %d/%d cells use the skinny kernel",
        sum(_POLICY.values()), len(_POLICY),
    )


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """F.linear with autotuned dispatch (no bias; Llama has none)."""
    if x.is_cuda and _POLICY:
        key = (x.shape[0], w.shape[0], w.shape[1])
        if _POLICY.get(key):
            from production_stack_amd import ops

            return ops.skinny_gemm(x, w)
    return F.linear(x, w)


def reset() -> None:
    _POLICY.clear()
